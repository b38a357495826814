// BatchNorm (K5) + fused activation (K6) + residual (K8) for gfx950.
//
// Design (2nd iteration, sized from rocprofv3 stats on MI355X):
//  - partial kernels fill the chip: grid = c_tiles x chunks with chunks
//    scaled so the launch has ~4096 workgroups (256 CUs x 8 XCDs need
//    well over 256 blocks); per-block mlane partials fold through LDS so
//    the slab count stays = chunks (keeps the combine cheap)
//  - finalize kernels are split-parallel: 256 threads = 16 channels x 16
//    slab lanes, LDS tree over lanes (the v1 one-thread-per-channel
//    serial loop over 4k slabs was 150 us per call)
//  - normalize/act kernels consume precomputed per-channel {a, b}
//    (a = gamma*invstd, b = beta - mean*a) so the hot loop loads 2
//    fp32 arrays, not 4-6
//
// Backward identity (training):
//   v0 = a*z + b ; da = dy * act'(v0)
//   dgamma = sum(da*zhat) ; dbeta = sum(da)
//   dz = a*da + K0 - K2*z,  K2 = a*invstd*dgamma/M,
//                           K0 = -a*dbeta/M + mean*K2
// Eval mode: dz = a*da.
// All reductions deterministic (fixed-order slabs, no atomics).

#include <algorithm>
#include <cstdlib>

#include <hip/hip_cooperative_groups.h>

#include "common.h"
#include "engine.h"

namespace cg = cooperative_groups;

namespace yams {

namespace {

// measurement knob: YAMS_BN_BLOCKS overrides the reduction grid target
// (512 = round-2 same-lease sweep optimum: 13.73 vs 13.84 at 1024,
// 14.15 at 4096; re-sweepable without a rebuild)
inline int bn_target_blocks() {
  static int t = [] {
    const char* e = getenv("YAMS_BN_BLOCKS");
    const int v = e ? atoi(e) : 512;
    return v > 0 ? v : 512;
  }();
  return t;
}

struct RedGeom {
  int CT, ML, c_tiles, chunks;
};

RedGeom red_geom(int64_t M, int C) {
  RedGeom g;
  const int C8 = C >> 3;
  g.CT = C8 < 256 ? C8 : 256;
  g.ML = 256 / g.CT;
  g.c_tiles = (C8 + g.CT - 1) / g.CT;
  int64_t max_chunks = (M + 63) / 64;
  int64_t want = bn_target_blocks() / g.c_tiles;
  g.chunks = int(std::min<int64_t>(std::max<int64_t>(want, 1), max_chunks));
  return g;
}

// --- stats partial: vec8 channels, per-(chunk) slab after mlane fold ---
__global__ __launch_bounds__(256) void bn_stats_partial_kernel(
    const ushort_t* __restrict__ z, float* __restrict__ ws, int64_t M, int C,
    int chunks) {
  __shared__ float red[256 * 8];
  const int C8 = C >> 3;
  const int CT8 = C8 < 256 ? C8 : 256;
  const int ML = 256 / CT8;
  const int c_tiles = (C8 + CT8 - 1) / CT8;
  const int c8 = (blockIdx.x % c_tiles) * CT8 + int(threadIdx.x) % CT8;
  const int chunk = blockIdx.x / c_tiles;
  const int mlane = int(threadIdx.x) / CT8;
  const int c = c8 * 8;
  const int64_t per = (M + chunks - 1) / chunks;
  const int64_t lo = chunk * per;
  const int64_t hi = std::min<int64_t>(lo + per, M);
  float s1[8] = {}, s2[8] = {};
  if (c8 < C8 && mlane < ML) {
    for (int64_t m = lo + mlane; m < hi; m += ML) {
      short8_t v8 = *reinterpret_cast<const short8_t*>(&z[m * C + c]);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float v = bf2f(ushort_t(v8[e]));
        s1[e] += v;
        s2[e] = fmaf(v, v, s2[e]);
      }
    }
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) red[threadIdx.x * 8 + e] = s1[e];
  __syncthreads();
  if (threadIdx.x < unsigned(CT8) && c8 < C8) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float t = 0.f;
      for (int l = 0; l < ML; ++l) t += red[(threadIdx.x + l * CT8) * 8 + e];
      ws[int64_t(chunk) * C + c + e] = t;
    }
  }
  __syncthreads();
#pragma unroll
  for (int e = 0; e < 8; ++e) red[threadIdx.x * 8 + e] = s2[e];
  __syncthreads();
  if (threadIdx.x < unsigned(CT8) && c8 < C8) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float t = 0.f;
      for (int l = 0; l < ML; ++l) t += red[(threadIdx.x + l * CT8) * 8 + e];
      ws[(int64_t(chunks) + chunk) * C + c + e] = t;
    }
  }
}

// --- split-parallel finalize: 16 channels x 16 slab lanes per block -----
__global__ __launch_bounds__(256) void bn_stats_finalize_kernel(
    const float* __restrict__ ws, float* __restrict__ mean,
    float* __restrict__ invstd, float* __restrict__ ab,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    int64_t M, int C, int splits, float momentum, float eps) {
  __shared__ float red[256];
  const int cl = int(threadIdx.x) % 16;          // channel lane
  const int sl = int(threadIdx.x) / 16;          // split lane
  const int c = blockIdx.x * 16 + cl;
  float s1 = 0.f, s2 = 0.f;
  if (c < C) {
    float s1b = 0.f, s2b = 0.f;  // second chain: ILP on the wait-bound
    int sp = sl;                 // serial slab walk
    for (; sp + 16 < splits; sp += 32) {
      s1 += ws[int64_t(sp) * C + c];
      s2 += ws[int64_t(splits + sp) * C + c];
      s1b += ws[int64_t(sp + 16) * C + c];
      s2b += ws[int64_t(splits + sp + 16) * C + c];
    }
    for (; sp < splits; sp += 16) {
      s1 += ws[int64_t(sp) * C + c];
      s2 += ws[int64_t(splits + sp) * C + c];
    }
    s1 += s1b;
    s2 += s2b;
  }
  red[threadIdx.x] = s1;
  __syncthreads();
  if (sl == 0 && c < C)
    for (int l = 1; l < 16; ++l) s1 += red[cl + l * 16];
  __syncthreads();
  red[threadIdx.x] = s2;
  __syncthreads();
  if (sl == 0 && c < C) {
    for (int l = 1; l < 16; ++l) s2 += red[cl + l * 16];
    const float mu = s1 / M;
    float var = s2 / M - mu * mu;
    var = fmaxf(var, 0.f);
    const float inv = rsqrtf(var + eps);
    mean[c] = mu;
    invstd[c] = inv;
    const float a = gamma[c] * inv;
    ab[c] = a;
    ab[C + c] = beta[c] - mu * a;
    if (running_mean != nullptr) {
      const float unbiased = M > 1 ? var * (float(M) / float(M - 1)) : var;
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
  }
}

// --- forward normalize+act(+residual): flat vec8, 2 coef arrays ---------
// (c8, mlane) layout: the per-channel {a, b} pairs load once per thread
// (same restructure as bn_bwd_apply_kernel; the flat version re-read
// 4 float4 of coefficients per 8 elements)
template <int ACT, bool RES>
__global__ __launch_bounds__(256) void bn_act_fwd_kernel(
    const ushort_t* __restrict__ z, const float* __restrict__ ab,
    ushort_t* __restrict__ y, const ushort_t* __restrict__ res,
    int64_t M, int C, int chunks) {
  const int C8 = C >> 3;
  const int CT8 = C8 < 256 ? C8 : 256;
  const int ML = 256 / CT8;
  const int c_tiles = (C8 + CT8 - 1) / CT8;
  const int c8 = (blockIdx.x % c_tiles) * CT8 + int(threadIdx.x) % CT8;
  const int chunk = blockIdx.x / c_tiles;
  const int mlane = int(threadIdx.x) / CT8;
  const int c = c8 * 8;
  if (c8 >= C8 || mlane >= ML) return;
  const int64_t per = (M + chunks - 1) / chunks;
  const int64_t lo = chunk * per;
  const int64_t hi = std::min<int64_t>(lo + per, M);
  float a[8], b[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    a[e] = ab[c + e];
    b[e] = ab[C + c + e];
  }
  for (int64_t m = lo + mlane; m < hi; m += ML) {
    const int64_t off = m * C + c;
    short8_t zv = *reinterpret_cast<const short8_t*>(&z[off]);
    short8_t rv;
    if (RES) rv = *reinterpret_cast<const short8_t*>(&res[off]);
    short8_t out;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = act_fwd<ACT>(fmaf(bf2f(ushort_t(zv[e])), a[e], b[e]));
      if (RES) v += bf2f(ushort_t(rv[e]));
      out[e] = short(f2bf(v));
    }
    *reinterpret_cast<short8_t*>(&y[off]) = out;
  }
}

// --- backward reduce: vec8 channels, per-chunk sum(da), sum(da*z) -------
// ADDPOOL: dy is the SE rowscale grad and the global-avg-pool backward
// contribution (dpooled/HW per image) still has to be added; doing the
// add HERE (writing the sum back to dy) removes the separate
// avgpool_bwd_add pass over the tensor.
template <int ACT, bool ADDPOOL = false>
__global__ __launch_bounds__(256) void bn_bwd_partial_kernel(
    ushort_t* __restrict__ dy, const ushort_t* __restrict__ z,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ ws, int64_t M, int C, int chunks,
    const ushort_t* __restrict__ dpooled = nullptr, int HW = 1,
    float inv_hw = 0.f) {
  __shared__ float red[256 * 8];
  const int C8 = C >> 3;
  const int CT8 = C8 < 256 ? C8 : 256;
  const int ML = 256 / CT8;
  const int c_tiles = (C8 + CT8 - 1) / CT8;
  const int c8 = (blockIdx.x % c_tiles) * CT8 + int(threadIdx.x) % CT8;
  const int chunk = blockIdx.x / c_tiles;
  const int mlane = int(threadIdx.x) / CT8;
  const int c = c8 * 8;
  const int64_t per = (M + chunks - 1) / chunks;
  const int64_t lo = chunk * per;
  const int64_t hi = std::min<int64_t>(lo + per, M);
  float s1[8] __attribute__((aligned(8))) = {};
  float s2[8] __attribute__((aligned(8))) = {};
  if (c8 < C8 && mlane < ML) {
    float a[8], b[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      a[e] = gamma[c + e] * invstd[c + e];
      b[e] = beta[c + e] - mean[c + e] * a[e];
    }
    for (int64_t m = lo + mlane; m < hi; m += ML) {
      short8_t z8 = *reinterpret_cast<const short8_t*>(&z[m * C + c]);
      short8_t g8 = *reinterpret_cast<const short8_t*>(&dy[m * C + c]);
      if (ADDPOOL) {
        const int n = int(m / HW);
        short8_t p8 = *reinterpret_cast<const short8_t*>(
            &dpooled[int64_t(n) * C + c]);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          g8[e] = short(f2bf(bf2f(ushort_t(g8[e])) +
                             bf2f(ushort_t(p8[e])) * inv_hw));
        *reinterpret_cast<short8_t*>(&dy[m * C + c]) = g8;
      }
      // packed accumulation: the two per-element fmas run as
      // v_pk_fma_f32 (the act' factor stays scalar — it has compares)
#pragma unroll
      for (int pp = 0; pp < 4; ++pp) {
        f32x2 zv = {bf2f(ushort_t(z8[pp * 2])),
                    bf2f(ushort_t(z8[pp * 2 + 1]))};
        f32x2 da = {
            bf2f(ushort_t(g8[pp * 2])) *
                act_bwd<ACT>(fmaf(zv[0], a[pp * 2], b[pp * 2])),
            bf2f(ushort_t(g8[pp * 2 + 1])) *
                act_bwd<ACT>(fmaf(zv[1], a[pp * 2 + 1], b[pp * 2 + 1]))};
        *reinterpret_cast<f32x2*>(&s1[pp * 2]) += da;
        *reinterpret_cast<f32x2*>(&s2[pp * 2]) += da * zv;
      }
    }
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) red[threadIdx.x * 8 + e] = s1[e];
  __syncthreads();
  if (threadIdx.x < unsigned(CT8) && c8 < C8) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float t = 0.f;
      for (int l = 0; l < ML; ++l) t += red[(threadIdx.x + l * CT8) * 8 + e];
      ws[int64_t(chunk) * C + c + e] = t;
    }
  }
  __syncthreads();
#pragma unroll
  for (int e = 0; e < 8; ++e) red[threadIdx.x * 8 + e] = s2[e];
  __syncthreads();
  if (threadIdx.x < unsigned(CT8) && c8 < C8) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float t = 0.f;
      for (int l = 0; l < ML; ++l) t += red[(threadIdx.x + l * CT8) * 8 + e];
      ws[(int64_t(chunks) + chunk) * C + c + e] = t;
    }
  }
}

// finalize -> dgamma/dbeta + apply coefficients {a, b, K0, K2}
template <bool TRAIN>
__global__ __launch_bounds__(256) void bn_bwd_finalize_kernel(
    const float* __restrict__ ws, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ dgamma,
    float* __restrict__ dbeta, float* __restrict__ coef, int64_t M, int C,
    int splits) {
  __shared__ float red[256][2];
  const int cl = int(threadIdx.x) % 16;
  const int sl = int(threadIdx.x) / 16;
  const int c = blockIdx.x * 16 + cl;
  float s1 = 0.f, s2 = 0.f;
  if (c < C) {
    float s1b = 0.f, s2b = 0.f;
    int sp = sl;
    for (; sp + 16 < splits; sp += 32) {
      s1 += ws[int64_t(sp) * C + c];
      s2 += ws[int64_t(splits + sp) * C + c];
      s1b += ws[int64_t(sp + 16) * C + c];
      s2b += ws[int64_t(splits + sp + 16) * C + c];
    }
    for (; sp < splits; sp += 16) {
      s1 += ws[int64_t(sp) * C + c];
      s2 += ws[int64_t(splits + sp) * C + c];
    }
    s1 += s1b;
    s2 += s2b;
  }
  red[threadIdx.x][0] = s1;
  red[threadIdx.x][1] = s2;
  __syncthreads();
  if (sl == 0 && c < C) {
    for (int l = 1; l < 16; ++l) {
      s1 += red[cl + l * 16][0];
      s2 += red[cl + l * 16][1];
    }
    const float mu = mean[c], inv = invstd[c];
    const float dg = (s2 - mu * s1) * inv;  // sum(da*zhat)
    dgamma[c] = dg;
    dbeta[c] = s1;
    const float a = gamma[c] * inv;
    coef[c] = a;
    coef[C + c] = beta[c] - mu * a;
    if (TRAIN) {
      const float inv_m = 1.f / float(M);
      const float k2 = a * inv * dg * inv_m;
      coef[2 * C + c] = -a * s1 * inv_m + mu * k2;  // K0
      coef[3 * C + c] = k2;                         // K2
    } else {
      coef[2 * C + c] = 0.f;
      coef[3 * C + c] = 0.f;
    }
  }
}

// (c8, mlane) layout like the partial kernel: the 4 coefficient vectors
// load ONCE per thread instead of 16 float4 loads per 8 elements (the
// flat version was load-issue bound at ~3.7 TB/s)
template <int ACT>
__global__ __launch_bounds__(256) void bn_bwd_apply_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ z,
    const float* __restrict__ coef, ushort_t* __restrict__ dz,
    int64_t M, int C, int chunks) {
  const int C8 = C >> 3;
  const int CT8 = C8 < 256 ? C8 : 256;
  const int ML = 256 / CT8;
  const int c_tiles = (C8 + CT8 - 1) / CT8;
  const int c8 = (blockIdx.x % c_tiles) * CT8 + int(threadIdx.x) % CT8;
  const int chunk = blockIdx.x / c_tiles;
  const int mlane = int(threadIdx.x) / CT8;
  const int c = c8 * 8;
  if (c8 >= C8 || mlane >= ML) return;
  const int64_t per = (M + chunks - 1) / chunks;
  const int64_t lo = chunk * per;
  const int64_t hi = std::min<int64_t>(lo + per, M);
  float a[8], b[8], k0[8], k2[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    a[e] = coef[c + e];
    b[e] = coef[C + c + e];
    k0[e] = coef[2 * C + c + e];
    k2[e] = coef[3 * C + c + e];
  }
  for (int64_t m = lo + mlane; m < hi; m += ML) {
    const int64_t off = m * C + c;
    short8_t zv = *reinterpret_cast<const short8_t*>(&z[off]);
    short8_t gv = *reinterpret_cast<const short8_t*>(&dy[off]);
    short8_t out;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float zf = bf2f(ushort_t(zv[e]));
      const float da =
          bf2f(ushort_t(gv[e])) * act_bwd<ACT>(fmaf(zf, a[e], b[e]));
      out[e] = short(f2bf(fmaf(a[e], da, k0[e]) - k2[e] * zf));
    }
    *reinterpret_cast<short8_t*>(&dz[off]) = out;
  }
}

// ONE cooperative launch for the whole BN backward: partial sums ->
// grid.sync -> finalize -> grid.sync -> apply.  Replaces three
// dependent launches per BN (two kernel ramps x 46 BNs of latency on
// the critical path; hipGraph replay measured unable to recover it).
// All reductions keep the fixed slab order — the grid sync is a
// barrier, not an atomic reduction.
template <int ACT, bool TRAIN>
__global__ __launch_bounds__(256) void bn_bwd_fused_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ z,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ dgamma, float* __restrict__ dbeta,
    float* __restrict__ ws, float* __restrict__ coef,
    ushort_t* __restrict__ dz, int64_t M, int C, int chunks) {
  __shared__ float red[256 * 8];
  const int C8 = C >> 3;
  const int CT8 = C8 < 256 ? C8 : 256;
  const int ML = 256 / CT8;
  const int c_tiles = (C8 + CT8 - 1) / CT8;
  // ---- phase 1: per-chunk partial sums (bn_bwd_partial body) ----------
  {
    const int c8 = (blockIdx.x % c_tiles) * CT8 + int(threadIdx.x) % CT8;
    const int chunk = blockIdx.x / c_tiles;
    const int mlane = int(threadIdx.x) / CT8;
    const int c = c8 * 8;
    const int64_t per = (M + chunks - 1) / chunks;
    const int64_t lo = chunk * per;
    const int64_t hi = std::min<int64_t>(lo + per, M);
    float s1[8] = {}, s2[8] = {};
    if (c8 < C8 && mlane < ML && chunk < chunks) {
      float a[8], b[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        a[e] = gamma[c + e] * invstd[c + e];
        b[e] = beta[c + e] - mean[c + e] * a[e];
      }
      for (int64_t m = lo + mlane; m < hi; m += ML) {
        short8_t z8 = *reinterpret_cast<const short8_t*>(&z[m * C + c]);
        short8_t g8 = *reinterpret_cast<const short8_t*>(&dy[m * C + c]);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float zv = bf2f(ushort_t(z8[e]));
          const float da =
              bf2f(ushort_t(g8[e])) * act_bwd<ACT>(fmaf(zv, a[e], b[e]));
          s1[e] += da;
          s2[e] = fmaf(da, zv, s2[e]);
        }
      }
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) red[threadIdx.x * 8 + e] = s1[e];
    __syncthreads();
    if (threadIdx.x < unsigned(CT8) && c8 < C8 && chunk < chunks) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float t = 0.f;
        for (int l = 0; l < ML; ++l)
          t += red[(threadIdx.x + l * CT8) * 8 + e];
        ws[int64_t(chunk) * C + c + e] = t;
      }
    }
    __syncthreads();
#pragma unroll
    for (int e = 0; e < 8; ++e) red[threadIdx.x * 8 + e] = s2[e];
    __syncthreads();
    if (threadIdx.x < unsigned(CT8) && c8 < C8 && chunk < chunks) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float t = 0.f;
        for (int l = 0; l < ML; ++l)
          t += red[(threadIdx.x + l * CT8) * 8 + e];
        ws[(int64_t(chunks) + chunk) * C + c + e] = t;
      }
    }
  }
  cg::this_grid().sync();
  // ---- phase 2: finalize (bn_bwd_finalize body on blocks 0..C/16) -----
  if (int(blockIdx.x) < (C + 15) / 16) {
    const int cl = int(threadIdx.x) % 16;
    const int sl = int(threadIdx.x) / 16;
    const int c = blockIdx.x * 16 + cl;
    float s1 = 0.f, s2 = 0.f;
    if (c < C) {
#pragma unroll 4
      for (int sp = sl; sp < chunks; sp += 16) {
        s1 += ws[int64_t(sp) * C + c];
        s2 += ws[int64_t(chunks + sp) * C + c];
      }
    }
    red[threadIdx.x * 2] = s1;
    red[threadIdx.x * 2 + 1] = s2;
    __syncthreads();
    if (sl == 0 && c < C) {
      for (int l = 1; l < 16; ++l) {
        s1 += red[(cl + l * 16) * 2];
        s2 += red[(cl + l * 16) * 2 + 1];
      }
      const float mu = mean[c], inv = invstd[c];
      const float dg = (s2 - mu * s1) * inv;
      dgamma[c] = dg;
      dbeta[c] = s1;
      const float a = gamma[c] * inv;
      coef[c] = a;
      coef[C + c] = beta[c] - mu * a;
      if (TRAIN) {
        const float inv_m = 1.f / float(M);
        const float k2 = a * inv * dg * inv_m;
        coef[2 * C + c] = -a * s1 * inv_m + mu * k2;
        coef[3 * C + c] = k2;
      } else {
        coef[2 * C + c] = 0.f;
        coef[3 * C + c] = 0.f;
      }
    }
  }
  cg::this_grid().sync();
  // ---- phase 3: apply (grid-stride; bn_bwd_apply body) ----------------
  {
    const int64_t total8 = M * C8;
    for (int64_t idx = int64_t(blockIdx.x) * 256 + threadIdx.x;
         idx < total8; idx += int64_t(gridDim.x) * 256) {
      const int c = int(idx % C8) * 8;
      const int64_t off = idx * 8;
      short8_t zv = *reinterpret_cast<const short8_t*>(&z[off]);
      short8_t gv = *reinterpret_cast<const short8_t*>(&dy[off]);
      short8_t out;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float a = coef[c + e];
        const float b = coef[C + c + e];
        const float k0 = coef[2 * C + c + e];
        const float k2 = coef[3 * C + c + e];
        const float zf = bf2f(ushort_t(zv[e]));
        const float da =
            bf2f(ushort_t(gv[e])) * act_bwd<ACT>(fmaf(zf, a, b));
        out[e] = short(f2bf(fmaf(a, da, k0) - k2 * zf));
      }
      *reinterpret_cast<short8_t*>(&dz[off]) = out;
    }
  }
}

}  // namespace

int bn_stats_chunks(int64_t M) { return red_geom(M, 256).chunks; }

int64_t bn_stats_ws_elems(int64_t M, int C) {
  return 2LL * red_geom(M, C).chunks * C;
}

void bn_stats(const void* z, float* mean, float* invstd, float* ab,
              const float* gamma, const float* beta, float* running_mean,
              float* running_var, float* ws, int64_t M, int C,
              float momentum, float eps, hipStream_t s) {
  const RedGeom g = red_geom(M, C);
  bn_stats_partial_kernel<<<dim3(g.c_tiles * g.chunks), 256, 0, s>>>(
      static_cast<const ushort_t*>(z), ws, M, C, g.chunks);
  bn_stats_finalize_kernel<<<dim3(ceil_div(C, 16)), 256, 0, s>>>(
      ws, mean, invstd, ab, gamma, beta, running_mean, running_var, M, C,
      g.chunks, momentum, eps);
}

// parallel pre-fold for very large slab counts (e.g. 112^2 layers produce
// ~25k GEMM m-blocks): out[2][folded][C] = grouped sums of ws[2][splits][C]
__global__ void bn_fold_kernel(const float* __restrict__ ws,
                               float* __restrict__ out, int splits,
                               int folded, int C) {
  const int64_t total = 2LL * folded * C;
  const int g = (splits + folded - 1) / folded;
  for (int64_t i = int64_t(blockIdx.x) * blockDim.x + threadIdx.x;
       i < total; i += int64_t(gridDim.x) * blockDim.x) {
    const int c = int(i % C);
    const int f = int((i / C) % folded);
    const int plane = int(i / (int64_t(folded) * C));
    const int lo = f * g;
    const int hi = lo + g < splits ? lo + g : splits;
    float v = 0.f;
    for (int sp = lo; sp < hi; ++sp)
      v += ws[(int64_t(plane) * splits + sp) * C + c];
    out[(int64_t(plane) * folded + f) * C + c] = v;
  }
}

void bn_finalize(const float* ws, int splits, float* mean, float* invstd,
                 float* ab, const float* gamma, const float* beta,
                 float* running_mean, float* running_var, int64_t M, int C,
                 float momentum, float eps, hipStream_t s) {
  static float* fold_buf = nullptr;
  static size_t fold_cap = 0;
  if (splits > 1024) {
    const int folded = 256;
    const size_t need = 2 * folded * size_t(C);
    if (need > fold_cap) {
      if (fold_buf) (void)hipFree(fold_buf);
      (void)hipMalloc(&fold_buf, need * sizeof(float));
      fold_cap = need;
    }
    const int64_t total = 2LL * folded * C;
    const int blocks = int(std::min<int64_t>((total + 255) / 256, 4096));
    bn_fold_kernel<<<dim3(blocks), 256, 0, s>>>(ws, fold_buf, splits,
                                                folded, C);
    bn_stats_finalize_kernel<<<dim3(ceil_div(C, 16)), 256, 0, s>>>(
        fold_buf, mean, invstd, ab, gamma, beta, running_mean, running_var,
        M, C, folded, momentum, eps);
    return;
  }
  bn_stats_finalize_kernel<<<dim3(ceil_div(C, 16)), 256, 0, s>>>(
      ws, mean, invstd, ab, gamma, beta, running_mean, running_var, M, C,
      splits, momentum, eps);
}

// SyncBN support: per-rank [sum, sumsq] of z reduced to ONE [2, C]
// vector (partial slabs + parallel fold) — the caller all-reduces it
// across ranks and finalizes with the GLOBAL element count via
// bn_finalize(splits=1, M=M*world) (SURVEY.md §2.4 optional stat sync).
void bn_sums(const void* z, float* ws, float* out, int64_t M, int C,
             hipStream_t s) {
  const RedGeom g = red_geom(M, C);
  bn_stats_partial_kernel<<<dim3(g.c_tiles * g.chunks), 256, 0, s>>>(
      static_cast<const ushort_t*>(z), ws, M, C, g.chunks);
  const int64_t total = 2LL * C;
  const int blocks = int(std::min<int64_t>((total + 255) / 256, 2048));
  bn_fold_kernel<<<dim3(blocks), 256, 0, s>>>(ws, out, g.chunks, 1, C);
}

void bn_fold_sums(const float* ws, int splits, float* out, int C,
                  hipStream_t s) {
  const int64_t total = 2LL * C;
  const int blocks = int(std::min<int64_t>((total + 255) / 256, 2048));
  bn_fold_kernel<<<dim3(blocks), 256, 0, s>>>(ws, out, splits, 1, C);
}

void bn_act_fwd(const void* z, const float* ab, int act, void* y,
                const void* residual, int64_t M, int C, hipStream_t s) {
  const RedGeom g = red_geom(M, C);
  auto* zp = static_cast<const ushort_t*>(z);
  auto* yp = static_cast<ushort_t*>(y);
  auto* rp = static_cast<const ushort_t*>(residual);
  ACT_SWITCH(act, {
    if (rp != nullptr)
      bn_act_fwd_kernel<kAct, true>
          <<<dim3(g.c_tiles * g.chunks), 256, 0, s>>>(zp, ab, yp, rp, M,
                                                      C, g.chunks);
    else
      bn_act_fwd_kernel<kAct, false>
          <<<dim3(g.c_tiles * g.chunks), 256, 0, s>>>(zp, ab, yp, nullptr,
                                                      M, C, g.chunks);
  });
}

void bn_act_bwd(const void* dy, const void* z, const float* mean,
                const float* invstd, const float* gamma, const float* beta,
                int act, bool training, float* dgamma, float* dbeta,
                void* dz, float* ws, float* coef, int64_t M, int C,
                bool apply, int ext_splits, hipStream_t s) {
  // ext_splits > 0: the partial sums were already produced by the kernel
  // that WROTE dy (conv-dgrad epilogue, EPI==2) into ws as [2, ext_splits,
  // C] — skip the bn_bwd_partial pass over dy+z entirely.  Very large
  // split counts (pw mblocks at 56²+ can be >6k) would serialize the
  // underfilled finalize, so pre-fold them with the parallel fold kernel.
  const RedGeom g = red_geom(M, C);
  int splits = ext_splits > 0 ? ext_splits : g.chunks;
  if (ext_splits > 1024) {
    constexpr int FOLDED = 256;
    static float* fold_buf = nullptr;
    static size_t fold_cap = 0;
    const size_t need = 2LL * FOLDED * C;
    if (need > fold_cap) {
      if (fold_buf) (void)hipFree(fold_buf);
      (void)hipMalloc(&fold_buf, need * sizeof(float));
      fold_cap = need;
    }
    const int blocks = int(std::min<int64_t>((2LL * FOLDED * C + 255) / 256,
                                             2048));
    bn_fold_kernel<<<dim3(blocks), 256, 0, s>>>(ws, fold_buf, splits,
                                                FOLDED, C);
    ws = fold_buf;
    splits = FOLDED;
  }
  auto* gp = static_cast<const ushort_t*>(dy);
  auto* zp = static_cast<const ushort_t*>(z);
  auto* dzp = static_cast<ushort_t*>(dz);
  // fused cooperative path: one launch instead of partial+finalize+apply.
  // Measured NEGATIVE and unstable (grid.sync cost ~18.3 vs 12.5 ms/step
  // and a crash in the small-shape tests) — default OFF, kept only as a
  // documented experiment (YAMS_BN_COOP=1 enables).
  static int coop_on = [] {
    const char* e = getenv("YAMS_BN_COOP");
    return e != nullptr && atoi(e) != 0;
  }();
  if (coop_on && ext_splits <= 0 && apply && training) {
    bool launched = false;
    ACT_SWITCH(act, {
      void* kernel = reinterpret_cast<void*>(
          &bn_bwd_fused_kernel<kAct, true>);
      int chunks = g.chunks;
      void* args[] = {&gp, &zp,
                      const_cast<float**>(&mean),
                      const_cast<float**>(&invstd),
                      const_cast<float**>(&gamma),
                      const_cast<float**>(&beta),
                      &dgamma, &dbeta, &ws, &coef, &dzp, &M,
                      const_cast<int*>(&C), &chunks};
      const hipError_t err = hipLaunchCooperativeKernel(
          kernel, dim3(g.c_tiles * g.chunks), dim3(256), args, 0, s);
      launched = err == hipSuccess;
      if (!launched) (void)hipGetLastError();  // clear, fall back
    });
    if (launched) return;
  }
  ACT_SWITCH(act, {
    if (ext_splits <= 0)
      bn_bwd_partial_kernel<kAct><<<dim3(g.c_tiles * g.chunks), 256, 0, s>>>(
          const_cast<ushort_t*>(gp), zp, mean, invstd, gamma, beta, ws, M,
          C, g.chunks);
    if (training)
      bn_bwd_finalize_kernel<true><<<dim3(ceil_div(C, 16)), 256, 0, s>>>(
          ws, mean, invstd, gamma, beta, dgamma, dbeta, coef, M, C,
          splits);
    else
      bn_bwd_finalize_kernel<false><<<dim3(ceil_div(C, 16)), 256, 0, s>>>(
          ws, mean, invstd, gamma, beta, dgamma, dbeta, coef, M, C,
          splits);
    if (apply)
      bn_bwd_apply_kernel<kAct><<<dim3(g.c_tiles * g.chunks), 256, 0, s>>>(
          gp, zp, coef, dzp, M, C, g.chunks);
  });
}

// apply-only entry: dz from precomputed coef[4C] (used when the
// partial/finalize ran via bn_bwd_stats and dz is materialized lazily,
// e.g. on the side stream for wgrad consumers)
void bn_bwd_apply(const void* dy, const void* z, const float* coef,
                  int act, void* dz, int64_t M, int C, hipStream_t s) {
  const RedGeom g = red_geom(M, C);
  ACT_SWITCH(act, {
    bn_bwd_apply_kernel<kAct><<<dim3(g.c_tiles * g.chunks), 256, 0, s>>>(
        static_cast<const ushort_t*>(dy), static_cast<const ushort_t*>(z),
        coef, static_cast<ushort_t*>(dz), M, C, g.chunks);
  });
}

// bn backward where dy still needs the avg-pool grad added (SE blocks):
// partial does add+reduce in one pass, then finalize + apply as usual.
void bn_act_bwd_addpool(void* dy, const void* z, const void* dpooled,
                        int HW, const float* mean, const float* invstd,
                        const float* gamma, const float* beta, int act,
                        bool training, float* dgamma, float* dbeta,
                        void* dz, float* ws, float* coef, int64_t M, int C,
                        hipStream_t s) {
  const RedGeom g = red_geom(M, C);
  auto* gp = static_cast<ushort_t*>(dy);
  auto* zp = static_cast<const ushort_t*>(z);
  auto* pp = static_cast<const ushort_t*>(dpooled);
  auto* dzp = static_cast<ushort_t*>(dz);
  ACT_SWITCH(act, {
    bn_bwd_partial_kernel<kAct, true>
        <<<dim3(g.c_tiles * g.chunks), 256, 0, s>>>(
            gp, zp, mean, invstd, gamma, beta, ws, M, C, g.chunks, pp, HW,
            1.f / float(HW));
    if (training)
      bn_bwd_finalize_kernel<true><<<dim3(ceil_div(C, 16)), 256, 0, s>>>(
          ws, mean, invstd, gamma, beta, dgamma, dbeta, coef, M, C,
          g.chunks);
    else
      bn_bwd_finalize_kernel<false><<<dim3(ceil_div(C, 16)), 256, 0, s>>>(
          ws, mean, invstd, gamma, beta, dgamma, dbeta, coef, M, C,
          g.chunks);
    bn_bwd_apply_kernel<kAct><<<dim3(g.c_tiles * g.chunks), 256, 0, s>>>(
        gp, zp, coef, dzp, M, C, g.chunks);
  });
}

}  // namespace yams
