// BatchNorm (K5) + fused activation (K6) + residual (K8) for gfx950.
//
// Training statistics are fp32 two-stage slab reductions over the bf16
// conv output z[M, C] (M = N*H*W rows): per-(chunk,mlane) partial
// sum/sumsq slabs, then a finalize kernel producing mean/invstd and the
// running-stat update (PyTorch convention: running_var gets the UNBIASED
// variance).  Deterministic — no atomics.
//
// Backward identity used (training mode):
//   v0   = gamma*zhat + beta,  zhat = (z-mean)*invstd
//   da   = dy * act'(v0)
//   dgamma = sum da*zhat ; dbeta = sum da
//   dz   = gamma*invstd * (da - dbeta/M - zhat*dgamma/M)
// Eval mode drops the two mean-correction terms.

#include <algorithm>

#include "common.h"
#include "engine.h"

namespace yams {

namespace {

// --- stats partial: thread = (channel lane, m lane) ---------------------
__global__ __launch_bounds__(256) void bn_stats_partial_kernel(
    const ushort_t* __restrict__ z, float* __restrict__ ws, int64_t M, int C,
    int chunks) {
  const int CT = C < 256 ? C : 256;
  const int ML = 256 / CT;
  const int c_tiles = (C + CT - 1) / CT;
  const int c = (blockIdx.x % c_tiles) * CT + int(threadIdx.x) % CT;
  const int chunk = blockIdx.x / c_tiles;
  const int mlane = int(threadIdx.x) / CT;
  if (c >= C || mlane >= ML) return;
  const int64_t per = (M + chunks - 1) / chunks;
  const int64_t lo = chunk * per;
  const int64_t hi = std::min<int64_t>(lo + per, M);
  float s1 = 0.f, s2 = 0.f;
  for (int64_t m = lo + mlane; m < hi; m += ML) {
    const float v = bf2f(z[m * C + c]);
    s1 += v;
    s2 = fmaf(v, v, s2);
  }
  const int64_t splits = int64_t(chunks) * ML;
  const int64_t slot = int64_t(chunk) * ML + mlane;
  ws[slot * C + c] = s1;
  ws[(splits + slot) * C + c] = s2;
}

__global__ void bn_stats_finalize_kernel(
    const float* __restrict__ ws, float* __restrict__ mean,
    float* __restrict__ invstd, float* __restrict__ running_mean,
    float* __restrict__ running_var, int64_t M, int C, int splits,
    float momentum, float eps) {
  for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += gridDim.x * blockDim.x) {
    float s1 = 0.f, s2 = 0.f;
    for (int sp = 0; sp < splits; ++sp) {
      s1 += ws[int64_t(sp) * C + c];
      s2 += ws[int64_t(splits + sp) * C + c];
    }
    const float mu = s1 / M;
    float var = s2 / M - mu * mu;
    var = fmaxf(var, 0.f);
    mean[c] = mu;
    invstd[c] = rsqrtf(var + eps);
    if (running_mean != nullptr) {
      const float unbiased = M > 1 ? var * (float(M) / float(M - 1)) : var;
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
  }
}

// --- forward normalize+act(+residual): flat vec8 ------------------------
template <int ACT, bool RES>
__global__ __launch_bounds__(256) void bn_act_fwd_kernel(
    const ushort_t* __restrict__ z, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, ushort_t* __restrict__ y,
    const ushort_t* __restrict__ res, int64_t total8, int C8) {
  for (int64_t idx = int64_t(blockIdx.x) * blockDim.x + threadIdx.x;
       idx < total8; idx += int64_t(gridDim.x) * blockDim.x) {
    const int c = int(idx % C8) * 8;
    const int64_t off = idx * 8;
    short8_t zv = *reinterpret_cast<const short8_t*>(&z[off]);
    short8_t rv;
    if (RES) rv = *reinterpret_cast<const short8_t*>(&res[off]);
    short8_t out;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float a = gamma[c + e] * invstd[c + e];
      const float b = beta[c + e] - mean[c + e] * a;
      float v = act_fwd<ACT>(fmaf(bf2f(ushort_t(zv[e])), a, b));
      if (RES) v += bf2f(ushort_t(rv[e]));
      out[e] = short(f2bf(v));
    }
    *reinterpret_cast<short8_t*>(&y[off]) = out;
  }
}

// --- backward reduce: per-channel sum(da), sum(da*z) --------------------
template <int ACT>
__global__ __launch_bounds__(256) void bn_bwd_partial_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ z,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ ws, int64_t M, int C, int chunks) {
  const int CT = C < 256 ? C : 256;
  const int ML = 256 / CT;
  const int c_tiles = (C + CT - 1) / CT;
  const int c = (blockIdx.x % c_tiles) * CT + int(threadIdx.x) % CT;
  const int chunk = blockIdx.x / c_tiles;
  const int mlane = int(threadIdx.x) / CT;
  if (c >= C || mlane >= ML) return;
  const float a = gamma[c] * invstd[c];
  const float b = beta[c] - mean[c] * a;
  const int64_t per = (M + chunks - 1) / chunks;
  const int64_t lo = chunk * per;
  const int64_t hi = std::min<int64_t>(lo + per, M);
  float s1 = 0.f, s2 = 0.f;
  for (int64_t m = lo + mlane; m < hi; m += ML) {
    const float zv = bf2f(z[m * C + c]);
    const float da = bf2f(dy[m * C + c]) * act_bwd<ACT>(fmaf(zv, a, b));
    s1 += da;
    s2 = fmaf(da, zv, s2);
  }
  const int64_t splits = int64_t(chunks) * ML;
  const int64_t slot = int64_t(chunk) * ML + mlane;
  ws[slot * C + c] = s1;
  ws[(splits + slot) * C + c] = s2;
}

__global__ void bn_bwd_finalize_kernel(
    const float* __restrict__ ws, const float* __restrict__ mean,
    const float* __restrict__ invstd, float* __restrict__ dgamma,
    float* __restrict__ dbeta, int C, int splits) {
  for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += gridDim.x * blockDim.x) {
    float s1 = 0.f, s2 = 0.f;
    for (int sp = 0; sp < splits; ++sp) {
      s1 += ws[int64_t(sp) * C + c];
      s2 += ws[int64_t(splits + sp) * C + c];
    }
    dbeta[c] = s1;
    dgamma[c] = (s2 - mean[c] * s1) * invstd[c];  // = sum(da * zhat)
  }
}

template <int ACT, bool TRAIN>
__global__ __launch_bounds__(256) void bn_bwd_apply_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ z,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ dgamma, const float* __restrict__ dbeta,
    ushort_t* __restrict__ dz, int64_t total8, int C8, float inv_m) {
  for (int64_t idx = int64_t(blockIdx.x) * blockDim.x + threadIdx.x;
       idx < total8; idx += int64_t(gridDim.x) * blockDim.x) {
    const int c = int(idx % C8) * 8;
    const int64_t off = idx * 8;
    short8_t zv = *reinterpret_cast<const short8_t*>(&z[off]);
    short8_t gv = *reinterpret_cast<const short8_t*>(&dy[off]);
    short8_t out;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float inv = invstd[c + e];
      const float a = gamma[c + e] * inv;
      const float b = beta[c + e] - mean[c + e] * a;
      const float zf = bf2f(ushort_t(zv[e]));
      const float da = bf2f(ushort_t(gv[e])) * act_bwd<ACT>(fmaf(zf, a, b));
      float d;
      if (TRAIN) {
        const float zhat = (zf - mean[c + e]) * inv;
        d = a * (da - dbeta[c + e] * inv_m - zhat * dgamma[c + e] * inv_m);
      } else {
        d = a * da;
      }
      out[e] = short(f2bf(d));
    }
    *reinterpret_cast<short8_t*>(&dz[off]) = out;
  }
}

int elem_grid(int64_t total8) {
  return int(std::min<int64_t>((total8 + 255) / 256, 4096));
}

}  // namespace

int bn_stats_chunks(int64_t M) {
  int c = int(std::min<int64_t>((M + 511) / 512, 256));
  return c < 1 ? 1 : c;
}

int64_t bn_stats_ws_elems(int64_t M, int C) {
  const int CT = C < 256 ? C : 256;
  const int ML = 256 / CT;
  return 2LL * bn_stats_chunks(M) * ML * C;
}

void bn_stats(const void* z, float* mean, float* invstd, float* running_mean,
              float* running_var, float* ws, int64_t M, int C,
              float momentum, float eps, hipStream_t s) {
  const int chunks = bn_stats_chunks(M);
  const int CT = C < 256 ? C : 256;
  const int ML = 256 / CT;
  const int c_tiles = (C + CT - 1) / CT;
  bn_stats_partial_kernel<<<dim3(c_tiles * chunks), 256, 0, s>>>(
      static_cast<const ushort_t*>(z), ws, M, C, chunks);
  bn_stats_finalize_kernel<<<dim3(ceil_div(C, 256)), 256, 0, s>>>(
      ws, mean, invstd, running_mean, running_var, M, C, chunks * ML,
      momentum, eps);
}

void bn_act_fwd(const void* z, const float* mean, const float* invstd,
                const float* gamma, const float* beta, int act, void* y,
                const void* residual, int64_t M, int C, hipStream_t s) {
  const int64_t total8 = M * (C >> 3);
  const int C8 = C >> 3;
  auto* zp = static_cast<const ushort_t*>(z);
  auto* yp = static_cast<ushort_t*>(y);
  auto* rp = static_cast<const ushort_t*>(residual);
  ACT_SWITCH(act, {
    if (rp != nullptr)
      bn_act_fwd_kernel<kAct, true><<<elem_grid(total8), 256, 0, s>>>(
          zp, mean, invstd, gamma, beta, yp, rp, total8, C8);
    else
      bn_act_fwd_kernel<kAct, false><<<elem_grid(total8), 256, 0, s>>>(
          zp, mean, invstd, gamma, beta, yp, nullptr, total8, C8);
  });
}

void bn_act_bwd(const void* dy, const void* z, const float* mean,
                const float* invstd, const float* gamma, const float* beta,
                int act, bool training, float* dgamma, float* dbeta,
                void* dz, float* ws, int64_t M, int C, hipStream_t s) {
  const int chunks = bn_stats_chunks(M);
  const int CT = C < 256 ? C : 256;
  const int ML = 256 / CT;
  const int c_tiles = (C + CT - 1) / CT;
  auto* gp = static_cast<const ushort_t*>(dy);
  auto* zp = static_cast<const ushort_t*>(z);
  auto* dzp = static_cast<ushort_t*>(dz);
  const int64_t total8 = M * (C >> 3);
  const float inv_m = 1.f / float(M);
  ACT_SWITCH(act, {
    bn_bwd_partial_kernel<kAct><<<dim3(c_tiles * chunks), 256, 0, s>>>(
        gp, zp, mean, invstd, gamma, beta, ws, M, C, chunks);
    bn_bwd_finalize_kernel<<<dim3(ceil_div(C, 256)), 256, 0, s>>>(
        ws, mean, invstd, dgamma, dbeta, C, chunks * ML);
    if (training)
      bn_bwd_apply_kernel<kAct, true><<<elem_grid(total8), 256, 0, s>>>(
          gp, zp, mean, invstd, gamma, beta, dgamma, dbeta, dzp, total8,
          C >> 3, inv_m);
    else
      bn_bwd_apply_kernel<kAct, false><<<elem_grid(total8), 256, 0, s>>>(
          gp, zp, mean, invstd, gamma, beta, dgamma, dbeta, dzp, total8,
          C >> 3, inv_m);
  });
}

}  // namespace yams
