// Small fused kernels: bias+activation (head/SE FCs), SE channel scale
// (K7 pieces), global average pool (K9), label-smoothed softmax CE (K11).
// All NHWC/bf16 with fp32 math; reductions are deterministic (fixed-order
// LDS/slab combines, no atomics).

#include <algorithm>

#include "common.h"
#include "engine.h"

namespace yams {

namespace {

inline Lop to_lop(const LopDesc& d) {
  Lop L;
  L.kind = d.kind;
  L.act = d.act;
  L.coef = d.coef;
  L.z = static_cast<const ushort_t*>(d.z);
  L.C = d.C;
  return L;
}

template <int LOPK>
DEV_INLINE void load8_lop(const Lop& L, const ushort_t* p, int64_t off,
                          int c, float* out) {
  short8_t v = *reinterpret_cast<const short8_t*>(&p[off]);
  if (LOPK == LOP_NONE) {
#pragma unroll
    for (int e = 0; e < 8; ++e) out[e] = bf2f(ushort_t(v[e]));
  } else if (LOPK == LOP_BNACT) {
#pragma unroll
    for (int e = 0; e < 8; ++e)
      out[e] = lop_applyT<LOP_BNACT>(L, bf2f(ushort_t(v[e])), 0.f, c + e);
  } else {
    short8_t z = *reinterpret_cast<const short8_t*>(&L.z[off]);
#pragma unroll
    for (int e = 0; e < 8; ++e)
      out[e] = lop_applyT<LOP_BNBWD>(L, bf2f(ushort_t(v[e])),
                                     bf2f(ushort_t(z[e])), c + e);
  }
}

int elem_grid(int64_t total) {
  return int(std::min<int64_t>((total + 255) / 256, 4096));
}

// --- bias + act on [M, C] (vec8) ---------------------------------------
template <int ACT>
__global__ __launch_bounds__(256) void bias_act_fwd_kernel(
    const ushort_t* __restrict__ z, const float* __restrict__ bias,
    ushort_t* __restrict__ y, int64_t total8, int C8) {
  const int tot = int(total8);  // < 2^31 for every supported shape
  for (int idx = int(blockIdx.x) * int(blockDim.x) + int(threadIdx.x);
       idx < tot; idx += int(gridDim.x) * int(blockDim.x)) {
    const int c = (idx % C8) * 8;
    short8_t zv = *reinterpret_cast<const short8_t*>(&z[idx * 8]);
    short8_t out;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = bf2f(ushort_t(zv[e]));
      if (bias != nullptr) v += bias[c + e];
      out[e] = short(f2bf(act_fwd<ACT>(v)));
    }
    *reinterpret_cast<short8_t*>(&y[idx * 8]) = out;
  }
}

// --- bias_act backward: dz everywhere + per-channel dbias slabs ---------
template <int ACT>
__global__ __launch_bounds__(256) void bias_act_bwd_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ z,
    const float* __restrict__ bias, ushort_t* __restrict__ dz,
    float* __restrict__ ws, int64_t M, int C, int chunks) {
  const int CT = C < 256 ? C : 256;
  const int ML = 256 / CT;
  const int c_tiles = (C + CT - 1) / CT;
  const int c = (blockIdx.x % c_tiles) * CT + int(threadIdx.x) % CT;
  const int chunk = blockIdx.x / c_tiles;
  const int mlane = int(threadIdx.x) / CT;
  if (c >= C || mlane >= ML) return;
  const float b = bias != nullptr ? bias[c] : 0.f;
  const int64_t per = (M + chunks - 1) / chunks;
  const int64_t lo = chunk * per;
  const int64_t hi = std::min<int64_t>(lo + per, M);
  float s = 0.f;
  for (int64_t m = lo + mlane; m < hi; m += ML) {
    const float v0 = bf2f(z[m * C + c]) + b;
    const float d = bf2f(dy[m * C + c]) * act_bwd<ACT>(v0);
    dz[m * C + c] = f2bf(d);
    s += d;
  }
  if (ws != nullptr) ws[int64_t(chunk * ML + mlane) * C + c] = s;
}

// --- SE channel scale: y[n,hw,c] = x * s[n,c] ---------------------------
template <int LOPK>
__global__ __launch_bounds__(256) void rowscale_fwd_kernel(
    const ushort_t* __restrict__ x, const ushort_t* __restrict__ scale,
    ushort_t* __restrict__ y, int64_t total8, int HW, int C8, Lop lopx) {
  const int tot = int(total8);  // < 2^31 for every supported shape
  for (int idx = int(blockIdx.x) * int(blockDim.x) + int(threadIdx.x);
       idx < tot; idx += int(gridDim.x) * int(blockDim.x)) {
    const int c8 = idx % C8;
    const int n = idx / (C8 * HW);
    float xv[8];
    load8_lop<LOPK>(lopx, x, idx * 8, c8 * 8, xv);
    short8_t sv = *reinterpret_cast<const short8_t*>(
        &scale[(int64_t(n) * C8 + c8) * 8]);
    short8_t out;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      out[e] = short(f2bf(xv[e] * bf2f(ushort_t(sv[e]))));
    *reinterpret_cast<short8_t*>(&y[idx * 8]) = out;
  }
}

// dx = dy * s (broadcast over hw); grid (n x c8 tiles); vec8 channels,
// hw lanes folded through LDS: ds[n,c] = sum_hw dy*x
template <int LOPK>
__global__ __launch_bounds__(256) void rowscale_bwd_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ x,
    const ushort_t* __restrict__ scale, ushort_t* __restrict__ dx,
    ushort_t* __restrict__ dscale, int HW, int C, int c8_tiles, Lop lopx) {
  __shared__ float red[256 * 8];
  const int C8 = C >> 3;
  const int CT8 = C8 < 256 ? C8 : 256;
  const int ML = 256 / CT8;
  const int n = blockIdx.x / c8_tiles;
  const int c8 = (blockIdx.x % c8_tiles) * CT8 + int(threadIdx.x) % CT8;
  const int hwlane = int(threadIdx.x) / CT8;
  const int64_t base = int64_t(n) * HW * C;
  const int c = c8 * 8;
  float acc[8] = {};
  float sv[8];
  LopReg<LOPK, 8> lr;
  if (c8 < C8 && hwlane < ML) {
    lr.init(lopx, c);
    short8_t s8 = *reinterpret_cast<const short8_t*>(
        &scale[int64_t(n) * C + c]);
#pragma unroll
    for (int e = 0; e < 8; ++e) sv[e] = bf2f(ushort_t(s8[e]));
    for (int hw = hwlane; hw < HW; hw += ML) {
      const int64_t off = base + int64_t(hw) * C + c;
      short8_t g8 = *reinterpret_cast<const short8_t*>(&dy[off]);
      float x8[8];
      {
        short8_t xv = *reinterpret_cast<const short8_t*>(&x[off]);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          x8[e] = lr.apply(bf2f(ushort_t(xv[e])), 0.f, e);
      }
      short8_t o8;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float g = bf2f(ushort_t(g8[e]));
        acc[e] = fmaf(g, x8[e], acc[e]);
        o8[e] = short(f2bf(g * sv[e]));
      }
      *reinterpret_cast<short8_t*>(&dx[off]) = o8;
    }
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) red[threadIdx.x * 8 + e] = acc[e];
  __syncthreads();
  if (threadIdx.x < unsigned(CT8) && c8 < C8) {
    short8_t out;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = 0.f;
      for (int l = 0; l < ML; ++l) v += red[(threadIdx.x + l * CT8) * 8 + e];
      out[e] = short(f2bf(v));
    }
    *reinterpret_cast<short8_t*>(&dscale[int64_t(n) * C + c]) = out;
  }
}

// --- global average pool ------------------------------------------------
template <int LOPK>
__global__ __launch_bounds__(256) void avgpool_fwd_kernel(
    const ushort_t* __restrict__ x, ushort_t* __restrict__ out, int HW,
    int C, int c8_tiles, Lop lopx) {
  __shared__ float red[256 * 8];
  const int C8 = C >> 3;
  const int CT8 = C8 < 256 ? C8 : 256;
  const int ML = 256 / CT8;
  const int n = blockIdx.x / c8_tiles;
  const int c8 = (blockIdx.x % c8_tiles) * CT8 + int(threadIdx.x) % CT8;
  const int hwlane = int(threadIdx.x) / CT8;
  const int64_t base = int64_t(n) * HW * C;
  const int c = c8 * 8;
  float acc[8] = {};
  LopReg<LOPK, 8> lr;
  if (c8 < C8 && hwlane < ML) {
    lr.init(lopx, c);
    for (int hw = hwlane; hw < HW; hw += ML) {
      short8_t xv = *reinterpret_cast<const short8_t*>(
          &x[base + int64_t(hw) * C + c]);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += lr.apply(bf2f(ushort_t(xv[e])),
                                                     0.f, e);
    }
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) red[threadIdx.x * 8 + e] = acc[e];
  __syncthreads();
  if (threadIdx.x < unsigned(CT8) && c8 < C8) {
    short8_t o8;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = 0.f;
      for (int l = 0; l < ML; ++l) v += red[(threadIdx.x + l * CT8) * 8 + e];
      o8[e] = short(f2bf(v / HW));
    }
    *reinterpret_cast<short8_t*>(&out[int64_t(n) * C + c]) = o8;
  }
}

template <bool ACCUM>
__global__ __launch_bounds__(256) void avgpool_bwd_kernel(
    const ushort_t* __restrict__ dpooled, ushort_t* __restrict__ dx,
    int64_t total8, int HW, int C8, float inv_hw) {
  const int tot = int(total8);  // < 2^31 for every supported shape
  for (int idx = int(blockIdx.x) * int(blockDim.x) + int(threadIdx.x);
       idx < tot; idx += int(gridDim.x) * int(blockDim.x)) {
    const int c8 = idx % C8;
    const int n = idx / (C8 * HW);
    short8_t gv = *reinterpret_cast<const short8_t*>(
        &dpooled[(int64_t(n) * C8 + c8) * 8]);
    short8_t out;
    if (ACCUM) {
      short8_t cur = *reinterpret_cast<const short8_t*>(&dx[idx * 8]);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        out[e] = short(f2bf(bf2f(ushort_t(cur[e])) +
                            bf2f(ushort_t(gv[e])) * inv_hw));
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e)
        out[e] = short(f2bf(bf2f(ushort_t(gv[e])) * inv_hw));
    }
    *reinterpret_cast<short8_t*>(&dx[idx * 8]) = out;
  }
}

// --- label-smoothed softmax CE: one block per row -----------------------
__global__ __launch_bounds__(256) void ce_row_kernel(
    const ushort_t* __restrict__ logits, const int64_t* __restrict__ target,
    float smoothing, float* __restrict__ dlogits,
    float* __restrict__ rowloss, int N, int C) {
  __shared__ float red[256];
  const int n = blockIdx.x;
  const int tid = threadIdx.x;
  const ushort_t* row = logits + int64_t(n) * C;
  // 1) max
  float m = -3.4e38f;
  for (int c = tid; c < C; c += 256) m = fmaxf(m, bf2f(row[c]));
  red[tid] = m;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (tid < off) red[tid] = fmaxf(red[tid], red[tid + off]);
    __syncthreads();
  }
  m = red[0];
  __syncthreads();
  // 2) sum exp and sum logits
  float se = 0.f, sl = 0.f;
  for (int c = tid; c < C; c += 256) {
    const float v = bf2f(row[c]);
    se += __expf(v - m);
    sl += v;
  }
  red[tid] = se;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (tid < off) red[tid] += red[tid + off];
    __syncthreads();
  }
  se = red[0];
  __syncthreads();
  red[tid] = sl;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (tid < off) red[tid] += red[tid + off];
    __syncthreads();
  }
  sl = red[0];
  __syncthreads();
  const float log_z = m + __logf(se);
  const int64_t y = target[n];
  const float inv_n = 1.f / float(N);
  // 3) dlogits = (softmax - ((1-s)*onehot + s/C)) / N ; rowloss
  for (int c = tid; c < C; c += 256) {
    const float p = __expf(bf2f(row[c]) - m) / se;
    float t = smoothing / C;
    if (c == int(y)) t += 1.f - smoothing;
    dlogits[int64_t(n) * C + c] = (p - t) * inv_n;
  }
  if (tid == 0) {
    const float xy = bf2f(row[y]);
    rowloss[n] = log_z - (1.f - smoothing) * xy - (smoothing / C) * sl;
  }
}

__global__ void ce_final_kernel(const float* __restrict__ rowloss,
                                float* __restrict__ loss, int N) {
  __shared__ float red[256];
  float s = 0.f;
  for (int n = threadIdx.x; n < N; n += 256) s += rowloss[n];
  red[threadIdx.x] = s;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) *loss = red[0] / N;
}

// ---- dropout (K14): stateless counter-based RNG ------------------------
// Philox-style: the mask is a pure function of (seed, element index), so
// backward regenerates it bit-exactly from the same seed — nothing is
// stored, and runs are reproducible given the training seed (SURVEY.md
// §5.2 determinism).  splitmix64 finalizer: passes practical uniformity
// tests, one mul-xor chain per element.
DEV_INLINE uint32_t yams_rand_u32(uint64_t seed, int64_t idx) {
  uint64_t z = seed + uint64_t(idx) * 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return uint32_t((z ^ (z >> 31)) >> 32);
}

template <bool FWD>
__global__ void dropout_kernel(const ushort_t* __restrict__ x,
                               ushort_t* __restrict__ y, int64_t total,
                               uint64_t seed, uint32_t keep_thresh,
                               float inv_keep) {
  for (int64_t i = int64_t(blockIdx.x) * blockDim.x + threadIdx.x;
       i < total; i += int64_t(gridDim.x) * blockDim.x) {
    const bool keep = yams_rand_u32(seed, i) < keep_thresh;
    y[i] = keep ? ushort_t(f2bf(bf2f(x[i]) * inv_keep)) : ushort_t(0);
  }
}

}  // namespace

void dropout(const void* x, void* y, int64_t total, uint64_t seed, float p,
             bool forward, hipStream_t s) {
  const float keep = 1.f - p;
  const uint32_t thresh =
      keep >= 1.f ? 0xFFFFFFFFu : uint32_t(keep * 4294967296.0);
  const int blocks = int(std::min<int64_t>((total + 255) / 256, 4096));
  // fwd and bwd run the SAME kernel: dx = dy * mask / keep
  if (forward)
    dropout_kernel<true><<<dim3(blocks), 256, 0, s>>>(
        static_cast<const ushort_t*>(x), static_cast<ushort_t*>(y), total,
        seed, thresh, 1.f / keep);
  else
    dropout_kernel<false><<<dim3(blocks), 256, 0, s>>>(
        static_cast<const ushort_t*>(x), static_cast<ushort_t*>(y), total,
        seed, thresh, 1.f / keep);
}

void bias_act_fwd(const void* z, const float* bias, int act, void* y,
                  int64_t M, int C, hipStream_t s) {
  const int64_t total8 = M * (C >> 3);
  ACT_SWITCH(act, {
    bias_act_fwd_kernel<kAct><<<elem_grid(total8), 256, 0, s>>>(
        static_cast<const ushort_t*>(z), bias, static_cast<ushort_t*>(y),
        total8, C >> 3);
  });
}

int bias_bwd_chunks(int64_t M) {
  int c = int(std::min<int64_t>((M + 3) / 4, 64));
  return c < 1 ? 1 : c;
}

int64_t bias_bwd_ws_elems(int64_t M, int C) {
  const int CT = C < 256 ? C : 256;
  const int ML = 256 / CT;
  return int64_t(bias_bwd_chunks(M)) * ML * C;
}

void bias_act_bwd(const void* dy, const void* z, const float* bias, int act,
                  void* dz, float* dbias, float* ws, int64_t M, int C,
                  hipStream_t s) {
  const int chunks = bias_bwd_chunks(M);
  const int CT = C < 256 ? C : 256;
  const int ML = 256 / CT;
  const int c_tiles = (C + CT - 1) / CT;
  ACT_SWITCH(act, {
    bias_act_bwd_kernel<kAct><<<dim3(c_tiles * chunks), 256, 0, s>>>(
        static_cast<const ushort_t*>(dy), static_cast<const ushort_t*>(z),
        bias, static_cast<ushort_t*>(dz), dbias != nullptr ? ws : nullptr,
        M, C, chunks);
  });
  if (dbias != nullptr) slab_reduce(ws, dbias, C, chunks * ML, s);
}

void rowscale_fwd(const void* x, const void* scale, void* y, int N, int HW,
                  int C, LopDesc lopx, hipStream_t s) {
  const int64_t total8 = int64_t(N) * HW * (C >> 3);
  const Lop L = to_lop(lopx);
  auto* xp = static_cast<const ushort_t*>(x);
  auto* sp = static_cast<const ushort_t*>(scale);
  auto* yp = static_cast<ushort_t*>(y);
  if (lopx.kind == 0)
    rowscale_fwd_kernel<0><<<elem_grid(total8), 256, 0, s>>>(
        xp, sp, yp, total8, HW, C >> 3, L);
  else
    rowscale_fwd_kernel<1><<<elem_grid(total8), 256, 0, s>>>(
        xp, sp, yp, total8, HW, C >> 3, L);
}

void rowscale_bwd(const void* dy, const void* x, const void* scale, void* dx,
                  void* dscale, int N, int HW, int C, LopDesc lopx,
                  hipStream_t s) {
  const int C8 = C >> 3;
  const int CT8 = C8 < 256 ? C8 : 256;
  const int c8_tiles = (C8 + CT8 - 1) / CT8;
  const Lop L = to_lop(lopx);
  if (lopx.kind == 0)
    rowscale_bwd_kernel<0><<<dim3(N * c8_tiles), 256, 0, s>>>(
        static_cast<const ushort_t*>(dy), static_cast<const ushort_t*>(x),
        static_cast<const ushort_t*>(scale), static_cast<ushort_t*>(dx),
        static_cast<ushort_t*>(dscale), HW, C, c8_tiles, L);
  else
    rowscale_bwd_kernel<1><<<dim3(N * c8_tiles), 256, 0, s>>>(
        static_cast<const ushort_t*>(dy), static_cast<const ushort_t*>(x),
        static_cast<const ushort_t*>(scale), static_cast<ushort_t*>(dx),
        static_cast<ushort_t*>(dscale), HW, C, c8_tiles, L);
}

void avgpool_fwd(const void* x, void* out, int N, int HW, int C,
                 LopDesc lopx, hipStream_t s) {
  const int C8 = C >> 3;
  const int CT8 = C8 < 256 ? C8 : 256;
  const int c8_tiles = (C8 + CT8 - 1) / CT8;
  const Lop L = to_lop(lopx);
  if (lopx.kind == 0)
    avgpool_fwd_kernel<0><<<dim3(N * c8_tiles), 256, 0, s>>>(
        static_cast<const ushort_t*>(x), static_cast<ushort_t*>(out), HW, C,
        c8_tiles, L);
  else
    avgpool_fwd_kernel<1><<<dim3(N * c8_tiles), 256, 0, s>>>(
        static_cast<const ushort_t*>(x), static_cast<ushort_t*>(out), HW, C,
        c8_tiles, L);
}

void avgpool_bwd(const void* dpooled, void* dx, int N, int HW, int C,
                 bool accumulate, hipStream_t s) {
  const int64_t total8 = int64_t(N) * HW * (C >> 3);
  if (accumulate)
    avgpool_bwd_kernel<true><<<elem_grid(total8), 256, 0, s>>>(
        static_cast<const ushort_t*>(dpooled), static_cast<ushort_t*>(dx),
        total8, HW, C >> 3, 1.f / HW);
  else
    avgpool_bwd_kernel<false><<<elem_grid(total8), 256, 0, s>>>(
        static_cast<const ushort_t*>(dpooled), static_cast<ushort_t*>(dx),
        total8, HW, C >> 3, 1.f / HW);
}

void label_smooth_ce(const void* logits, const int64_t* target,
                     float smoothing, float* loss, float* dlogits,
                     float* rowloss, int N, int C, hipStream_t s) {
  ce_row_kernel<<<dim3(N), 256, 0, s>>>(
      static_cast<const ushort_t*>(logits), target, smoothing, dlogits,
      rowloss, N, C);
  ce_final_kernel<<<dim3(1), 256, 0, s>>>(rowloss, loss, N);
}

}  // namespace yams
