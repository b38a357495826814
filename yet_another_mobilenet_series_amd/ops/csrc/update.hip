// Fused multi-tensor parameter updates (K12/K13):
// SGD(momentum, nesterov, per-tensor weight decay), TF-semantics RMSprop,
// and the EMA shadow update.  A launch covers up to 32 tensors via a
// by-value chunk table (block ranges precomputed on the host) — one
// kernel per optimizer step instead of ~200 per-tensor torch ops.

#include <algorithm>

#include "common.h"
#include "engine.h"

namespace yams {

namespace {

constexpr int MAXC = 32;
constexpr int TPB = 256;
constexpr int VEC = 4;

struct ChunkTable {
  const float* g[MAXC];
  float* p[MAXC];
  float* m[MAXC];   // momentum buf (SGD) / square-avg (RMSprop)
  float* m2[MAXC];  // momentum buf (RMSprop) / unused
  float wd[MAXC];
  int numel[MAXC];
  int start_block[MAXC + 1];
  int count;
};

DEV_INLINE int find_chunk(const ChunkTable& t, int bid) {
  int c = 0;
  while (c + 1 < t.count && bid >= t.start_block[c + 1]) ++c;
  return c;
}

__global__ __launch_bounds__(TPB) void sgd_kernel(
    ChunkTable t, float lr, float momentum, bool nesterov) {
  const int c = find_chunk(t, blockIdx.x);
  const int nb = t.start_block[c + 1] - t.start_block[c];
  const int lb = blockIdx.x - t.start_block[c];
  const int n = t.numel[c];
  const float wd = t.wd[c];
  float* p = t.p[c];
  float* m = t.m[c];
  const float* g = t.g[c];
  for (int64_t i = (int64_t(lb) * TPB + threadIdx.x) * VEC; i < n;
       i += int64_t(nb) * TPB * VEC) {
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      const int64_t j = i + e;
      if (j >= n) break;
      float gv = g[j] + wd * p[j];
      if (momentum != 0.f) {
        const float mv = momentum * m[j] + gv;
        m[j] = mv;
        gv = nesterov ? gv + momentum * mv : mv;
      }
      p[j] -= lr * gv;
    }
  }
}

// TF semantics (utils/optim.py RMSpropTF): ms init 1 (host), eps inside
// sqrt, momentum accumulates the lr-scaled update.
__global__ __launch_bounds__(TPB) void rmsprop_tf_kernel(
    ChunkTable t, float lr, float alpha, float eps, float momentum) {
  const int c = find_chunk(t, blockIdx.x);
  const int nb = t.start_block[c + 1] - t.start_block[c];
  const int lb = blockIdx.x - t.start_block[c];
  const int n = t.numel[c];
  const float wd = t.wd[c];
  float* p = t.p[c];
  float* ms = t.m[c];
  float* mom = t.m2[c];
  const float* g = t.g[c];
  for (int64_t i = (int64_t(lb) * TPB + threadIdx.x) * VEC; i < n;
       i += int64_t(nb) * TPB * VEC) {
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      const int64_t j = i + e;
      if (j >= n) break;
      const float gv = g[j] + wd * p[j];
      const float msv = alpha * ms[j] + (1.f - alpha) * gv * gv;
      ms[j] = msv;
      const float u = gv / sqrtf(msv + eps);
      if (momentum != 0.f) {
        const float mv = momentum * mom[j] + lr * u;
        mom[j] = mv;
        p[j] -= mv;
      } else {
        p[j] -= lr * u;
      }
    }
  }
}

// shadow = d*shadow + (1-d)*param  (shadow fp32; param fp32 or bf16)
__global__ __launch_bounds__(TPB) void ema_kernel(ChunkTable t, float decay) {
  const int c = find_chunk(t, blockIdx.x);
  const int nb = t.start_block[c + 1] - t.start_block[c];
  const int lb = blockIdx.x - t.start_block[c];
  const int n = t.numel[c];
  float* s = t.p[c];
  const float* v = t.g[c];
  for (int64_t i = (int64_t(lb) * TPB + threadIdx.x) * VEC; i < n;
       i += int64_t(nb) * TPB * VEC) {
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      const int64_t j = i + e;
      if (j >= n) break;
      s[j] = decay * s[j] + (1.f - decay) * v[j];
    }
  }
}

// multi-tensor bf16 weight-pack refresh: after the fused optimizer writes
// the fp32 masters through raw pointers, ONE launch re-derives every
// cached kernel-layout bf16 pack (flat cast, or 2D transpose-cast for the
// dgrad/dw layouts) instead of ~95 per-tensor torch ops per step.
struct CastTable {
  const float* src[MAXC];
  ushort_t* dst[MAXC];
  int numel[MAXC];
  int adim[MAXC];  // 0 = flat cast; else dst is [A, B] = src[B, A]^T
  int start_block[MAXC + 1];
  int count;
};

__global__ __launch_bounds__(TPB) void repack_kernel(CastTable t) {
  int c = 0;
  while (c + 1 < t.count && int(blockIdx.x) >= t.start_block[c + 1]) ++c;
  const int nb = t.start_block[c + 1] - t.start_block[c];
  const int lb = blockIdx.x - t.start_block[c];
  const int n = t.numel[c];
  const int A = t.adim[c];
  const float* src = t.src[c];
  ushort_t* dst = t.dst[c];
  for (int64_t i = (int64_t(lb) * TPB + threadIdx.x) * VEC; i < n;
       i += int64_t(nb) * TPB * VEC) {
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      const int64_t j = i + e;
      if (j >= n) break;
      // weights are tiny (<1 MB total): the strided transpose read is
      // L2-served; no LDS tiling needed.
      // dst[a, b] = src[b, a]: a = j / B_cols, b = j % B_cols, where dst
      // is [A, n/A] — so B_cols = n / A
      const int bc = A > 0 ? n / A : 1;
      dst[j] = f2bf(A > 0 ? src[(j % bc) * A + (j / bc)] : src[j]);
    }
  }
}

}  // namespace

int update_max_chunks() { return MAXC; }

static int fill_blocks(ChunkTable& t) {
  t.start_block[0] = 0;
  for (int c = 0; c < t.count; ++c) {
    const int nb = std::max(1, std::min(
        256, int((int64_t(t.numel[c]) + TPB * VEC - 1) / (TPB * VEC))));
    t.start_block[c + 1] = t.start_block[c] + nb;
  }
  return t.start_block[t.count];
}

void fused_sgd(const float** g, float** p, float** m, const float* wd,
               const int* numel, int count, float lr, float momentum,
               bool nesterov, hipStream_t s) {
  ChunkTable t;
  t.count = count;
  for (int c = 0; c < count; ++c) {
    t.g[c] = g[c];
    t.p[c] = p[c];
    t.m[c] = m[c];
    t.m2[c] = nullptr;
    t.wd[c] = wd[c];
    t.numel[c] = numel[c];
  }
  const int blocks = fill_blocks(t);
  sgd_kernel<<<dim3(blocks), TPB, 0, s>>>(t, lr, momentum, nesterov);
}

void fused_rmsprop_tf(const float** g, float** p, float** ms, float** mom,
                      const float* wd, const int* numel, int count,
                      float lr, float alpha, float eps, float momentum,
                      hipStream_t s) {
  ChunkTable t;
  t.count = count;
  for (int c = 0; c < count; ++c) {
    t.g[c] = g[c];
    t.p[c] = p[c];
    t.m[c] = ms[c];
    t.m2[c] = mom[c];
    t.wd[c] = wd[c];
    t.numel[c] = numel[c];
  }
  const int blocks = fill_blocks(t);
  rmsprop_tf_kernel<<<dim3(blocks), TPB, 0, s>>>(t, lr, alpha, eps,
                                                 momentum);
}

void fused_repack(const float** src, void** dst, const int* numel,
                  const int* adim, int count, hipStream_t s) {
  CastTable t;
  t.count = count;
  int blocks = 0;
  t.start_block[0] = 0;
  for (int c = 0; c < count; ++c) {
    t.src[c] = src[c];
    t.dst[c] = static_cast<ushort_t*>(dst[c]);
    t.numel[c] = numel[c];
    t.adim[c] = adim[c];
    const int nb = std::max(1, std::min(
        64, int((int64_t(numel[c]) + TPB * VEC - 1) / (TPB * VEC))));
    t.start_block[c + 1] = t.start_block[c] + nb;
  }
  blocks = t.start_block[count];
  repack_kernel<<<dim3(blocks), TPB, 0, s>>>(t);
}

void fused_ema(const float** v, float** s_, const int* numel, int count,
               float decay, hipStream_t st) {
  ChunkTable t;
  t.count = count;
  for (int c = 0; c < count; ++c) {
    t.g[c] = v[c];
    t.p[c] = s_[c];
    t.m[c] = nullptr;
    t.m2[c] = nullptr;
    t.wd[c] = 0.f;
    t.numel[c] = numel[c];
  }
  const int blocks = fill_blocks(t);
  ema_kernel<<<dim3(blocks), TPB, 0, st>>>(t, decay);
}

}  // namespace yams
