// Depthwise conv (K3/K4) and stem dense conv (K1) for gfx950, NHWC bf16.
//
// NHWC is the natural depthwise layout on CDNA4: channels are the fastest
// axis, so a wave's 64 lanes reading one spatial position across channels
// is one coalesced transaction, and the per-channel filter taps broadcast
// from L1.  Forward/dgrad are flat vec8-per-thread elementwise kernels
// (memory-bound; the k*k window reuse is served by L1/L2); wgrad reduces
// deterministically via per-(chunk,mlane) fp32 slabs + slab_reduce.

#include <algorithm>

#include "common.h"
#include "engine.h"

namespace yams {

void slab_reduce(const float* ws, float* out, int64_t elems, int splits,
                 hipStream_t s);

namespace {

constexpr int MAX_BLOCKS = 4096;

// forward: 4 consecutive wo outputs per thread share the x row window in
// registers (k*k -> ~k*(k+3)/4 x loads per output).  FLIP reuses the same
// kernel as the stride-1 dgrad (correlation with the flipped filter).
template <int K, int S, bool FLIP>
__global__ __launch_bounds__(256) void dw_fwd_kernel(
    const ushort_t* __restrict__ x, const ushort_t* __restrict__ w,
    ushort_t* __restrict__ y, int N, int H, int W, int C, int Ho, int Wo,
    int pad, int64_t total8) {
  constexpr int XW = 3 * S + K;
  const int C8 = C >> 3;
  const int Wo4 = (Wo + 3) / 4;
  for (int64_t idx = int64_t(blockIdx.x) * blockDim.x + threadIdx.x;
       idx < total8; idx += int64_t(gridDim.x) * blockDim.x) {
    const int c8 = int(idx % C8);
    int64_t rest = idx / C8;
    const int w4 = int(rest % Wo4);
    rest /= Wo4;
    const int ho = int(rest % Ho);
    const int n = int(rest / Ho);
    const int c = c8 * 8;
    const int wo0 = w4 * 4;
    float acc[4][8] = {};
#pragma unroll
    for (int i = 0; i < K; ++i) {
      const int hi = ho * S + i - pad;
      if (hi < 0 || hi >= H) continue;
      float xr[XW][8];
      const int wbase = wo0 * S - pad;
#pragma unroll
      for (int t = 0; t < XW; ++t) {
        const int wi = wbase + t;
        if (wi >= 0 && wi < W) {
          short8_t v = *reinterpret_cast<const short8_t*>(
              &x[idx4(n, hi, wi, c, H, W, C)]);
#pragma unroll
          for (int e = 0; e < 8; ++e) xr[t][e] = bf2f(ushort_t(v[e]));
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) xr[t][e] = 0.f;
        }
      }
#pragma unroll
      for (int j = 0; j < K; ++j) {
        const int tap = FLIP ? (K - 1 - i) * K + (K - 1 - j) : i * K + j;
        short8_t wv = *reinterpret_cast<const short8_t*>(&w[tap * C + c]);
        float wf[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) wf[e] = bf2f(ushort_t(wv[e]));
#pragma unroll
        for (int q = 0; q < 4; ++q)
#pragma unroll
          for (int e = 0; e < 8; ++e)
            acc[q][e] = fmaf(xr[q * S + j][e], wf[e], acc[q][e]);
      }
    }
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      if (wo0 + q >= Wo) break;
      short8_t out;
#pragma unroll
      for (int e = 0; e < 8; ++e) out[e] = short(f2bf(acc[q][e]));
      *reinterpret_cast<short8_t*>(
          &y[idx4(n, ho, wo0 + q, c, Ho, Wo, C)]) = out;
    }
  }
}

// dgrad for stride 2 (stride 1 routes through dw_fwd_kernel<K,1,FLIP=true>)
template <int K>
__global__ __launch_bounds__(256) void dw_dgrad_s2_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ w,
    ushort_t* __restrict__ dx, int N, int H, int W, int C, int Ho, int Wo,
    int stride, int pad, int64_t total8) {
  const int C8 = C >> 3;
  for (int64_t idx = int64_t(blockIdx.x) * blockDim.x + threadIdx.x;
       idx < total8; idx += int64_t(gridDim.x) * blockDim.x) {
    const int c8 = int(idx % C8);
    int64_t rest = idx / C8;
    const int wi = int(rest % W);
    rest /= W;
    const int hi = int(rest % H);
    const int n = int(rest / H);
    const int c = c8 * 8;
    float acc[8] = {};
#pragma unroll
    for (int i = 0; i < K; ++i) {
      const int hnum = hi + pad - i;
      if (hnum < 0 || hnum % stride) continue;
      const int ho = hnum / stride;
      if (ho >= Ho) continue;
#pragma unroll
      for (int j = 0; j < K; ++j) {
        const int wnum = wi + pad - j;
        if (wnum < 0 || wnum % stride) continue;
        const int wo = wnum / stride;
        if (wo >= Wo) continue;
        short8_t gv = *reinterpret_cast<const short8_t*>(
            &dy[idx4(n, ho, wo, c, Ho, Wo, C)]);
        short8_t wv = *reinterpret_cast<const short8_t*>(
            &w[(i * K + j) * C + c]);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          acc[e] = fmaf(bf2f(ushort_t(gv[e])), bf2f(ushort_t(wv[e])), acc[e]);
      }
    }
    short8_t out;
#pragma unroll
    for (int e = 0; e < 8; ++e) out[e] = short(f2bf(acc[e]));
    *reinterpret_cast<short8_t*>(&dx[idx4(n, hi, wi, c, H, W, C)]) = out;
  }
}

// wgrad: thread = 4-channel vector lane; 4 consecutive wo outputs share
// one x row window in registers; per-chunk fp32 slab after an LDS fold
// over the m lanes.
template <int K, int S>
__global__ __launch_bounds__(256) void dw_wgrad_kernel(
    const ushort_t* __restrict__ x, const ushort_t* __restrict__ dy,
    float* __restrict__ ws, int N, int H, int W, int C, int Ho, int Wo,
    int pad, int chunks) {
  __shared__ float red[256 * 4];
  constexpr int XW = 3 * S + K;  // x window for 4 outputs
  const int C4 = C >> 2;
  const int CT4 = C4 < 256 ? C4 : 256;
  const int ML = 256 / CT4;
  const int c_tiles = (C4 + CT4 - 1) / CT4;
  const int c4 = (blockIdx.x % c_tiles) * CT4 + int(threadIdx.x) % CT4;
  const int chunk = blockIdx.x / c_tiles;
  const int mlane = int(threadIdx.x) / CT4;
  const int c = c4 * 4;

  const int Wo4 = (Wo + 3) / 4;
  const int64_t P = int64_t(N) * Ho * Wo4;
  const int64_t per = (P + chunks - 1) / chunks;
  const int64_t lo = chunk * per;
  const int64_t hi_ = std::min<int64_t>(lo + per, P);

  float acc[K * K][4] = {};
  if (c4 < C4 && mlane < ML) {
    for (int64_t p = lo + mlane; p < hi_; p += ML) {
      const int w4 = int(p % Wo4);
      int64_t rest = p / Wo4;
      const int ho = int(rest % Ho);
      const int n = int(rest / Ho);
      const int wo0 = w4 * 4;
      float g[4][4];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (wo0 + q < Wo) {
          short4_t g4 = *reinterpret_cast<const short4_t*>(
              &dy[idx4(n, ho, wo0 + q, c, Ho, Wo, C)]);
#pragma unroll
          for (int e = 0; e < 4; ++e) g[q][e] = bf2f(ushort_t(g4[e]));
        } else {
#pragma unroll
          for (int e = 0; e < 4; ++e) g[q][e] = 0.f;
        }
      }
#pragma unroll
      for (int i = 0; i < K; ++i) {
        const int hi2 = ho * S + i - pad;
        if (hi2 < 0 || hi2 >= H) continue;
        float xr[XW][4];
        const int wbase = wo0 * S - pad;
#pragma unroll
        for (int t = 0; t < XW; ++t) {
          const int wi = wbase + t;
          if (wi >= 0 && wi < W) {
            short4_t x4 = *reinterpret_cast<const short4_t*>(
                &x[idx4(n, hi2, wi, c, H, W, C)]);
#pragma unroll
            for (int e = 0; e < 4; ++e) xr[t][e] = bf2f(ushort_t(x4[e]));
          } else {
#pragma unroll
            for (int e = 0; e < 4; ++e) xr[t][e] = 0.f;
          }
        }
#pragma unroll
        for (int j = 0; j < K; ++j)
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            float sum = 0.f;
#pragma unroll
            for (int q = 0; q < 4; ++q)
              sum = fmaf(g[q][e], xr[q * S + j][e], sum);
            acc[i * K + j][e] += sum;
          }
      }
    }
  }
  float* slab = ws + int64_t(chunk) * K * K * C;
#pragma unroll
  for (int t = 0; t < K * K; ++t) {
#pragma unroll
    for (int e = 0; e < 4; ++e) red[threadIdx.x * 4 + e] = acc[t][e];
    __syncthreads();
    if (threadIdx.x < unsigned(CT4) && c4 < C4) {
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        float v = 0.f;
        for (int l = 0; l < ML; ++l) v += red[(threadIdx.x + l * CT4) * 4 + e];
        slab[t * C + c + e] = v;
      }
    }
    __syncthreads();
  }
}

// im2row for the stem (K1): x[N,H,W,3] -> patches[M, KP] with
// KP = round_up(k*k*3, 8) zero-padded columns, row m = (n,ho,wo).
// The stem then runs on the MFMA GEMM: y = patches @ w[KP,Cout]^T, and its
// wgrad on the split-M atb GEMM — no bespoke dense-conv kernels.
__global__ __launch_bounds__(256) void im2row_kernel(
    const ushort_t* __restrict__ x, ushort_t* __restrict__ patches,
    int N, int H, int W, int Ho, int Wo, int k, int stride, int pad,
    int KP, int64_t total) {
  const int Cin = 3;
  for (int64_t m = int64_t(blockIdx.x) * blockDim.x + threadIdx.x;
       m < total; m += int64_t(gridDim.x) * blockDim.x) {
    const int wo = int(m % Wo);
    int64_t rest = m / Wo;
    const int ho = int(rest % Ho);
    const int n = int(rest / Ho);
    ushort_t buf[32];
    int t = 0;
    for (int i = 0; i < k; ++i) {
      const int hi = ho * stride + i - pad;
      for (int j = 0; j < k; ++j) {
        const int wi = wo * stride + j - pad;
        const bool ok = hi >= 0 && hi < H && wi >= 0 && wi < W;
#pragma unroll
        for (int ci = 0; ci < 3; ++ci)
          buf[t++] = ok ? x[idx4(n, hi, wi, ci, H, W, Cin)] : ushort_t(0);
      }
    }
    for (; t < KP; ++t) buf[t] = ushort_t(0);
    ushort_t* dst = &patches[m * KP];
#pragma unroll
    for (int v = 0; v < 32; v += 8) {
      if (v >= KP) break;
      *reinterpret_cast<short8_t*>(&dst[v]) =
          *reinterpret_cast<const short8_t*>(&buf[v]);
    }
  }
}

int grid_for(int64_t total) {
  return int(std::min<int64_t>((total + 255) / 256, MAX_BLOCKS));
}

}  // namespace

void dw_conv_fwd(const void* x, const void* w, void* y, int N, int H, int W,
                 int C, int Ho, int Wo, int k, int stride, int pad,
                 hipStream_t s) {
  const int64_t total8 = int64_t(N) * Ho * ((Wo + 3) / 4) * (C >> 3);
  auto* xp = static_cast<const ushort_t*>(x);
  auto* wp = static_cast<const ushort_t*>(w);
  auto* yp = static_cast<ushort_t*>(y);
  dim3 g(grid_for(total8));
  if (k == 3 && stride == 1)
    dw_fwd_kernel<3, 1, false><<<g, 256, 0, s>>>(xp, wp, yp, N, H, W, C, Ho,
                                                 Wo, pad, total8);
  else if (k == 3 && stride == 2)
    dw_fwd_kernel<3, 2, false><<<g, 256, 0, s>>>(xp, wp, yp, N, H, W, C, Ho,
                                                 Wo, pad, total8);
  else if (k == 5 && stride == 1)
    dw_fwd_kernel<5, 1, false><<<g, 256, 0, s>>>(xp, wp, yp, N, H, W, C, Ho,
                                                 Wo, pad, total8);
  else
    dw_fwd_kernel<5, 2, false><<<g, 256, 0, s>>>(xp, wp, yp, N, H, W, C, Ho,
                                                 Wo, pad, total8);
}

void dw_conv_dgrad(const void* dy, const void* w, void* dx, int N, int H,
                   int W, int C, int Ho, int Wo, int k, int stride, int pad,
                   hipStream_t s) {
  auto* gp = static_cast<const ushort_t*>(dy);
  auto* wp = static_cast<const ushort_t*>(w);
  auto* xp = static_cast<ushort_t*>(dx);
  if (stride == 1) {
    // dgrad == correlation of dy with the flipped filter at same pad
    const int64_t total8 = int64_t(N) * H * ((W + 3) / 4) * (C >> 3);
    dim3 g(grid_for(total8));
    if (k == 3)
      dw_fwd_kernel<3, 1, true><<<g, 256, 0, s>>>(gp, wp, xp, N, Ho, Wo, C,
                                                  H, W, pad, total8);
    else
      dw_fwd_kernel<5, 1, true><<<g, 256, 0, s>>>(gp, wp, xp, N, Ho, Wo, C,
                                                  H, W, pad, total8);
    return;
  }
  const int64_t total8 = int64_t(N) * H * W * (C >> 3);
  if (k == 3)
    dw_dgrad_s2_kernel<3><<<grid_for(total8), 256, 0, s>>>(
        gp, wp, xp, N, H, W, C, Ho, Wo, stride, pad, total8);
  else
    dw_dgrad_s2_kernel<5><<<grid_for(total8), 256, 0, s>>>(
        gp, wp, xp, N, H, W, C, Ho, Wo, stride, pad, total8);
}

int dw_wgrad_chunks(int N, int Ho, int Wo) {
  const int64_t P = int64_t(N) * Ho * ((Wo + 3) / 4);
  int c = int(std::min<int64_t>((P + 3) / 4, 4096));
  return c < 1 ? 1 : c;
}

int64_t dw_wgrad_ws_elems(int N, int Ho, int Wo, int C, int k) {
  const int c_tiles = ((C >> 2) + 255) / 256;
  return int64_t((dw_wgrad_chunks(N, Ho, Wo) + c_tiles - 1) / c_tiles)
         * k * k * C;
}

void dw_conv_wgrad(const void* x, const void* dy, float* dwgt, float* ws,
                   int N, int H, int W, int C, int Ho, int Wo, int k,
                   int stride, int pad, hipStream_t s) {
  const int C4 = C >> 2;
  const int CT4 = C4 < 256 ? C4 : 256;
  const int c_tiles = (C4 + CT4 - 1) / CT4;
  const int chunks = (dw_wgrad_chunks(N, Ho, Wo) + c_tiles - 1) / c_tiles;
  auto* xp = static_cast<const ushort_t*>(x);
  auto* gp = static_cast<const ushort_t*>(dy);
  dim3 grid(c_tiles * chunks);
  if (k == 3 && stride == 1)
    dw_wgrad_kernel<3, 1><<<grid, 256, 0, s>>>(xp, gp, ws, N, H, W, C, Ho,
                                               Wo, pad, chunks);
  else if (k == 3 && stride == 2)
    dw_wgrad_kernel<3, 2><<<grid, 256, 0, s>>>(xp, gp, ws, N, H, W, C, Ho,
                                               Wo, pad, chunks);
  else if (k == 5 && stride == 1)
    dw_wgrad_kernel<5, 1><<<grid, 256, 0, s>>>(xp, gp, ws, N, H, W, C, Ho,
                                               Wo, pad, chunks);
  else
    dw_wgrad_kernel<5, 2><<<grid, 256, 0, s>>>(xp, gp, ws, N, H, W, C, Ho,
                                               Wo, pad, chunks);
  slab_reduce(ws, dwgt, int64_t(k) * k * C, chunks, s);
}

void im2row(const void* x, void* patches, int N, int H, int W, int Ho,
            int Wo, int k, int stride, int pad, int KP, hipStream_t s) {
  const int64_t total = int64_t(N) * Ho * Wo;
  im2row_kernel<<<grid_for((total + 3) / 4), 256, 0, s>>>(
      static_cast<const ushort_t*>(x), static_cast<ushort_t*>(patches),
      N, H, W, Ho, Wo, k, stride, pad, KP, total);
}

}  // namespace yams
