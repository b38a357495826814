// Depthwise conv (K3/K4) and stem dense conv (K1) for gfx950, NHWC bf16.
//
// NHWC is the natural depthwise layout on CDNA4: channels are the fastest
// axis, so a wave's 64 lanes reading one spatial position across channels
// is one coalesced transaction, and the per-channel filter taps broadcast
// from L1.  Forward/dgrad are flat vec8-per-thread elementwise kernels
// (memory-bound; the k*k window reuse is served by L1/L2); wgrad reduces
// deterministically via per-(chunk,mlane) fp32 slabs + slab_reduce.

#include <algorithm>
#include <cstdlib>

#include "common.h"
#include "engine.h"

namespace yams {

void slab_reduce_dwt(const float* ws, float* out, int C, int KK,
                     int splits, hipStream_t s);

namespace {

constexpr int MAX_BLOCKS = 4096;

inline Lop to_lop(const LopDesc& d) {
  Lop L;
  L.kind = d.kind;
  L.act = d.act;
  L.coef = d.coef;
  L.z = static_cast<const ushort_t*>(d.z);
  L.C = d.C;
  return L;
}

// forward: each thread computes a TH x 4 (ho x wo) output tile for 4
// channels, so every input row is loaded ONCE per thread and reused by
// the K output rows it feeds (the row-per-thread version re-read x up to
// K times through L2 — the dominant cost on these latency-bound shapes).
// FLIP reuses the kernel as the stride-1 dgrad; lopx fuses the producer
// BN+act (or the BNBWD dz for dgrad) into the loads.
// EPI: 0 none, 1 forward BN stats of the output, 2 BN-BACKWARD stats:
// FLIP dgrad output dy feeds a BN backward, and the epilogue accumulates
// (sum da, sum da*z) with da = dy * act'(a*z+b) so no separate
// bn_bwd_partial pass reads dy+z again.  lopE carries z/ab/act for EPI==2.
template <int K, int S, bool FLIP, int EPI, int LOPK, int ACT,
          bool WL = false>
__global__ __launch_bounds__(256) void dw_fwd_kernel(
    const ushort_t* __restrict__ x, const ushort_t* __restrict__ w,
    ushort_t* __restrict__ y, int N, int H, int W, int C, int Ho, int Wo,
    int pad, Lop lopx, Lop lopE, float* __restrict__ stats_ws, int chunks) {
  __shared__ float red[EPI ? 256 * 4 : 4];
  constexpr int XW = 3 * S + K;        // x cols for 4 wo outputs
  constexpr int TH = (S == 1 && K == 3) ? 4 : 2;  // rows per thread
  // (TH=4 for k5 measured 4x SLOWER: xr+acc registers spill at 256
  // threads — the k5 fix is the traversal order below, not deeper tiles)
  constexpr int RR = (TH - 1) * S + K; // input rows touched
  // k5's 6-row window thrashes L1 when quads walk along w (row reuse
  // distance = a full row of quads ~40-80 KB); walking DOWN the column
  // stripe (hg fastest) keeps the reuse window at XW cols x RR rows
  // (~12 KB) — measured 1.1-1.6 TB/s -> see profiles/r02 A/B
  constexpr bool HGFAST = (K == 5);
  const int C8 = C >> 2;
  // WL (k5): the register-cached weight decode costs 256 VGPR + 148
  // AGPR per thread = ONE wave/SIMD (objdump count, docs/ROADMAP §3b);
  // staging the decoded f32 filter bank in LDS instead caps the
  // register demand (LDS is never register-promoted) at the price of a
  // ds_read_b128 per tap use.  CT8 is capped at 64 so the bank fits.
  const int CT8 = WL ? (C8 < 64 ? C8 : 64) : (C8 < 256 ? C8 : 256);
  const int ML = 256 / CT8;
  const int c_tiles = (C8 + CT8 - 1) / CT8;
  const int c8 = (blockIdx.x % c_tiles) * CT8 + int(threadIdx.x) % CT8;
  const int chunk = blockIdx.x / c_tiles;
  const int mlane = int(threadIdx.x) / CT8;
  const int c = c8 * 4;
  __shared__ float wlds[WL ? 64 * K * K * 4 : 4];
  if (WL) {
    const int cbase = (blockIdx.x % c_tiles) * CT8;
    for (int idx = int(threadIdx.x); idx < CT8 * K * K; idx += 256) {
      const int c8l = idx / (K * K);
      const int tap = idx % (K * K);
      const int cg = cbase + c8l;
      if (cg < C8) {
        // bake the FLIP into the bank so the inner loop indexes i*K+j
        const int tsrc = FLIP ? K * K - 1 - tap : tap;
        const short4_t wv =
            *reinterpret_cast<const short4_t*>(&w[tsrc * C + cg * 4]);
#pragma unroll
        for (int e = 0; e < 4; ++e)
          wlds[(c8l * K * K + tap) * 4 + e] = bf2f(ushort_t(wv[e]));
      }
    }
    __syncthreads();
  }

  const int Wo4 = (Wo + 3) / 4;
  const int HoG = (Ho + TH - 1) / TH;
  // 32-bit position decode: P = N*HoG*Wo4 < 2^31 always for these nets,
  // and a 64-bit div is ~4x the instructions of a 32-bit one on gfx9
  const int P = N * HoG * Wo4;
  const int per = (P + chunks - 1) / chunks;
  const int lo = chunk * per;
  const int hi_ = min(lo + per, P);

  float s1[4] = {}, s2[4] = {};
  float ea[4], eb[4];
  LopRegT<LOPK, ACT, 4> lr;
  if (c8 < C8 && mlane < ML) {
    lr.init(lopx, c);
    if (EPI == 2) {
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        ea[e] = lopE.coef[c + e];
        eb[e] = lopE.coef[C + c + e];
      }
    }
    for (int p = lo + mlane; p < hi_; p += ML) {
      int w4, hg, n;
      if (HGFAST) {
        hg = p % HoG;
        const int rest = p / HoG;
        w4 = rest % Wo4;
        n = rest / Wo4;
      } else {
        w4 = p % Wo4;
        const int rest = p / Wo4;
        hg = rest % HoG;
        n = rest / HoG;
      }
      const int ho0 = hg * TH;
      const int wo0 = w4 * 4;
      const int wbase = wo0 * S - pad;
      const int64_t imgbase = int64_t(n) * H * W * C + c;
      // packed-f32 accumulators: the inner math runs on v_pk_fma_f32
      // (2 MACs/instruction) — the scalar version measured VALU-bound
      // (SQ_INSTS_VALU ~56% of all SIMD cycles, profiles/r02)
      f32x2 acc[TH][4][2] = {};
      // column clamps are row-invariant: hoist out of the r loop
      int coff[XW];
      bool cok[XW];
#pragma unroll
      for (int t = 0; t < XW; ++t) {
        const int wi = wbase + t;
        const int wic = wi < 0 ? 0 : (wi >= W ? W - 1 : wi);
        cok[t] = wi >= 0 && wi < W;
        coff[t] = wic * C;
      }
      int rr = RR;
      // WL: opaque bound keeps the row loop ROLLED — full unrolling is
      // what drives the load hoisting + weight caching that exhausts
      // the register file; with 2+ waves/SIMD the waves hide the row
      // load latency instead of one wave's software pipeline.
      if (WL) asm volatile("" : "+r"(rr));
#pragma unroll
      for (int r = 0; r < rr; ++r) {
        const int hi = ho0 * S + r - pad;
        const int hic = hi < 0 ? 0 : (hi >= H ? H - 1 : hi);
        const bool row_ok = hi >= 0 && hi < H;
        const ushort_t* xrow = x + imgbase + int64_t(hic) * W * C;
        const ushort_t* zrow =
            LOPK == LOP_BNBWD ? lopx.z + imgbase + int64_t(hic) * W * C
                              : nullptr;
        f32x2 xr[XW][2];
        // branchless: always load from a CLAMPED in-bounds address and
        // zero the raw bf16 vector by ONE select — a per-element branch
        // around a load makes hipcc wait vmcnt(0) per load
#pragma unroll
        for (int t = 0; t < XW; ++t) {
          short4_t v = *reinterpret_cast<const short4_t*>(&xrow[coff[t]]);
          const bool keep = cok[t] && row_ok;
          if (!keep) v = short4_t{0, 0, 0, 0};
          if (LOPK == LOP_NONE) {
#pragma unroll
            for (int pp = 0; pp < 2; ++pp) {
              xr[t][pp][0] = bf2f(ushort_t(v[pp * 2]));
              xr[t][pp][1] = bf2f(ushort_t(v[pp * 2 + 1]));
            }
          } else if (LOPK == LOP_BNBWD) {
            short4_t zv = *reinterpret_cast<const short4_t*>(
                &zrow[coff[t]]);
            if (!keep) zv = short4_t{0, 0, 0, 0};
#pragma unroll
            for (int e = 0; e < 4; ++e)
              xr[t][e >> 1][e & 1] = lr.apply(bf2f(ushort_t(v[e])),
                                              bf2f(ushort_t(zv[e])), e);
          } else {  // LOP_BNACT
#pragma unroll
            for (int e = 0; e < 4; ++e)
              xr[t][e >> 1][e & 1] =
                  lr.apply(bf2f(ushort_t(v[e])), 0.f, e);
          }
        }
#pragma unroll
        for (int t = 0; t < TH; ++t) {
          const int i = r - t * S;
          if (i < 0 || i >= K) continue;
#pragma unroll
          for (int j = 0; j < K; ++j) {
            f32x2 wf[2];
            if (WL) {
              const float4 wf4 = *reinterpret_cast<const float4*>(
                  &wlds[((int(threadIdx.x) % CT8) * K * K + i * K + j) * 4]);
              wf[0][0] = wf4.x;
              wf[0][1] = wf4.y;
              wf[1][0] = wf4.z;
              wf[1][1] = wf4.w;
            } else {
              const int tap =
                  FLIP ? (K - 1 - i) * K + (K - 1 - j) : i * K + j;
              short4_t wv =
                  *reinterpret_cast<const short4_t*>(&w[tap * C + c]);
#pragma unroll
              for (int pp = 0; pp < 2; ++pp) {
                wf[pp][0] = bf2f(ushort_t(wv[pp * 2]));
                wf[pp][1] = bf2f(ushort_t(wv[pp * 2 + 1]));
              }
            }
#pragma unroll
            for (int q = 0; q < 4; ++q)
#pragma unroll
              for (int pp = 0; pp < 2; ++pp)
                acc[t][q][pp] += xr[q * S + j][pp] * wf[pp];
          }
        }
      }
#pragma unroll
      for (int t = 0; t < TH; ++t) {
        if (ho0 + t >= Ho) break;
        const int64_t obase =
            (int64_t(n) * Ho + ho0 + t) * int64_t(Wo) * C +
            int64_t(wo0) * C + c;
        ushort_t* yout = y + obase;
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          if (wo0 + q >= Wo) break;
          short4_t out;
          short4_t zq;
          if (EPI == 2)
            zq = *reinterpret_cast<const short4_t*>(&lopE.z[obase + q * C]);
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            const ushort_t o = f2bf(acc[t][q][e >> 1][e & 1]);
            out[e] = short(o);
            if (EPI == 1) {
              const float v = bf2f(o);
              s1[e] += v;
              s2[e] = fmaf(v, v, s2[e]);
            } else if (EPI == 2) {
              const float v = bf2f(o);
              const float zv = bf2f(ushort_t(zq[e]));
              const float da = v * act_bwd<ACT>(fmaf(zv, ea[e], eb[e]));
              s1[e] += da;
              s2[e] = fmaf(da, zv, s2[e]);
            }
          }
          *reinterpret_cast<short4_t*>(&yout[q * C]) = out;
        }
      }
    }
  }
  if (EPI) {
#pragma unroll
    for (int e = 0; e < 4; ++e) red[threadIdx.x * 4 + e] = s1[e];
    __syncthreads();
    if (threadIdx.x < unsigned(CT8) && c8 < C8) {
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        float t = 0.f;
        for (int l = 0; l < ML; ++l)
          t += red[(threadIdx.x + l * CT8) * 4 + e];
        stats_ws[int64_t(chunk) * C + c + e] = t;
      }
    }
    __syncthreads();
#pragma unroll
    for (int e = 0; e < 4; ++e) red[threadIdx.x * 4 + e] = s2[e];
    __syncthreads();
    if (threadIdx.x < unsigned(CT8) && c8 < C8) {
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        float t = 0.f;
        for (int l = 0; l < ML; ++l)
          t += red[(threadIdx.x + l * CT8) * 4 + e];
        stats_ws[(int64_t(chunks) + chunk) * C + c + e] = t;
      }
    }
  }
}

// dgrad for stride 2 (stride 1 routes through dw_fwd_kernel FLIP=true)
template <int K, int LOPK, int ACT>
__global__ __launch_bounds__(256) void dw_dgrad_s2_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ w,
    ushort_t* __restrict__ dx, int N, int H, int W, int C, int Ho, int Wo,
    int stride, int pad, Lop lopdy, int64_t total8) {
  // stride==2 only.  Tap parity is fixed by (hi+pad)&1 / (wi+pad)&1, so
  // iterate ONLY the parity-valid taps (no per-tap %), and mask range
  // edges with clamped in-bounds loads + value select (a per-tap branch
  // around the load serializes on vmcnt).  Index decode is 32-bit.
  const int C8 = C >> 3;
  const int tot = int(total8);
  for (int idx = int(blockIdx.x) * 256 + int(threadIdx.x); idx < tot;
       idx += int(gridDim.x) * 256) {
    const int c8 = idx % C8;
    int rest = idx / C8;
    const int wi = rest % W;
    rest /= W;
    const int hi = rest % H;
    const int n = rest / H;
    const int c = c8 * 8;
    LopRegT<LOPK, ACT, 8> lr;
    lr.init(lopdy, c);
    f32x2 acc[4] = {};  // packed pairs: taps accumulate on v_pk_fma_f32
    const int ip0 = (hi + pad) & 1;
    const int jp0 = (wi + pad) & 1;
#pragma unroll
    for (int ii = 0; ii < (K + 1) / 2; ++ii) {
      const int i = ip0 + ii * 2;
      if (i >= K) break;
      const int ho = (hi + pad - i) >> 1;      // exact by parity
      const bool hok = ho >= 0 && ho < Ho;
      const int hoc = ho < 0 ? 0 : (ho >= Ho ? Ho - 1 : ho);
#pragma unroll
      for (int jj = 0; jj < (K + 1) / 2; ++jj) {
        const int j = jp0 + jj * 2;
        if (j >= K) break;
        const int wo = (wi + pad - j) >> 1;
        const bool ok = hok && wo >= 0 && wo < Wo;
        const int woc = wo < 0 ? 0 : (wo >= Wo ? Wo - 1 : wo);
        const int64_t off = idx4(n, hoc, woc, c, Ho, Wo, C);
        short8_t gv = *reinterpret_cast<const short8_t*>(&dy[off]);
        if (!ok) gv = short8_t{0, 0, 0, 0, 0, 0, 0, 0};
        short8_t wv = *reinterpret_cast<const short8_t*>(
            &w[(i * K + j) * C + c]);
        if (LOPK == LOP_NONE) {
#pragma unroll
          for (int pp = 0; pp < 4; ++pp) {
            f32x2 g2 = {bf2f(ushort_t(gv[pp * 2])),
                        bf2f(ushort_t(gv[pp * 2 + 1]))};
            f32x2 w2v = {bf2f(ushort_t(wv[pp * 2])),
                         bf2f(ushort_t(wv[pp * 2 + 1]))};
            acc[pp] += g2 * w2v;
          }
        } else {
          // (gv pre-zeroing is NOT a valid mask for the BNBWD lop — its
          // affine terms are nonzero at 0 input — so mask the applied
          // value explicitly)
          short8_t zv = *reinterpret_cast<const short8_t*>(&lopdy.z[off]);
#pragma unroll
          for (int e = 0; e < 8; ++e)
            acc[e >> 1][e & 1] =
                fmaf(ok ? lr.apply(bf2f(ushort_t(gv[e])),
                                   bf2f(ushort_t(zv[e])), e)
                        : 0.f,
                     bf2f(ushort_t(wv[e])), acc[e >> 1][e & 1]);
        }
      }
    }
    short8_t out;
#pragma unroll
    for (int e = 0; e < 8; ++e) out[e] = short(f2bf(acc[e >> 1][e & 1]));
    *reinterpret_cast<short8_t*>(&dx[idx4(n, hi, wi, c, H, W, C)]) = out;
  }
}

// wgrad: thread = 4-channel vector lane; 4 consecutive wo outputs share
// one x row window in registers; per-chunk fp32 slab after an LDS fold
// over the m lanes.
// IP: kernel-row parts — K=5 accumulates 25 taps x 4ch = 100 VGPRs in one
// block, capping occupancy at 2 waves/SIMD; splitting the K rows across IP
// blocks (rows re-read dy, x-row traffic unchanged) trades a little dy
// bandwidth for 2x occupancy on the latency-bound load stream.
template <int K, int S, int LX, int LXACT, int LD, int IP>
__global__ __launch_bounds__(256) void dw_wgrad_kernel(
    const ushort_t* __restrict__ x, const ushort_t* __restrict__ dy,
    float* __restrict__ ws, int N, int H, int W, int C, int Ho, int Wo,
    int pad, Lop lopx, Lop lopdy, int chunks) {
  __shared__ float red[256 * 4];
  constexpr int XW = 3 * S + K;  // x window for 4 outputs
  const int C4 = C >> 2;
  const int CT4 = C4 < 256 ? C4 : 256;
  const int ML = 256 / CT4;
  const int c_tiles = (C4 + CT4 - 1) / CT4;
  constexpr int KR = (K + IP - 1) / IP;  // kernel rows per part
  const int c4 = (blockIdx.x % c_tiles) * CT4 + int(threadIdx.x) % CT4;
  const int part = int(blockIdx.x / c_tiles) % IP;
  const int chunk = blockIdx.x / (c_tiles * IP);
  const int mlane = int(threadIdx.x) / CT4;
  const int c = c4 * 4;
  const int i0 = part * KR;

  const int Wo4 = (Wo + 3) / 4;
  const int P = N * Ho * Wo4;   // < 2^31; 32-bit div is 4x cheaper
  const int per = (P + chunks - 1) / chunks;
  const int lo = chunk * per;
  const int hi_ = min(lo + per, P);

  float acc[KR * K][4] = {};
  LopRegT<LX, LXACT, 4> lrx;
  LopRegT<LD, 0, 4> lrd;
  if (c4 < C4 && mlane < ML) {
    lrx.init(lopx, c);
    lrd.init(lopdy, c);
    for (int p = lo + mlane; p < hi_; p += ML) {
      // k5: walk down the column stripe (ho fastest) so the K-row
      // re-reads stay in L1 (same reasoning as dw_fwd_kernel HGFAST)
      int w4, ho, n;
      if (K == 5) {
        ho = p % Ho;
        const int rest = p / Ho;
        w4 = rest % Wo4;
        n = rest / Wo4;
      } else {
        w4 = p % Wo4;
        const int rest = p / Wo4;
        ho = rest % Ho;
        n = rest / Ho;
      }
      const int wo0 = w4 * 4;
      const ushort_t* dyrow =
          dy + (int64_t(n) * Ho + ho) * int64_t(Wo) * C + c;
      const ushort_t* dzrow =
          LD == LOP_BNBWD
              ? lopdy.z + (int64_t(n) * Ho + ho) * int64_t(Wo) * C + c
              : nullptr;
      float g[4][4];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const int wo = wo0 + q;
        const int woc = wo < Wo ? wo : Wo - 1;
        const bool ok = wo < Wo;
        const int off = woc * C;
        short4_t g4 = *reinterpret_cast<const short4_t*>(&dyrow[off]);
        if (LD == LOP_NONE) {
#pragma unroll
          for (int e = 0; e < 4; ++e)
            g[q][e] = ok ? bf2f(ushort_t(g4[e])) : 0.f;
        } else {
          short4_t z4 = *reinterpret_cast<const short4_t*>(&dzrow[off]);
#pragma unroll
          for (int e = 0; e < 4; ++e)
            g[q][e] = ok ? lrd.apply(bf2f(ushort_t(g4[e])),
                                     bf2f(ushort_t(z4[e])), e)
                         : 0.f;
        }
      }
      const int wbase = wo0 * S - pad;
      const int64_t imgbase = int64_t(n) * H * W * C + c;
      // column clamps are kernel-row invariant: hoist
      int coff[XW];
      bool cok[XW];
#pragma unroll
      for (int t = 0; t < XW; ++t) {
        const int wi = wbase + t;
        const int wic = wi < 0 ? 0 : (wi >= W ? W - 1 : wi);
        cok[t] = wi >= 0 && wi < W;
        coff[t] = wic * C;
      }
#pragma unroll
      for (int ii = 0; ii < KR; ++ii) {
        const int i = i0 + ii;
        if (IP > 1 && i >= K) break;
        const int hi2 = ho * S + i - pad;
        const int hic = hi2 < 0 ? 0 : (hi2 >= H ? H - 1 : hi2);
        const bool row_ok = hi2 >= 0 && hi2 < H;
        const ushort_t* xrow = x + imgbase + int64_t(hic) * W * C;
        float xr[XW][4];
#pragma unroll
        for (int t = 0; t < XW; ++t) {
          const bool keep = row_ok && cok[t];
          short4_t x4 = *reinterpret_cast<const short4_t*>(
              &xrow[coff[t]]);
          if (LX == LOP_NONE) {
#pragma unroll
            for (int e = 0; e < 4; ++e)
              xr[t][e] = keep ? bf2f(ushort_t(x4[e])) : 0.f;
          } else {
#pragma unroll
            for (int e = 0; e < 4; ++e)
              xr[t][e] = keep ? lrx.apply(bf2f(ushort_t(x4[e])), 0.f, e)
                              : 0.f;
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
            acc[ii * K + j][e] += sum;
          }
      }
    }
  }
  float* slab = ws + int64_t(chunk) * K * K * C;
#pragma unroll
  for (int t = 0; t < KR * K; ++t) {
    const int tap = (i0 + t / K) * K + t % K;
    if (IP > 1 && tap >= K * K) break;
#pragma unroll
    for (int e = 0; e < 4; ++e) red[threadIdx.x * 4 + e] = acc[t][e];
    __syncthreads();
    if (threadIdx.x < unsigned(CT4) && c4 < C4) {
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        float v = 0.f;
        for (int l = 0; l < ML; ++l) v += red[(threadIdx.x + l * CT4) * 4 + e];
        slab[tap * C + c + e] = v;
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
      const int hic = hi < 0 ? 0 : (hi >= H ? H - 1 : hi);
      for (int j = 0; j < k; ++j) {
        const int wi = wo * stride + j - pad;
        const int wic = wi < 0 ? 0 : (wi >= W ? W - 1 : wi);
        const bool ok = hi >= 0 && hi < H && wi >= 0 && wi < W;
#pragma unroll
        for (int ci = 0; ci < 3; ++ci) {
          const ushort_t v = x[idx4(n, hic, wic, ci, H, W, Cin)];
          buf[t++] = ok ? v : ushort_t(0);
        }
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

int dw_stats_chunks(int N, int Ho, int Wo, int C, int stride, int k,
                    bool wl) {
  (void)stride;
  // measurement knob (roadmap §3): YAMS_DW_CHUNKS_DIV divides the block
  // target so the work-per-wave vs launch-count tradeoff can be swept
  // without a rebuild.  Unset => the per-shape table below.
  static int div_ = [] {
    const char* e = getenv("YAMS_DW_CHUNKS_DIV");
    return e ? atoi(e) : 0;
  }();
  const int C8 = C >> 2;
  const int CT8 = wl ? (C8 < 64 ? C8 : 64) : (C8 < 256 ? C8 : 256);
  const int c_tiles = (C8 + CT8 - 1) / CT8;
  const int64_t P = int64_t(N) * ((Ho + 1) / 2) * ((Wo + 3) / 4);
  // per-shape block target from the round-1 chunks sweep
  // (tools/probe_dw_sweep.py; per-wave prologue + launch ramp dominate
  // when waves run ~1 loop iteration): k5 prefers 4x fewer chunks than
  // the 512 base, k3 at small spatial 1x (512); env div overrides.
  // The WL k5 variant runs at ~134 VGPR (3 blocks/CU) instead of the
  // register-saturated 1 block/CU the 256 target was tuned for, so its
  // grid target is 3x — without this the extra occupancy sits idle
  // (measured: the c_tiles==1 shapes did not move under WL alone).
  int64_t target = 512;
  if (div_ <= 0) {
    if (k == 5) {
      // WL wants ~3 blocks/CU (768) for its occupancy, but only while
      // every thread still gets >=1.5 position iterations — per-wave
      // prologue (LDS bank staging + sync) dominates below that.
      // iterations at G blocks = P*C8/(G*256), so G <= P*C8/384.
      // Fixed 768 measured -10% on V3-Small's thin 14² k5 layers.
      target = wl ? std::min<int64_t>(
                        768, std::max<int64_t>(256, P * C8 / 384))
                  : 256;
    }
    if (k == 3 && wl) target = 1024;
  } else {
    target = 2048 / div_;
  }
  int64_t want = target / c_tiles;
  int64_t maxc = P;
  int c = int(std::min(std::max<int64_t>(want, 1), maxc));
  return c < 1 ? 1 : c;
}

// YAMS_DW_K5LDS=0 disables the LDS filter-bank k5 variant (see the WL
// template comment in dw_fwd_kernel — default on, A/B in profiles/r02)
bool dw_k5_wlds() {
  static const bool on = [] {
    const char* e = getenv("YAMS_DW_K5LDS");
    return !(e && e[0] == '0');
  }();
  return on;
}

// YAMS_DW_K3LDS=1 extends the LDS filter bank to k3 (9.2 KB bank; k3
// sits at ~200-256 VGPR = 2 waves/SIMD — default off pending the A/B)
bool dw_k3_wlds() {
  static const bool on = [] {
    const char* e = getenv("YAMS_DW_K3LDS");
    return e && e[0] == '1';
  }();
  return on;
}

void dw_conv_fwd(const void* x, const void* w, void* y, int N, int H, int W,
                 int C, int Ho, int Wo, int k, int stride, int pad,
                 LopDesc lopx, float* stats_ws, hipStream_t s) {
  auto* xp = static_cast<const ushort_t*>(x);
  auto* wp = static_cast<const ushort_t*>(w);
  auto* yp = static_cast<ushort_t*>(y);
  const Lop L = to_lop(lopx);
  const bool wl = k == 5 ? dw_k5_wlds() : (k == 3 && dw_k3_wlds());
  const int C8 = C >> 2;
  const int CT8 = wl ? (C8 < 64 ? C8 : 64) : (C8 < 256 ? C8 : 256);
  const int c_tiles = (C8 + CT8 - 1) / CT8;
  const int chunks = dw_stats_chunks(N, Ho, Wo, C, stride, k, wl);
  dim3 g(c_tiles * chunks);
  // fwd lop kinds: none, or bnact with act in {relu, relu6, hswish}
  Lop LE{};
#define DWF_CASE(KK, SS, ST, LK, AC, WLW)                                  \
  if (k == KK && stride == SS && wl == WLW &&                              \
      (stats_ws != nullptr) == (ST != 0) &&                                \
      lopx.kind == LK && (LK == 0 || lopx.act == AC)) {                    \
    dw_fwd_kernel<KK, SS, false, ST, LK, AC, WLW><<<g, 256, 0, s>>>(       \
        xp, wp, yp, N, H, W, C, Ho, Wo, pad, L, LE, stats_ws, chunks);     \
    return;                                                                \
  }
#define DWF_ST(KK, SS, ST, WLW)                                            \
  DWF_CASE(KK, SS, ST, 0, 0, WLW) DWF_CASE(KK, SS, ST, 1, 1, WLW)          \
  DWF_CASE(KK, SS, ST, 1, 2, WLW) DWF_CASE(KK, SS, ST, 1, 3, WLW)
#define DWF_KS(KK, SS, WLW)                                                \
  DWF_ST(KK, SS, false, WLW) DWF_ST(KK, SS, true, WLW)
  DWF_KS(3, 1, false) DWF_KS(3, 2, false)
  DWF_KS(5, 1, false) DWF_KS(5, 2, false)
  DWF_KS(3, 1, true) DWF_KS(3, 2, true)
  DWF_KS(5, 1, true) DWF_KS(5, 2, true)
#undef DWF_KS
#undef DWF_ST
#undef DWF_CASE
  yams_launch_fail("dw_conv_fwd", k, stride, lopx.kind, lopx.act);
}

void dw_conv_dgrad(const void* dy, const void* w, void* dx, int N, int H,
                   int W, int C, int Ho, int Wo, int k, int stride, int pad,
                   LopDesc lopdy, LopDesc epi, float* stats_ws,
                   hipStream_t s) {
  auto* gp = static_cast<const ushort_t*>(dy);
  auto* wp = static_cast<const ushort_t*>(w);
  auto* xp = static_cast<ushort_t*>(dx);
  const Lop L = to_lop(lopdy);
  const Lop LE = to_lop(epi);
  if (stride == 1) {
    // dgrad == correlation of dy with the flipped filter at same pad
    const bool wl = k == 5 ? dw_k5_wlds() : (k == 3 && dw_k3_wlds());
    const int C8 = C >> 2;
    const int CT8 = wl ? (C8 < 64 ? C8 : 64) : (C8 < 256 ? C8 : 256);
    const int c_tiles = (C8 + CT8 - 1) / CT8;
    const int chunks = dw_stats_chunks(N, H, W, C, 1, k, wl);
    dim3 g(c_tiles * chunks);
    // dgrad supports only lop none (dz is materialized by bn_act_bwd);
    // with stats_ws + epi it also emits the consumer BN's backward
    // partial sums (EPI==2) so bn_bwd_partial is skipped
#define DWD_EPI(KK, AC, WLW)                                               \
    if (stats_ws != nullptr && k == KK && wl == WLW &&                     \
        lopdy.kind == 0 && epi.act == AC) {                                \
      dw_fwd_kernel<KK, 1, true, 2, 0, AC, WLW><<<g, 256, 0, s>>>(         \
          gp, wp, xp, N, Ho, Wo, C, H, W, pad, L, LE, stats_ws, chunks);   \
      return;                                                              \
    }
#define DWD_CASE(KK, WLW)                                                  \
    DWD_EPI(KK, 0, WLW) DWD_EPI(KK, 1, WLW)                                \
    DWD_EPI(KK, 2, WLW) DWD_EPI(KK, 3, WLW)                                \
    if (k == KK && wl == WLW && lopdy.kind == 0) {                         \
      dw_fwd_kernel<KK, 1, true, 0, 0, 0, WLW><<<g, 256, 0, s>>>(          \
          gp, wp, xp, N, Ho, Wo, C, H, W, pad, L, LE, nullptr, chunks);    \
      return;                                                              \
    }
    DWD_CASE(3, false) DWD_CASE(3, true)
    DWD_CASE(5, false) DWD_CASE(5, true)
#undef DWD_CASE
#undef DWD_EPI
    yams_launch_fail("dw_conv_dgrad_s1", k, lopdy.kind, lopdy.act, epi.act);
  }
  const int64_t total8 = int64_t(N) * H * W * (C >> 3);
  dim3 g2(grid_for(total8));
#define DWD2_CASE(KK)                                                      \
  if (k == KK && lopdy.kind == 0) {                                        \
    dw_dgrad_s2_kernel<KK, 0, 0><<<g2, 256, 0, s>>>(                       \
        gp, wp, xp, N, H, W, C, Ho, Wo, stride, pad, L, total8);           \
    return;                                                                \
  }
  DWD2_CASE(3) DWD2_CASE(5)
#undef DWD2_CASE
  yams_launch_fail("dw_conv_dgrad_s2", k, stride, lopdy.kind, lopdy.act);
}

int dw_wgrad_chunks(int N, int Ho, int Wo) {
  const int64_t P = int64_t(N) * Ho * ((Wo + 3) / 4);
  int c = int(std::min<int64_t>((P + 3) / 4, 1024));  // see dw_stats_chunks
  return c < 1 ? 1 : c;
}

int64_t dw_wgrad_ws_elems(int N, int Ho, int Wo, int C, int k) {
  const int c_tiles = ((C >> 2) + 255) / 256;
  return int64_t((dw_wgrad_chunks(N, Ho, Wo) + c_tiles - 1) / c_tiles)
         * k * k * C;
}

void dw_conv_wgrad(const void* x, const void* dy, float* dwgt, float* ws,
                   int N, int H, int W, int C, int Ho, int Wo, int k,
                   int stride, int pad, LopDesc lopx, LopDesc lopdy,
                   hipStream_t s) {
  const int C4 = C >> 2;
  const int CT4 = C4 < 256 ? C4 : 256;
  const int c_tiles = (C4 + CT4 - 1) / CT4;
  const int chunks = (dw_wgrad_chunks(N, Ho, Wo) + c_tiles - 1) / c_tiles;
  auto* xp = static_cast<const ushort_t*>(x);
  auto* gp = static_cast<const ushort_t*>(dy);
  const Lop Lx = to_lop(lopx);
  const Lop Ldy = to_lop(lopdy);
  // wgrad x-stream lop: none or bnact (act 1..3); dy-stream: none only
#define DWW_CASE(KK, SS, LXX, AC)                                          \
  if (k == KK && stride == SS && lopx.kind == LXX &&                       \
      (LXX == 0 || lopx.act == AC) && lopdy.kind == 0) {                   \
    constexpr int IP = (KK == 5) ? (SS == 2 ? 3 : 2) : 1;                  \
    dim3 grid(c_tiles * chunks * IP);                                      \
    dw_wgrad_kernel<KK, SS, LXX, AC, 0, IP><<<grid, 256, 0, s>>>(          \
        xp, gp, ws, N, H, W, C, Ho, Wo, pad, Lx, Ldy, chunks);             \
    goto combine;                                                          \
  }
#define DWW_KS(KK, SS)                                                     \
  DWW_CASE(KK, SS, 0, 0) DWW_CASE(KK, SS, 1, 1) DWW_CASE(KK, SS, 1, 2)     \
  DWW_CASE(KK, SS, 1, 3)
  DWW_KS(3, 1) DWW_KS(3, 2) DWW_KS(5, 1) DWW_KS(5, 2)
#undef DWW_KS
#undef DWW_CASE
  yams_launch_fail("dw_conv_wgrad", k, stride, lopx.kind, lopdy.kind);
combine:
  // transposed combine: dwgt lands as [C, k*k] (OIHW view, no copy)
  slab_reduce_dwt(ws, dwgt, C, k * k, chunks, s);
}

void im2row(const void* x, void* patches, int N, int H, int W, int Ho,
            int Wo, int k, int stride, int pad, int KP, hipStream_t s) {
  const int64_t total = int64_t(N) * Ho * Wo;
  im2row_kernel<<<grid_for((total + 3) / 4), 256, 0, s>>>(
      static_cast<const ushort_t*>(x), static_cast<ushort_t*>(patches),
      N, H, W, Ho, Wo, k, stride, pad, KP, total);
}

}  // namespace yams
