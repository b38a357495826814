// MFMA bf16 GEMM family for gfx950: the pointwise-conv (K2) / FC (K10)
// compute path, plus the deterministic split-M wgrad GEMM.
//
// pw_conv:  y[M,N] = x[M,K] @ w[N,K]^T
//   - implicit GEMM over NHWC rows: M = N*H*W, K = Cin, N = Cout
//   - both operands K-major, so A-frag and B-frag LDS reads are contiguous
//     16-byte ds_read_b128 (no transpose staging)
//   - v_mfma_f32_16x16x32_bf16, fp32 accumulate; masked edges so any
//     (M, K, N) with K % 8 == 0 works
//
// pw_conv_wgrad: dw[N1,N2] = a[M,N1]^T @ b[M,N2]
//   - M (= batch*spatial) is the reduction axis; fixed split-M partial
//     slabs + a separate reduce kernel: deterministic, no atomics
//     (SURVEY.md §5.2).

#include <algorithm>
#include <cstdlib>

#include "common.h"
#include "engine.h"

namespace yams {

// ---------------------------------------------------------------------------
// pw_conv: 128x64 tile, BK=32, 4 waves as 2(M)x2(N), 64x32 per wave.
// ---------------------------------------------------------------------------
namespace {

constexpr int BM = 128, BN = 64, BK = 32;
constexpr int LDA = BK + 8;  // +16B pad: conflict-free b128 column reads
constexpr int FRAG_M = 4, FRAG_N = 2;  // 16x16 frags per wave

inline Lop to_lop(const LopDesc& d) {
  Lop L;
  L.kind = d.kind;
  L.act = d.act;
  L.coef = d.coef;
  L.z = static_cast<const ushort_t*>(d.z);
  L.C = d.C;
  return L;
}

// EPI: output epilogue — 0 none, 1 forward BN stats (sum, sumsq of the
// bf16-rounded output per column), 2 BN-BACKWARD stats: the output is a
// dgrad dy feeding a BN backward, and the epilogue computes that BN's
// partial sums (sum da, sum da*z with da = dy * act'(a*z+b)) so the
// separate bn_bwd_partial pass over dy+z never runs.  lopE carries the
// z tensor, the ab[2C] coefs and the act for EPI==2.
// RES: epilogue residual add (C = A@B^T + res) — used by the backward's
// d_x = pw_dgrad(dz1) + dyb so the residual path costs zero extra passes
// TBM: M-tile — 128 default; 64 for small-M dgrads, where ceil(M/128)
// blocks underfill the 256-CU chip (e.g. 7^2 layers: 98x3 blocks)
template <int EPI, int LOPK, int EACT = 0, bool RES = false, int TBM = BM>
__global__ __launch_bounds__(256) void pw_conv_kernel(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ Bt,
    ushort_t* __restrict__ C, int64_t M, int K, int N, Lop lopA, Lop lopE,
    float* __restrict__ stats_ws, const ushort_t* __restrict__ res) {
  // double-buffered register staging (guide T14: issue the next tile's
  // global loads before the current tile's MFMAs, write LDS after the
  // barrier) — the single-buffered loop parked 47% of wave cycles on
  // staging waits (profiles/gemm_counter_collection.csv)
  constexpr int FRAG_Mv = TBM / 32;   // 16x16 frags per wave (M dim)
  constexpr int TPRA = 256 / TBM;     // staging threads per A row
  constexpr int APT = BK / TPRA;      // A elems staged per thread
  __shared__ ushort_t Alds[2][TBM * LDA];
  __shared__ ushort_t Blds[2][BN * LDA];

  const int tid = threadIdx.x;
  const int64_t bm = int64_t(blockIdx.x) * TBM;
  const int bn = blockIdx.y * BN;
  const int wave = tid >> 6, lane = tid & 63;
  const int wrow0 = (wave >> 1) * (TBM / 2), wcol0 = (wave & 1) * 32;
  const int fr = lane & 15, fk = (lane >> 4) * 8;

  f32x4 acc[FRAG_Mv][FRAG_N];
#pragma unroll
  for (int i = 0; i < FRAG_Mv; ++i)
#pragma unroll
    for (int j = 0; j < FRAG_N; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int a_row = tid / TPRA, a_seg = (tid % TPRA) * APT;
  const int b_row = tid >> 2, b_seg = (tid & 3) * 8;    // 4 thr/row x 64
  const int64_t gm = bm + a_row;
  const int gn = bn + b_row;

  ushort_t areg[APT];
  ushort_t breg[8];

  auto stage_regs = [&](int k0) {
    const int gka = k0 + a_seg;
    if (gm < M && gka + APT <= K) {
      const int64_t off = gm * K + gka;
#pragma unroll
      for (int g8 = 0; g8 < APT / 8; ++g8)
        lop_vec8<LOPK>(lopA, A, off + g8 * 8, gka + g8 * 8, areg + g8 * 8);
    } else if (gm >= M || gka >= K) {
#pragma unroll
      for (int j = 0; j < APT; ++j) areg[j] = ushort_t(0);
    } else {
      // partial segment: per-element conditional loads (the segment-level
      // all-OOB cases above are wave-uniform and skip loads entirely;
      // always-executed clamped loads measured slower for lop variants)
#pragma unroll
      for (int j = 0; j < APT; ++j) {
        const int gkj = gka + j;
        areg[j] = (gm < M && gkj < K)
                      ? ushort_t(f2bf(lop_loadT<LOPK>(lopA, A,
                                                      gm * K + gkj, gkj)))
                      : ushort_t(0);
      }
    }
    const int gkb = k0 + b_seg;
    if (gn < N && gkb + 8 <= K) {
      *reinterpret_cast<short4_t*>(breg) =
          *reinterpret_cast<const short4_t*>(&Bt[int64_t(gn) * K + gkb]);
      *reinterpret_cast<short4_t*>(breg + 4) =
          *reinterpret_cast<const short4_t*>(&Bt[int64_t(gn) * K + gkb + 4]);
    } else if (gn >= N || gkb >= K) {
#pragma unroll
      for (int j = 0; j < 8; ++j) breg[j] = ushort_t(0);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        breg[j] = (gn < N && gkb + j < K)
                      ? Bt[int64_t(gn) * K + gkb + j] : ushort_t(0);
    }
  };

  auto write_lds = [&](int buf) {
    ushort_t* da = &Alds[buf][a_row * LDA + a_seg];
#pragma unroll
    for (int g8 = 0; g8 < APT / 8; ++g8)
      *reinterpret_cast<short8_t*>(da + g8 * 8) =
          *reinterpret_cast<const short8_t*>(areg + g8 * 8);
    ushort_t* db = &Blds[buf][b_row * LDA + b_seg];
    *reinterpret_cast<short8_t*>(db) =
        *reinterpret_cast<const short8_t*>(breg);
  };

  const int nt = (K + BK - 1) / BK;
  stage_regs(0);
  write_lds(0);
  __syncthreads();
  for (int t = 0; t < nt; ++t) {
    if (t + 1 < nt) stage_regs((t + 1) * BK);  // loads fly under the MFMAs
    const int cur = t & 1;
    bf16x8 a_frag[FRAG_Mv], b_frag[FRAG_N];
#pragma unroll
    for (int i = 0; i < FRAG_Mv; ++i)
      a_frag[i] = *reinterpret_cast<const bf16x8*>(
          &Alds[cur][(wrow0 + i * 16 + fr) * LDA + fk]);
#pragma unroll
    for (int j = 0; j < FRAG_N; ++j)
      b_frag[j] = *reinterpret_cast<const bf16x8*>(
          &Blds[cur][(wcol0 + j * 16 + fr) * LDA + fk]);
#pragma unroll
    for (int i = 0; i < FRAG_Mv; ++i)
#pragma unroll
      for (int j = 0; j < FRAG_N; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    if (t + 1 < nt) {
      __syncthreads();       // everyone past buf[cur^1] reads of tile t-1
      write_lds(cur ^ 1);
      __syncthreads();       // tile t+1 visible
    }
  }

  // epilogue: C/D frag layout col = lane&15, row = (lane>>4)*4 + r.
  // Stores and the EPI statistics run in ONE pass over acc (the split
  // version converted every element to bf16 twice — measurable on the
  // skinny-K shapes where the epilogue rivals the 1-2 mainloop tiles).
  const int col_l = fr, row_l4 = (lane >> 4) * 4;
  float s1[FRAG_N] = {}, s2[FRAG_N] = {};
#pragma unroll
  for (int j = 0; j < FRAG_N; ++j) {
    const int col = bn + wcol0 + j * 16 + col_l;
    if (col >= N) continue;
    float ea = 0.f, eb = 0.f;
    if (EPI == 2) {
      ea = lopE.coef[col];
      eb = lopE.coef[N + col];
    }
#pragma unroll
    for (int i = 0; i < FRAG_Mv; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int64_t row = bm + wrow0 + i * 16 + row_l4 + r;
        if (row < M) {
          float v = acc[i][j][r];
          if (RES) v += bf2f(res[row * N + col]);
          const ushort_t o = f2bf(v);
          C[row * N + col] = o;
          if (EPI == 1) {
            const float vb = bf2f(o);
            s1[j] += vb;
            s2[j] = fmaf(vb, vb, s2[j]);
          } else if (EPI == 2) {
            const float vb = bf2f(o);
            const float zv = bf2f(lopE.z[row * N + col]);
            const float da = vb * act_bwd<EACT>(fmaf(zv, ea, eb));
            s1[j] += da;
            s2[j] = fmaf(da, zv, s2[j]);
          }
        }
      }
    }
  }

  if (EPI) {
    // per-block partial column sums of the (bf16-rounded) output.
    float* red = reinterpret_cast<float*>(Alds);  // 512 floats
    const int slot = (wave >> 1) * 4 + (lane >> 4);  // 0..7
    const int mblocks = gridDim.x;
    __syncthreads();
#pragma unroll
    for (int j = 0; j < FRAG_N; ++j)
      red[slot * 64 + wcol0 + j * 16 + col_l] = s1[j];
    __syncthreads();
    if (tid < 64 && bn + tid < N) {
      float t = 0.f;
#pragma unroll
      for (int l = 0; l < 8; ++l) t += red[l * 64 + tid];
      stats_ws[int64_t(blockIdx.x) * N + bn + tid] = t;
    }
    __syncthreads();
#pragma unroll
    for (int j = 0; j < FRAG_N; ++j)
      red[slot * 64 + wcol0 + j * 16 + col_l] = s2[j];
    __syncthreads();
    if (tid < 64 && bn + tid < N) {
      float t = 0.f;
#pragma unroll
      for (int l = 0; l < 8; ++l) t += red[l * 64 + tid];
      stats_ws[(int64_t(mblocks) + blockIdx.x) * N + bn + tid] = t;
    }
  }
}

}  // namespace

// BM policy shared by dispatch and the stats-ws sizing: BM=64 doubles
// the block count (12 waves/CU) — measured faster for the dgrads and
// the forward alike; YAMS_PW_BM64{,_MAX} sweep it.
static int pw_bm_for(int64_t M) {
  static int on = [] {
    const char* e = getenv("YAMS_PW_BM64");
    return e == nullptr || atoi(e) != 0;
  }();
  static int64_t mx = [] {
    const char* e = getenv("YAMS_PW_BM64_MAX");
    return e ? atoll(e) : (1LL << 40);
  }();
  return (on && M <= mx) ? 64 : BM;
}

int pw_stats_mblocks(int64_t M) { return ceil_div(M, pw_bm_for(M)); }

void pw_conv(const void* x, const void* w, void* y,
             int64_t M, int K, int N, LopDesc lopa, LopDesc epi,
             float* stats_ws, const void* res, hipStream_t s) {
  const Lop L = to_lop(lopa);
  const Lop LE = to_lop(epi);
  const int ep = stats_ws == nullptr ? 0 : (epi.kind == LOP_BNACT ? 2 : 1);
  auto* ap = static_cast<const ushort_t*>(x);
  auto* wp = static_cast<const ushort_t*>(w);
  auto* yp = static_cast<ushort_t*>(y);
  auto* rp = static_cast<const ushort_t*>(res);
  // stats-free launches (dgrads) use the BM=64 tile: twice the blocks
  // and 12 waves/CU on a chip that needs >>256 workgroups
  // (YAMS_PW_BM64=0 disables, YAMS_PW_BM64_MAX sweeps the threshold)
  // the BM policy must match pw_stats_mblocks (the Python side sizes
  // stats workspaces with it), so EVERY ep follows it
  const bool small = pw_bm_for(M) == 64;
  dim3 grid(ceil_div(M, small ? 64 : BM), ceil_div(N, BN));
  if (rp != nullptr) {  // residual epilogue: plain or BNBWD-lop dgrad
    if (ep == 0 && lopa.kind == 0) {
      if (small)
        pw_conv_kernel<0, 0, 0, true, 64><<<grid, 256, 0, s>>>(
            ap, wp, yp, M, K, N, L, LE, stats_ws, rp);
      else
        pw_conv_kernel<0, 0, 0, true><<<grid, 256, 0, s>>>(
            ap, wp, yp, M, K, N, L, LE, stats_ws, rp);
      return;
    }
    if (ep == 0 && lopa.kind == 2) {
      if (small)
        pw_conv_kernel<0, 2, 0, true, 64><<<grid, 256, 0, s>>>(
            ap, wp, yp, M, K, N, L, LE, stats_ws, rp);
      else
        pw_conv_kernel<0, 2, 0, true><<<grid, 256, 0, s>>>(
            ap, wp, yp, M, K, N, L, LE, stats_ws, rp);
      return;
    }
    yams_launch_fail("pw_conv_res", ep, lopa.kind, lopa.act, epi.act);
  }
  if (small) {
#define PW64_CASE(EP, LK)                                                 \
    if (ep == EP && lopa.kind == LK) {                                    \
      pw_conv_kernel<EP, LK, 0, false, 64><<<grid, 256, 0, s>>>(          \
          ap, wp, yp, M, K, N, L, LE, stats_ws, nullptr);                 \
      return;                                                             \
    }
    PW64_CASE(0, 0) PW64_CASE(0, 1) PW64_CASE(0, 2)
    PW64_CASE(1, 0) PW64_CASE(1, 1) PW64_CASE(1, 2)
#undef PW64_CASE
#define PW64_EPI2(AC)                                                     \
    if (ep == 2 && lopa.kind == 0 && epi.act == AC) {                     \
      pw_conv_kernel<2, 0, AC, false, 64><<<grid, 256, 0, s>>>(           \
          ap, wp, yp, M, K, N, L, LE, stats_ws, nullptr);                 \
      return;                                                             \
    }
    PW64_EPI2(0) PW64_EPI2(1) PW64_EPI2(2) PW64_EPI2(3)
#undef PW64_EPI2
    yams_launch_fail("pw_conv64", ep, lopa.kind, lopa.act, epi.act);
  }
#define PW_CASE(EP, LK)                                                   \
  if (ep == EP && lopa.kind == LK) {                                      \
    pw_conv_kernel<EP, LK><<<grid, 256, 0, s>>>(ap, wp, yp, M, K, N, L,   \
                                                LE, stats_ws, nullptr);   \
    return;                                                               \
  }
#define PW_EPI2(AC)                                                       \
  if (ep == 2 && lopa.kind == 0 && epi.act == AC) {                       \
    pw_conv_kernel<2, 0, AC><<<grid, 256, 0, s>>>(ap, wp, yp, M, K, N, L, \
                                                  LE, stats_ws, nullptr); \
    return;                                                               \
  }
  PW_CASE(0, 0) PW_CASE(0, 1) PW_CASE(0, 2)
  PW_CASE(1, 0) PW_CASE(1, 1) PW_CASE(1, 2)
  PW_EPI2(0) PW_EPI2(1) PW_EPI2(2) PW_EPI2(3)
#undef PW_EPI2
#undef PW_CASE
  yams_launch_fail("pw_conv", ep, lopa.kind, lopa.act, epi.act);
}

// ---------------------------------------------------------------------------
// wgrad: dw[N1,N2] = a[M,N1]^T @ b[M,N2] via split-M fp32 slabs.
// 64x64 tile per block, 16x16 threads x (4x4 per thread), VALU fp32 FMA
// (the reduction axis streams through once — memory-bound, MFMA not needed).
// ---------------------------------------------------------------------------
namespace {

// MFMA version: C[N1,N2] = A^T B with A:[M,N1], B:[M,N2].  Both operands
// are M-major (M is the reduction axis), so staging TRANSPOSES tiles into
// K-major LDS rows ([n][k], k = m-chunk) and then the frag-read + MFMA
// structure is identical to pw_conv_kernel.  The transposed scatter writes
// use an XOR chunk swizzle (phys k-chunk = (k>>3) ^ ((n>>3)&7)): the
// staging loop walks n in steps of 8, so swizzling on n&7 left every
// lane of a write on ONE bank group (PMC: 24.8% of atb cycles in LDS
// conflicts, profiles/r02/pmc_summary); the n>>3 bits vary across the
// write lanes AND across the 16 consecutive n of a frag read.  Replaces the VALU fp32-FMA version,
// which was compute-bound at ~1/16 of MFMA throughput.
constexpr int KT = 64;        // m rows per LDS pass
constexpr int LDK = KT + 8;   // shorts per LDS row (pad keeps b128 align)

template <int W1, int W2, int LA, int LB>
__global__ __launch_bounds__(256) void atb_kernel(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ B,
    float* __restrict__ ws, int64_t M, int N1, int N2, int splits,
    Lop lopA, Lop lopB) {
  constexpr int HW1 = W1 / 2, HW2 = W2 / 2;   // per-wave sub-tile (2x2 waves)
  constexpr int F1 = HW1 / 16, F2 = HW2 / 16; // 16x16 frags per wave
  __shared__ ushort_t Al[W1 * LDK];
  __shared__ ushort_t Bl[W2 * LDK];

  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wrow0 = (wave >> 1) * HW1, wcol0 = (wave & 1) * HW2;
  const int fr = lane & 15, fk = (lane >> 4) * 8;
  const int n1_tiles = ceil_div(N1, W1);
  const int n_tiles = n1_tiles * ceil_div(N2, W2);
  const int tile = blockIdx.x % n_tiles;
  const int split = blockIdx.x / n_tiles;
  const int n1_0 = (tile % n1_tiles) * W1;
  const int n2_0 = (tile / n1_tiles) * W2;

  const int64_t m_per = (M + splits - 1) / splits;
  const int64_t m_lo = split * m_per;
  const int64_t m_hi = m_lo + m_per < M ? m_lo + m_per : M;

  f32x4 acc[F1][F2];
#pragma unroll
  for (int i = 0; i < F1; ++i)
#pragma unroll
    for (int j = 0; j < F2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  constexpr int TPR1 = W1 / 8;  // threads per m-row (vec8 along n)
  constexpr int TPR2 = W2 / 8;

  for (int64_t m0 = m_lo; m0 < m_hi; m0 += KT) {
    __syncthreads();
    {
      const int s_m0 = tid / TPR1, s_n = (tid % TPR1) * 8;
      ushort_t tmp[8];
      for (int r = s_m0; r < KT; r += 256 / TPR1) {
        const int64_t gm = m0 + r;
        const int gn = n1_0 + s_n;
        if (gm < m_hi && gn + 8 <= N1) {
          lop_vec8<LA>(lopA, A, gm * N1 + gn, gn, tmp);
        } else if (gm >= m_hi || gn >= N1) {
#pragma unroll
          for (int j = 0; j < 8; ++j) tmp[j] = ushort_t(0);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            tmp[j] = (gm < m_hi && gn + j < N1)
                         ? ushort_t(f2bf(lop_loadT<LA>(lopA, A,
                                                       gm * N1 + gn + j,
                                                       gn + j)))
                         : ushort_t(0);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int n = s_n + j;
          Al[n * LDK +
             ((((r >> 3) ^ ((n >> 3) & 7)) << 3) | (r & 7))] = tmp[j];
        }
      }
    }
    {
      const int s_m0 = tid / TPR2, s_n = (tid % TPR2) * 8;
      ushort_t tmp[8];
      for (int r = s_m0; r < KT; r += 256 / TPR2) {
        const int64_t gm = m0 + r;
        const int gn = n2_0 + s_n;
        if (gm < m_hi && gn + 8 <= N2) {
          lop_vec8<LB>(lopB, B, gm * N2 + gn, gn, tmp);
        } else if (gm >= m_hi || gn >= N2) {
#pragma unroll
          for (int j = 0; j < 8; ++j) tmp[j] = ushort_t(0);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            tmp[j] = (gm < m_hi && gn + j < N2)
                         ? ushort_t(f2bf(lop_loadT<LB>(lopB, B,
                                                       gm * N2 + gn + j,
                                                       gn + j)))
                         : ushort_t(0);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int n = s_n + j;
          Bl[n * LDK +
             ((((r >> 3) ^ ((n >> 3) & 7)) << 3) | (r & 7))] = tmp[j];
        }
      }
    }
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < KT / 32; ++ks) {
      bf16x8 a_frag[F1], b_frag[F2];
#pragma unroll
      for (int i = 0; i < F1; ++i) {
        const int n = wrow0 + i * 16 + fr;
        const int kc = ((ks * 32 + fk) >> 3) ^ ((n >> 3) & 7);
        a_frag[i] = *reinterpret_cast<const bf16x8*>(&Al[n * LDK + kc * 8]);
      }
#pragma unroll
      for (int j = 0; j < F2; ++j) {
        const int n = wcol0 + j * 16 + fr;
        const int kc = ((ks * 32 + fk) >> 3) ^ ((n >> 3) & 7);
        b_frag[j] = *reinterpret_cast<const bf16x8*>(&Bl[n * LDK + kc * 8]);
      }
#pragma unroll
      for (int i = 0; i < F1; ++i)
#pragma unroll
        for (int j = 0; j < F2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    }
  }

  // D frag layout: row (=n1) = (lane>>4)*4 + r, col (=n2) = lane&15
  float* out = ws + int64_t(split) * N1 * N2;
  const int row_l4 = (lane >> 4) * 4;
#pragma unroll
  for (int j = 0; j < F2; ++j) {
    const int n2 = n2_0 + wcol0 + j * 16 + fr;
    if (n2 >= N2) continue;
#pragma unroll
    for (int i = 0; i < F1; ++i)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int n1 = n1_0 + wrow0 + i * 16 + row_l4 + r;
        if (n1 < N1) out[int64_t(n1) * N2 + n2] = acc[i][j][r];
      }
  }
}

int tile_of(int n) { return n <= 32 ? 32 : (n <= 64 ? 64 : 128); }

void atb_launch(const ushort_t* a, const ushort_t* b, float* ws, int64_t M,
                int N1, int N2, int splits, int blocks, Lop lopA, Lop lopB,
                int ka, int kb, hipStream_t s) {
  const int w1 = tile_of(N1), w2 = tile_of(N2);
  dim3 g(blocks);
  // lop kinds used in practice: A in {none, bnbwd}, B in {none, bnact}
#define ATB_CASE(X, Y, LA, LB)                                           \
  if (w1 == X && w2 == Y && ka == LA && kb == LB) {                      \
    atb_kernel<X, Y, LA, LB><<<g, 256, 0, s>>>(a, b, ws, M, N1, N2,      \
                                               splits, lopA, lopB);      \
    return;                                                              \
  }
#define ATB_TILE(X, Y)                                                   \
  ATB_CASE(X, Y, 0, 0) ATB_CASE(X, Y, 2, 0) ATB_CASE(X, Y, 0, 1)         \
  ATB_CASE(X, Y, 2, 1)
  ATB_TILE(32, 32) ATB_TILE(32, 64) ATB_TILE(32, 128)
  ATB_TILE(64, 32) ATB_TILE(64, 64) ATB_TILE(64, 128)
  ATB_TILE(128, 32) ATB_TILE(128, 64) ATB_TILE(128, 128)
#undef ATB_TILE
#undef ATB_CASE
  yams_launch_fail("atb", w1, w2, ka, kb);
}

// vec4 variant: thread = 4 consecutive fp32 elems x one of SL split lanes;
// reads are 16B/lane fully coalesced (the scalar kernel read 4B chunks and
// serialized the split loop -> 4-5x off the slab-stream floor).
template <int SL>
__global__ __launch_bounds__(256) void slab_reduce4_kernel(
    const float* __restrict__ ws, float* __restrict__ out, int64_t elems,
    int splits) {
  constexpr int EL = 256 / SL;          // float4 lanes per block
  __shared__ float red[256][4];
  const int el = int(threadIdx.x) % EL;
  const int sl = int(threadIdx.x) / EL;
  for (int64_t base = int64_t(blockIdx.x) * EL * 4; base < elems;
       base += int64_t(gridDim.x) * EL * 4) {
    const int64_t i = base + el * 4;
    float v[4] = {};
    if (i + 4 <= elems) {
      for (int sp = sl; sp < splits; sp += SL) {
        const float4 w4 = *reinterpret_cast<const float4*>(
            &ws[int64_t(sp) * elems + i]);
        v[0] += w4.x; v[1] += w4.y; v[2] += w4.z; v[3] += w4.w;
      }
    } else if (i < elems) {
      for (int sp = sl; sp < splits; sp += SL)
        for (int e = 0; e < 4; ++e)
          if (i + e < elems) v[e] += ws[int64_t(sp) * elems + i + e];
    }
#pragma unroll
    for (int e = 0; e < 4; ++e) red[threadIdx.x][e] = v[e];
    __syncthreads();
    if (sl == 0 && i < elems) {
      for (int l = 1; l < SL; ++l)
#pragma unroll
        for (int e = 0; e < 4; ++e) v[e] += red[el + l * EL][e];
      if (i + 4 <= elems) {
        *reinterpret_cast<float4*>(&out[i]) = float4{v[0], v[1], v[2], v[3]};
      } else {
        for (int e = 0; e < 4 && i + e < elems; ++e) out[i + e] = v[e];
      }
    }
    __syncthreads();
  }
}

// dw-wgrad variant: input slabs are [splits][KK][C] (C fastest) but the
// OUTPUT is written transposed as [C][KK] so it reshapes to the OIHW
// [C,1,k,k] param-grad layout without a separate transpose copy (15
// eager copies/step removed).  Reads stay fully coalesced; the scattered
// 4B stores are trivial (<100 KB).
template <int SL>
__global__ __launch_bounds__(256) void slab_reduce_dwt_kernel(
    const float* __restrict__ ws, float* __restrict__ out, int C, int KK,
    int splits) {
  constexpr int EL = 256 / SL;
  __shared__ float red[256][4];
  const int elems = C * KK;
  const int el = int(threadIdx.x) % EL;
  const int sl = int(threadIdx.x) / EL;
  for (int64_t base = int64_t(blockIdx.x) * EL * 4; base < elems;
       base += int64_t(gridDim.x) * EL * 4) {
    const int64_t i = base + el * 4;
    float v[4] = {};
    if (i + 4 <= elems) {
      for (int sp = sl; sp < splits; sp += SL) {
        const float4 w4 = *reinterpret_cast<const float4*>(
            &ws[int64_t(sp) * elems + i]);
        v[0] += w4.x; v[1] += w4.y; v[2] += w4.z; v[3] += w4.w;
      }
    } else if (i < elems) {
      for (int sp = sl; sp < splits; sp += SL)
        for (int e = 0; e < 4; ++e)
          if (i + e < elems) v[e] += ws[int64_t(sp) * elems + i + e];
    }
#pragma unroll
    for (int e = 0; e < 4; ++e) red[threadIdx.x][e] = v[e];
    __syncthreads();
    if (sl == 0 && i < elems) {
      for (int l = 1; l < SL; ++l)
#pragma unroll
        for (int e = 0; e < 4; ++e) v[e] += red[el + l * EL][e];
      for (int e = 0; e < 4 && i + e < elems; ++e) {
        const int flat = int(i) + e;         // = j*C + c in slab order
        const int j = flat / C;
        const int cc = flat % C;
        out[int64_t(cc) * KK + j] = v[e];
      }
    }
    __syncthreads();
  }
}

}  // namespace

void slab_reduce_dwt(const float* ws, float* out, int C, int KK,
                     int splits, hipStream_t s) {
  const int sl = splits >= 64 ? 64 : (splits >= 16 ? 16 : 4);
  const int el4 = (256 / sl) * 4;
  const int64_t elems = int64_t(C) * KK;
  const int blocks = int(std::min<int64_t>((elems + el4 - 1) / el4, 8192));
  if (sl == 64)
    slab_reduce_dwt_kernel<64><<<dim3(blocks), 256, 0, s>>>(ws, out, C, KK,
                                                            splits);
  else if (sl == 16)
    slab_reduce_dwt_kernel<16><<<dim3(blocks), 256, 0, s>>>(ws, out, C, KK,
                                                            splits);
  else
    slab_reduce_dwt_kernel<4><<<dim3(blocks), 256, 0, s>>>(ws, out, C, KK,
                                                           splits);
}

void slab_reduce(const float* ws, float* out, int64_t elems, int splits,
                 hipStream_t s) {
  const int sl = splits >= 64 ? 64 : (splits >= 16 ? 16 : 4);
  const int el4 = (256 / sl) * 4;
  const int blocks = int(std::min<int64_t>((elems + el4 - 1) / el4, 8192));
  if (sl == 64)
    slab_reduce4_kernel<64><<<dim3(blocks), 256, 0, s>>>(ws, out, elems,
                                                         splits);
  else if (sl == 16)
    slab_reduce4_kernel<16><<<dim3(blocks), 256, 0, s>>>(ws, out, elems,
                                                         splits);
  else
    slab_reduce4_kernel<4><<<dim3(blocks), 256, 0, s>>>(ws, out, elems,
                                                        splits);
}

int pw_wgrad_splits(int64_t M, int N1, int N2) {
  // measurement knob: YAMS_ATB_BLOCKS scales the split-M block target
  // (default 1024; the split count trades slab-combine traffic against
  // fill — swept on hardware, see profiles/)
  static int target = [] {
    const char* e = getenv("YAMS_ATB_BLOCKS");
    const int v = e ? atoi(e) : 1024;
    return v > 0 ? v : 1024;
  }();
  const int tiles = ceil_div(N1, tile_of(N1)) * ceil_div(N2, tile_of(N2));
  const int64_t max_s = (M + 127) / 128;
  int64_t s = target / tiles;
  // cap the split count so the fp32 slab traffic (8*N1*N2 bytes per
  // split, written then re-read) stays of the order of the bf16 input
  // traffic (2*M*(N1+N2)) — uncapped, the 14^2/7^2 wgrads wrote 4-8x
  // their input in slabs and ran at 300-800 GB/s (profiles/r02)
  const int64_t ws_cap = std::max<int64_t>(
      16, M * (N1 + N2) / (2LL * N1 * N2));
  if (s > ws_cap) s = ws_cap;
  if (s < 1) s = 1;
  if (s > max_s) s = max_s;
  return int(s < 1 ? 1 : s);
}

int64_t pw_wgrad_ws_elems(int64_t M, int N1, int N2) {
  return int64_t(pw_wgrad_splits(M, N1, N2)) * N1 * N2;
}

void pw_conv_wgrad(const void* a, const void* b, float* dw, float* ws,
                   int64_t M, int N1, int N2, LopDesc lopa, LopDesc lopb,
                   hipStream_t s) {
  const int splits = pw_wgrad_splits(M, N1, N2);
  const int tiles = ceil_div(N1, tile_of(N1)) * ceil_div(N2, tile_of(N2));
  atb_launch(static_cast<const ushort_t*>(a),
             static_cast<const ushort_t*>(b), ws, M, N1, N2, splits,
             tiles * splits, to_lop(lopa), to_lop(lopb), lopa.kind,
             lopb.kind, s);
  slab_reduce(ws, dw, int64_t(N1) * N2, splits, s);
}

}  // namespace yams
