// Kernel launcher declarations: the C ABI between bindings.cpp (compiled
// against torch headers) and the gfx950 kernels in *.hip.
//
// Pointer conventions: `bf16*` passed as const void*/void*; fp32 as float*.
// All launches are asynchronous on the given hipStream_t.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

namespace yams {

// Host-side descriptor of a load-op (see common.h Lop): a per-element
// transform applied while an operand stream is staged.  kind 0 = none,
// 1 = bnact (coef = ab[2C]), 2 = bnbwd (coef = [4C] {a,b,K0,K2}, z = the
// paired conv-output stream with the main stream's layout).
struct LopDesc {
  int kind = 0;
  int act = 0;
  const float* coef = nullptr;
  const void* z = nullptr;
  int C = 0;
};

// --- GEMM family (pointwise conv K2, FC K10) ----------------------------
// y[M,N] = x[M,K] @ w[N,K]^T   (both operands K-major; w is the natural
// [Cout, Cin] torch layout for forward, its transpose for dgrad).
// lopa transforms the A stream (channel = K index).  stats_ws, if
// non-null, receives per-m-block partial column sums/sumsq of the OUTPUT
// ([2, ceil(M/128), N] fp32) for BN-stats fusion; combine with
// bn_finalize(splits = pw_stats_mblocks(M)).
// epi: kind!=0 => stats_ws receives BN-BACKWARD partials (sum da, sum
// da*z) of the consumer BN instead of forward sum/sumsq; epi carries the
// z tensor, ab[2C] coefs and act (see gemm.hip EPI==2).
// res: optional epilogue residual (y = x@w^T + res; plain-GEMM case only)
void pw_conv(const void* x, const void* w, void* y,
             int64_t M, int K, int N, LopDesc lopa, LopDesc epi,
             float* stats_ws, const void* res, hipStream_t s);
int pw_stats_mblocks(int64_t M);

// dw[N1,N2] = a[M,N1]^T @ b[M,N2], fp32 out, deterministic split-M slabs
// (lopa/lopb transform the streams; channel = the N1/N2 index)
void pw_conv_wgrad(const void* a, const void* b, float* dw, float* ws,
                   int64_t M, int N1, int N2, LopDesc lopa, LopDesc lopb,
                   hipStream_t s);
int pw_wgrad_splits(int64_t M, int N1, int N2);
int64_t pw_wgrad_ws_elems(int64_t M, int N1, int N2);

// out[i] = sum_s ws[s*elems + i] (deterministic slab combine)
void slab_reduce(const float* ws, float* out, int64_t elems, int splits,
                 hipStream_t s);
// dw-wgrad combine with transposed store: ws [splits][KK][C] -> out [C][KK]
void slab_reduce_dwt(const float* ws, float* out, int C, int KK,
                     int splits, hipStream_t s);

// --- depthwise conv (K3/K4) --------------------------------------------
// w layout [k*k, C] (C fastest).  lopx transforms the input stream;
// stats_ws ([2, dw_stats_chunks(...), C]) receives output stats partials.
void dw_conv_fwd(const void* x, const void* w, void* y,
                 int N, int H, int W, int C, int Ho, int Wo,
                 int k, int stride, int pad, LopDesc lopx, float* stats_ws,
                 hipStream_t s);
int dw_stats_chunks(int N, int Ho, int Wo, int C, int stride, int k,
                    bool wl);
bool dw_k5_wlds();  // YAMS_DW_K5LDS gate (LDS filter-bank k5 variant)
bool dw_k3_wlds();  // YAMS_DW_K3LDS gate (default off)
void dw_conv_dgrad(const void* dy, const void* w, void* dx,
                   int N, int H, int W, int C, int Ho, int Wo,
                   int k, int stride, int pad, LopDesc lopdy, LopDesc epi,
                   float* stats_ws, hipStream_t s);
// dwgt fp32 [C, k*k] (transposed combine -> reshapes to OIHW [C,1,k,k]
// without a copy); ws fp32 [chunks, k*k, C]
void dw_conv_wgrad(const void* x, const void* dy, float* dwgt, float* ws,
                   int N, int H, int W, int C, int Ho, int Wo,
                   int k, int stride, int pad, LopDesc lopx, LopDesc lopdy,
                   hipStream_t s);
int dw_wgrad_chunks(int N, int Ho, int Wo);
int64_t dw_wgrad_ws_elems(int N, int Ho, int Wo, int C, int k);

// --- stem (K1) runs as im2row + the GEMM family ------------------------
void im2row(const void* x, void* patches, int N, int H, int W, int Ho,
            int Wo, int k, int stride, int pad, int KP, hipStream_t s);

// --- BatchNorm (K5) -----------------------------------------------------
// training stats over M=N*H*W rows of z[M,C]; ws fp32 [2, chunks, C].
// Also emits ab[2C] = {a = gamma*invstd, b = beta - mean*a} for the
// normalize kernels.
void bn_stats(const void* z, float* mean, float* invstd, float* ab,
              const float* gamma, const float* beta, float* running_mean,
              float* running_var, float* ws, int64_t M, int C,
              float momentum, float eps, hipStream_t s);
int bn_stats_chunks(int64_t M);
int64_t bn_stats_ws_elems(int64_t M, int C);

// SyncBN: one [2, C] {sum, sumsq} vector for cross-rank all-reduce;
// ws sized bn_stats_ws_elems(M, C)
void bn_sums(const void* z, float* ws, float* out, int64_t M, int C,
             hipStream_t s);
// fold epilogue slabs [2, splits, C] -> [2, C]
void bn_fold_sums(const float* ws, int splits, float* out, int C,
                  hipStream_t s);

// finalize-only entry: combine sum/sumsq slabs ws[2, splits, C] produced
// by a conv epilogue into mean/invstd/ab (+running update)
void bn_finalize(const float* ws, int splits, float* mean, float* invstd,
                 float* ab, const float* gamma, const float* beta,
                 float* running_mean, float* running_var, int64_t M, int C,
                 float momentum, float eps, hipStream_t s);

// y = act(a*z + b) [+ residual], ab as produced by bn_stats
void bn_act_fwd(const void* z, const float* ab, int act, void* y,
                const void* residual, int64_t M, int C, hipStream_t s);

// backward: dgamma/dbeta + dz; ws fp32 [2, chunks, C]; coef fp32 [4C]
// apply=false computes only dgamma/dbeta and the coef[4C] (for LOP_BNBWD
// consumers); dz may be null then
void bn_act_bwd(const void* dy, const void* z, const float* mean,
                const float* invstd, const float* gamma, const float* beta,
                int act, bool training, float* dgamma, float* dbeta,
                void* dz, float* ws, float* coef, int64_t M, int C,
                bool apply, int ext_splits, hipStream_t s);
// apply-only: dz = f(dy, z, coef[4C]) with precomputed coefs
void bn_bwd_apply(const void* dy, const void* z, const float* coef,
                  int act, void* dz, int64_t M, int C, hipStream_t s);
// SE blocks: dy += avgpool-bwd(dpooled) fused into the partial pass
void bn_act_bwd_addpool(void* dy, const void* z, const void* dpooled,
                        int HW, const float* mean, const float* invstd,
                        const float* gamma, const float* beta, int act,
                        bool training, float* dgamma, float* dbeta,
                        void* dz, float* ws, float* coef, int64_t M, int C,
                        hipStream_t s);

// --- small fused elementwise/reduction kernels --------------------------
// y = act(z + bias) on [M, C] (bias nullable)
void bias_act_fwd(const void* z, const float* bias, int act, void* y,
                  int64_t M, int C, hipStream_t s);
// dz = dy * act'(z+bias); dbias = sum_m dz (nullable); ws [chunks, C]
void bias_act_bwd(const void* dy, const void* z, const float* bias, int act,
                  void* dz, float* dbias, float* ws, int64_t M, int C,
                  hipStream_t s);
int bias_bwd_chunks(int64_t M);
int64_t bias_bwd_ws_elems(int64_t M, int C);

// y[n,hw,c] = lop(x[n,hw,c]) * scale[n,c]
void rowscale_fwd(const void* x, const void* scale, void* y,
                  int N, int HW, int C, LopDesc lopx, hipStream_t s);
// dx = dy*scale ; dscale[n,c] = sum_hw dy*lop(x)
void rowscale_bwd(const void* dy, const void* x, const void* scale,
                  void* dx, void* dscale, int N, int HW, int C, LopDesc lopx,
                  hipStream_t s);

// out[n,c] = mean_hw lop(x[n,hw,c])
void avgpool_fwd(const void* x, void* out, int N, int HW, int C,
                 LopDesc lopx, hipStream_t s);
// dx[n,hw,c] = dpooled[n,c]/HW   (accumulate=true adds into dx)
void avgpool_bwd(const void* dpooled, void* dx, int N, int HW, int C,
                 bool accumulate, hipStream_t s);

// --- fused squeeze-excite (K7) ------------------------------------------
// ONE kernel per direction (se.hip).  w1 [R,C] bf16, w2 [C,R] bf16,
// b1/b2 fp32.  Saved-for-backward intermediates pooled[N,C], ha[N,R],
// sz[N,C] are fp32 (no bf16 rounding inside the SE chain).
// lopx: LOP_NONE, or LOP_BNACT to apply the producing BN+act on load and
// stream the activated input to `ymat` (folds bn_act_fwd into SE).
void se_fwd(const void* x, const void* w1, const float* b1, const void* w2,
            const float* b2, void* y, void* ymat, float* pooled,
            float* ha, float* sz, int N, int HW, int C, int R,
            LopDesc lopx, hipStream_t s);
// dx = dy*s + avgpool_bwd(dpooled); dsz[N,C]/dhz[N,R] staged for se_wgrad.
// stats_ws != null: also emit the dw BN's backward partials [2, N, C]
// (z2/bnab/bnact describe that BN) while writing dx — combine with
// bn_act_bwd(partial_splits = N)
void se_bwd(const void* dy, const void* x, const void* w1, const void* w2,
            const float* ha, const float* sz, void* dx, float* dsz,
            float* dhz, int N, int HW, int C, int R, const void* z2,
            const float* bnab, int bnact, float* stats_ws, hipStream_t s);
// dw1[R,C], dw2[C,R], db1[R], db2[C] — deterministic serial-N reduction
void se_wgrad(const float* dsz, const float* ha, const float* dhz,
              const float* pooled, float* dw1, float* dw2, float* db1,
              float* db2, int N, int C, int R, hipStream_t s);

// --- dropout (K14): stateless counter-hash mask — backward regenerates
// the mask from the same seed; y = x*mask/(1-p) fwd, dx = dy*mask/(1-p)
void dropout(const void* x, void* y, int64_t total, uint64_t seed, float p,
             bool forward, hipStream_t s);

// --- loss (K11): label-smoothed softmax CE, mean over batch -------------
// loss: fp32 scalar; dlogits fp32 [N, C]; rowloss ws fp32 [N]
void label_smooth_ce(const void* logits, const int64_t* target,
                     float smoothing, float* loss, float* dlogits,
                     float* rowloss, int N, int C, hipStream_t s);

// --- fused multi-tensor updates (K12/K13) -------------------------------
int update_max_chunks();
void fused_sgd(const float** g, float** p, float** m, const float* wd,
               const int* numel, int count, float lr, float momentum,
               bool nesterov, hipStream_t s);
void fused_rmsprop_tf(const float** g, float** p, float** ms, float** mom,
                      const float* wd, const int* numel, int count,
                      float lr, float alpha, float eps, float momentum,
                      hipStream_t s);
void fused_ema(const float** v, float** s_, const int* numel, int count,
               float decay, hipStream_t st);
// multi-tensor bf16 pack refresh: dst = bf16(src) flat (adim 0) or
// dst[A,B] = bf16(src[B,A]^T) (adim = A) — one launch per ~32 packs
void fused_repack(const float** src, void** dst, const int* numel,
                  const int* adim, int count, hipStream_t s);

}  // namespace yams
