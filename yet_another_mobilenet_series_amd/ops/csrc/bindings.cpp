// Torch extension bindings for the gfx950 kernel engine (_yams_hip).
//
// Tensor contract (enforced here; see ops/hip_ops.py):
//  - activations: 4D NCHW-logical, channels_last memory (or 2D [M, C]
//    contiguous, or 4D with H=W=1 plainly contiguous), dtype bf16
//  - per-channel params/stats: fp32 contiguous
//  - weight packs: bf16 contiguous in the layout each kernel documents

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "engine.h"

namespace {

using torch::Tensor;

hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

struct Rows {
  int64_t M;
  int C;
  const void* ptr;
};

bool nhwc_ok(const Tensor& t) {
  if (t.dim() == 2) return t.is_contiguous();
  if (t.dim() != 4) return false;
  if (t.size(2) == 1 && t.size(3) == 1)
    return t.is_contiguous() || t.is_contiguous(at::MemoryFormat::ChannelsLast);
  return t.is_contiguous(at::MemoryFormat::ChannelsLast);
}

Rows rows_of(const Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a CUDA tensor");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(nhwc_ok(t), name, " must be channels_last (NHWC) contiguous");
  if (t.dim() == 2)
    return {t.size(0), int(t.size(1)), t.data_ptr()};
  return {t.size(0) * t.size(2) * t.size(3), int(t.size(1)), t.data_ptr()};
}

const float* fp32_ptr(const c10::optional<Tensor>& t, const char* name) {
  if (!t.has_value()) return nullptr;
  TORCH_CHECK(t->scalar_type() == torch::kFloat32 && t->is_contiguous(),
              name, " must be fp32 contiguous");
  return t->data_ptr<float>();
}

const float* fp32_req(const Tensor& t, const char* name) {
  TORCH_CHECK(t.scalar_type() == torch::kFloat32 && t.is_contiguous(),
              name, " must be fp32 contiguous");
  return t.data_ptr<float>();
}

void check_w(const Tensor& w) {
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 &&
                  w.is_contiguous(),
              "weight pack must be bf16 contiguous");
}

Tensor f32_ws(const Tensor& like, int64_t elems) {
  return torch::empty({elems}, like.options().dtype(torch::kFloat32));
}

yams::LopDesc make_lop(int64_t kind, int64_t act,
                       const c10::optional<Tensor>& coef,
                       const c10::optional<Tensor>& z, int C) {
  yams::LopDesc d;
  d.kind = int(kind);
  d.act = int(act);
  d.C = C;
  if (kind != 0) {
    TORCH_CHECK(coef.has_value(), "load-op requires coef tensor");
    TORCH_CHECK(coef->scalar_type() == torch::kFloat32 &&
                    coef->is_contiguous(),
                "lop coef must be fp32 contiguous");
    TORCH_CHECK(coef->numel() >= (kind == 2 ? 4 : 2) * int64_t(C),
                "lop coef too small");
    d.coef = coef->data_ptr<float>();
  }
  if (kind == 2) {
    TORCH_CHECK(z.has_value(), "bnbwd load-op requires z tensor");
    TORCH_CHECK(z->scalar_type() == torch::kBFloat16, "lop z must be bf16");
    d.z = z->data_ptr();
  }
  return d;
}

// ---- GEMM family -------------------------------------------------------

void pw_conv(const Tensor& x, const Tensor& w, Tensor& y,
             int64_t lop_kind, int64_t lop_act,
             c10::optional<Tensor> lop_coef, c10::optional<Tensor> lop_z,
             c10::optional<Tensor> stats_ws,
             int64_t bstats_act, c10::optional<Tensor> bstats_coef,
             c10::optional<Tensor> bstats_z,
             c10::optional<Tensor> residual) {
  auto xr = rows_of(x, "x");
  auto yr = rows_of(y, "y");
  check_w(w);
  TORCH_CHECK(w.dim() == 2 && w.size(1) == xr.C && w.size(0) == yr.C,
              "pw_conv weight shape mismatch");
  TORCH_CHECK(xr.M == yr.M, "pw_conv row mismatch");
  TORCH_CHECK(xr.C % 8 == 0, "pw_conv K must be a multiple of 8");
  auto lop = make_lop(lop_kind, lop_act, lop_coef, lop_z, xr.C);
  float* sw = nullptr;
  if (stats_ws.has_value()) {
    TORCH_CHECK(stats_ws->numel() >=
                    2 * int64_t(yams::pw_stats_mblocks(xr.M)) * yr.C,
                "stats_ws too small");
    sw = stats_ws->data_ptr<float>();
  }
  // bstats: the output is a dgrad dy; emit the consumer BN's backward
  // partial sums into stats_ws (EPI==2) instead of forward sum/sumsq
  yams::LopDesc epi{};
  if (bstats_coef.has_value()) {
    TORCH_CHECK(sw != nullptr, "bstats requires stats_ws");
    TORCH_CHECK(bstats_z.has_value() &&
                    bstats_z->scalar_type() == torch::kBFloat16,
                "bstats requires bf16 z");
    epi = make_lop(1, bstats_act, bstats_coef, bstats_z, yr.C);
    epi.z = bstats_z->data_ptr();
  }
  const void* rp = nullptr;
  if (residual.has_value()) {
    auto rr = rows_of(*residual, "residual");
    TORCH_CHECK(rr.M == yr.M && rr.C == yr.C,
                "pw_conv residual shape mismatch");
    rp = rr.ptr;
  }
  yams::pw_conv(xr.ptr, w.data_ptr(), y.data_ptr(), xr.M, xr.C, yr.C, lop,
                epi, sw, rp, cur_stream());
}

int64_t pw_stats_mblocks(int64_t M) { return yams::pw_stats_mblocks(M); }

void pw_conv_wgrad(const Tensor& a, const Tensor& b, Tensor& dw,
                   int64_t lopa_kind, int64_t lopa_act,
                   c10::optional<Tensor> lopa_coef,
                   c10::optional<Tensor> lopa_z, int64_t lopb_kind,
                   int64_t lopb_act, c10::optional<Tensor> lopb_coef,
                   c10::optional<Tensor> lopb_z) {
  auto ar = rows_of(a, "a");
  auto br = rows_of(b, "b");
  TORCH_CHECK(ar.M == br.M, "wgrad row mismatch");
  TORCH_CHECK(dw.scalar_type() == torch::kFloat32 && dw.is_contiguous() &&
                  dw.dim() == 2 && dw.size(0) == ar.C && dw.size(1) == br.C,
              "wgrad out must be fp32 [N1,N2]");
  auto ws = f32_ws(a, yams::pw_wgrad_ws_elems(ar.M, ar.C, br.C));
  yams::pw_conv_wgrad(ar.ptr, br.ptr, dw.data_ptr<float>(),
                      ws.data_ptr<float>(), ar.M, ar.C, br.C,
                      make_lop(lopa_kind, lopa_act, lopa_coef, lopa_z, ar.C),
                      make_lop(lopb_kind, lopb_act, lopb_coef, lopb_z, br.C),
                      cur_stream());
}

// ---- depthwise ---------------------------------------------------------

void dw_conv_fwd(const Tensor& x, const Tensor& w, Tensor& y, int64_t k,
                 int64_t stride, int64_t pad, int64_t lop_kind,
                 int64_t lop_act, c10::optional<Tensor> lop_coef,
                 c10::optional<Tensor> lop_z,
                 c10::optional<Tensor> stats_ws) {
  auto xr = rows_of(x, "x");
  auto yr = rows_of(y, "y");
  check_w(w);
  TORCH_CHECK(xr.C == yr.C && xr.C % 8 == 0, "dw channel mismatch");
  float* sw = stats_ws.has_value() ? stats_ws->data_ptr<float>() : nullptr;
  yams::dw_conv_fwd(xr.ptr, w.data_ptr(), y.data_ptr(), int(x.size(0)),
                    int(x.size(2)), int(x.size(3)), xr.C, int(y.size(2)),
                    int(y.size(3)), int(k), int(stride), int(pad),
                    make_lop(lop_kind, lop_act, lop_coef, lop_z, xr.C), sw,
                    cur_stream());
}

int64_t dw_stats_chunks(int64_t N, int64_t Ho, int64_t Wo, int64_t C,
                        int64_t stride, int64_t k) {
  return yams::dw_stats_chunks(int(N), int(Ho), int(Wo), int(C),
                               int(stride), int(k),
                               k == 5 ? yams::dw_k5_wlds()
                                      : k == 3 && yams::dw_k3_wlds());
}

void dw_conv_dgrad(const Tensor& dy, const Tensor& w, Tensor& dx, int64_t k,
                   int64_t stride, int64_t pad, int64_t lop_kind,
                   int64_t lop_act, c10::optional<Tensor> lop_coef,
                   c10::optional<Tensor> lop_z,
                   c10::optional<Tensor> stats_ws, int64_t bstats_act,
                   c10::optional<Tensor> bstats_coef,
                   c10::optional<Tensor> bstats_z) {
  auto gr = rows_of(dy, "dy");
  auto xr = rows_of(dx, "dx");
  check_w(w);
  TORCH_CHECK(gr.C == xr.C, "dw channel mismatch");
  float* sw = nullptr;
  yams::LopDesc epi{};
  if (stats_ws.has_value()) {
    TORCH_CHECK(stride == 1 && bstats_coef.has_value(),
                "dgrad bstats: stride-1 only, needs coef");
    TORCH_CHECK(bstats_z.has_value() &&
                    bstats_z->scalar_type() == torch::kBFloat16,
                "bstats requires bf16 z");
    sw = stats_ws->data_ptr<float>();
    epi = make_lop(1, bstats_act, bstats_coef, bstats_z, xr.C);
    epi.z = bstats_z->data_ptr();
  }
  yams::dw_conv_dgrad(gr.ptr, w.data_ptr(), dx.data_ptr(), int(dx.size(0)),
                      int(dx.size(2)), int(dx.size(3)), xr.C,
                      int(dy.size(2)), int(dy.size(3)), int(k), int(stride),
                      int(pad),
                      make_lop(lop_kind, lop_act, lop_coef, lop_z, gr.C),
                      epi, sw, cur_stream());
}

void dw_conv_wgrad(const Tensor& x, const Tensor& dy, Tensor& dwgt,
                   int64_t k, int64_t stride, int64_t pad,
                   int64_t lopx_kind, int64_t lopx_act,
                   c10::optional<Tensor> lopx_coef,
                   c10::optional<Tensor> lopx_z, int64_t lopdy_kind,
                   int64_t lopdy_act, c10::optional<Tensor> lopdy_coef,
                   c10::optional<Tensor> lopdy_z) {
  auto xr = rows_of(x, "x");
  auto gr = rows_of(dy, "dy");
  TORCH_CHECK(dwgt.scalar_type() == torch::kFloat32 && dwgt.is_contiguous(),
              "dw wgrad out must be fp32");
  const int N = int(x.size(0)), H = int(x.size(2)), W = int(x.size(3));
  const int Ho = int(dy.size(2)), Wo = int(dy.size(3));
  auto ws = f32_ws(x, yams::dw_wgrad_ws_elems(N, Ho, Wo, xr.C, int(k)));
  yams::dw_conv_wgrad(
      xr.ptr, gr.ptr, dwgt.data_ptr<float>(), ws.data_ptr<float>(), N, H, W,
      xr.C, Ho, Wo, int(k), int(stride), int(pad),
      make_lop(lopx_kind, lopx_act, lopx_coef, lopx_z, xr.C),
      make_lop(lopdy_kind, lopdy_act, lopdy_coef, lopdy_z, gr.C),
      cur_stream());
}

// ---- fused squeeze-excite (K7) -----------------------------------------

void se_fwd(const Tensor& x, const Tensor& w1t, const Tensor& b1,
            const Tensor& w2t, const Tensor& b2, Tensor& y,
            Tensor& pooled, Tensor& ha, Tensor& sz,
            int64_t lop_kind, int64_t lop_act,
            c10::optional<Tensor> lop_coef, c10::optional<Tensor> ymat) {
  auto xr = rows_of(x, "x");
  auto yr = rows_of(y, "y");
  check_w(w1t);
  check_w(w2t);
  const int C = xr.C, R = int(w1t.size(1));
  TORCH_CHECK(yr.C == C && yr.M == xr.M, "se_fwd shape mismatch");
  TORCH_CHECK(w1t.dim() == 2 && w1t.size(0) == C,
              "w1t must be [C, R] (transposed fc1 weight)");
  TORCH_CHECK(w2t.dim() == 2 && w2t.size(0) == R && w2t.size(1) == C,
              "w2t must be [R, C] (transposed fc2 weight)");
  const int N = int(x.size(0));
  const int HW = int(xr.M / N);
  TORCH_CHECK(pooled.numel() >= int64_t(N) * C &&
                  ha.numel() >= int64_t(N) * R &&
                  sz.numel() >= int64_t(N) * C,
              "se_fwd save buffers too small");
  void* mp = nullptr;
  if (ymat.has_value()) {
    auto mr = rows_of(*ymat, "ymat");
    TORCH_CHECK(mr.C == C && mr.M == xr.M, "ymat shape mismatch");
    mp = const_cast<void*>(mr.ptr);
  }
  yams::se_fwd(xr.ptr, w1t.data_ptr(), fp32_req(b1, "b1"),
               w2t.data_ptr(), fp32_req(b2, "b2"), y.data_ptr(), mp,
               pooled.data_ptr<float>(), ha.data_ptr<float>(),
               sz.data_ptr<float>(), N, HW, C, R,
               make_lop(lop_kind, lop_act, lop_coef, c10::nullopt, C),
               cur_stream());
}

void se_bwd(const Tensor& dy, const Tensor& x, const Tensor& w1,
            const Tensor& w2, const Tensor& ha, const Tensor& sz,
            Tensor& dx, Tensor& dsz, Tensor& dhz,
            c10::optional<Tensor> bn_z, c10::optional<Tensor> bn_ab,
            int64_t bn_act, c10::optional<Tensor> bn_stats_ws) {
  auto gr = rows_of(dy, "dy");
  auto xr = rows_of(x, "x");
  auto dxr = rows_of(dx, "dx");
  check_w(w1);
  check_w(w2);
  const int C = xr.C, R = int(w1.size(0));
  TORCH_CHECK(gr.C == C && dxr.C == C && gr.M == xr.M && dxr.M == xr.M,
              "se_bwd shape mismatch");
  const int N = int(x.size(0));
  const int HW = int(xr.M / N);
  TORCH_CHECK(dsz.numel() >= int64_t(N) * C &&
                  dhz.numel() >= int64_t(N) * R,
              "se_bwd stage buffers too small");
  const void* z2p = nullptr;
  const float* abp = nullptr;
  float* swp = nullptr;
  if (bn_stats_ws.has_value()) {
    TORCH_CHECK(bn_z.has_value() && bn_ab.has_value(),
                "se_bwd bn stats need z and ab");
    auto zr = rows_of(*bn_z, "bn_z");
    TORCH_CHECK(zr.M == xr.M && zr.C == C, "se_bwd bn_z mismatch");
    TORCH_CHECK(bn_stats_ws->numel() >= 2LL * N * C,
                "bn_stats_ws too small");
    z2p = zr.ptr;
    abp = fp32_req(*bn_ab, "bn_ab");
    swp = bn_stats_ws->data_ptr<float>();
  }
  yams::se_bwd(gr.ptr, xr.ptr, w1.data_ptr(), w2.data_ptr(),
               fp32_req(ha, "ha"), fp32_req(sz, "sz"), dx.data_ptr(),
               dsz.data_ptr<float>(), dhz.data_ptr<float>(), N, HW, C, R,
               z2p, abp, int(bn_act), swp, cur_stream());
}

void se_wgrad(const Tensor& dsz, const Tensor& ha, const Tensor& dhz,
              const Tensor& pooled, Tensor& dw1, Tensor& dw2, Tensor& db1,
              Tensor& db2) {
  const int N = int(dsz.size(0));
  const int C = int(dsz.size(1));
  const int R = int(dhz.size(1));
  TORCH_CHECK(ha.size(0) == N && ha.size(1) == R &&
                  pooled.size(0) == N && pooled.size(1) == C,
              "se_wgrad shape mismatch");
  TORCH_CHECK(dw1.sizes() == torch::IntArrayRef({R, C}) &&
                  dw2.sizes() == torch::IntArrayRef({C, R}),
              "se_wgrad out shape mismatch");
  yams::se_wgrad(fp32_req(dsz, "dsz"), fp32_req(ha, "ha"),
                 fp32_req(dhz, "dhz"), fp32_req(pooled, "pooled"),
                 dw1.data_ptr<float>(), dw2.data_ptr<float>(),
                 db1.data_ptr<float>(), db2.data_ptr<float>(), N, C, R,
                 cur_stream());
}

// ---- dropout (K14) -----------------------------------------------------

void dropout(const Tensor& x, Tensor& y, int64_t seed, double p,
             bool forward) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16,
              "dropout expects bf16 CUDA input");
  TORCH_CHECK(x.strides() == y.strides() && x.numel() == y.numel() &&
                  y.scalar_type() == torch::kBFloat16,
              "dropout out mismatch");
  TORCH_CHECK(nhwc_ok(x) || x.is_contiguous(),
              "dropout input must be contiguous");
  TORCH_CHECK(p >= 0.0 && p < 1.0, "dropout p must be in [0, 1)");
  yams::dropout(x.data_ptr(), y.data_ptr(), x.numel(), uint64_t(seed),
                float(p), forward, cur_stream());
}

// ---- stem: im2row (conv itself runs on the GEMM kernels) ---------------

void im2row(const Tensor& x, Tensor& patches, int64_t k, int64_t stride,
            int64_t pad) {
  auto xr = rows_of(x, "x");
  TORCH_CHECK(xr.C == 3, "im2row expects Cin=3");
  TORCH_CHECK(patches.dim() == 2 && patches.is_contiguous() &&
                  patches.scalar_type() == torch::kBFloat16,
              "patches must be bf16 [M, KP]");
  const int KP = int(patches.size(1));
  const int Ho = int((x.size(2) + 2 * pad - k) / stride + 1);
  const int Wo = int((x.size(3) + 2 * pad - k) / stride + 1);
  TORCH_CHECK(patches.size(0) == x.size(0) * Ho * Wo, "patches row mismatch");
  yams::im2row(xr.ptr, patches.data_ptr(), int(x.size(0)), int(x.size(2)),
               int(x.size(3)), Ho, Wo, int(k), int(stride), int(pad), KP,
               cur_stream());
}

// ---- BatchNorm ---------------------------------------------------------

void bn_stats(const Tensor& z, Tensor& mean, Tensor& invstd, Tensor& ab,
              const Tensor& gamma, const Tensor& beta,
              c10::optional<Tensor> running_mean,
              c10::optional<Tensor> running_var, double momentum,
              double eps) {
  auto zr = rows_of(z, "z");
  auto ws = f32_ws(z, yams::bn_stats_ws_elems(zr.M, zr.C));
  TORCH_CHECK(ab.numel() == 2 * zr.C, "ab must be [2C]");
  float* rm = nullptr;
  float* rv = nullptr;
  if (running_mean.has_value()) {
    rm = running_mean->data_ptr<float>();
    rv = running_var->data_ptr<float>();
  }
  yams::bn_stats(zr.ptr, mean.data_ptr<float>(), invstd.data_ptr<float>(),
                 ab.data_ptr<float>(), fp32_req(gamma, "gamma"),
                 fp32_req(beta, "beta"), rm, rv, ws.data_ptr<float>(), zr.M,
                 zr.C, float(momentum), float(eps), cur_stream());
}

void bn_act_fwd(const Tensor& z, const Tensor& ab, int64_t act, Tensor& y,
                c10::optional<Tensor> residual) {
  auto zr = rows_of(z, "z");
  TORCH_CHECK(zr.C % 8 == 0, "C must be a multiple of 8");
  const void* res = nullptr;
  if (residual.has_value()) res = rows_of(*residual, "residual").ptr;
  yams::bn_act_fwd(zr.ptr, fp32_req(ab, "ab"), int(act), y.data_ptr(), res,
                   zr.M, zr.C, cur_stream());
}

void bn_act_bwd(const Tensor& dy, const Tensor& z, const Tensor& mean,
                const Tensor& invstd, const Tensor& gamma,
                const Tensor& beta, int64_t act, bool training,
                Tensor& dgamma, Tensor& dbeta, Tensor& dz,
                c10::optional<Tensor> partial_ws, int64_t partial_splits) {
  auto zr = rows_of(z, "z");
  TORCH_CHECK(zr.C % 8 == 0, "C must be a multiple of 8");
  auto gr = rows_of(dy, "dy");
  TORCH_CHECK(zr.M == gr.M && zr.C == gr.C, "bn bwd shape mismatch");
  auto coef = f32_ws(z, 4 * zr.C);
  float* wp;
  Tensor ws;
  if (partial_ws.has_value()) {
    TORCH_CHECK(partial_splits > 0 &&
                    partial_ws->numel() >= 2 * partial_splits * zr.C,
                "partial_ws too small");
    wp = partial_ws->data_ptr<float>();
  } else {
    ws = f32_ws(z, yams::bn_stats_ws_elems(zr.M, zr.C));
    wp = ws.data_ptr<float>();
  }
  yams::bn_act_bwd(gr.ptr, zr.ptr, fp32_req(mean, "mean"),
                   fp32_req(invstd, "invstd"), fp32_req(gamma, "gamma"),
                   fp32_req(beta, "beta"), int(act), training,
                   dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                   dz.data_ptr(), wp, coef.data_ptr<float>(), zr.M, zr.C,
                   true, int(partial_splits), cur_stream());
}

// SE path: bn backward with the avg-pool grad added to dy in the
// partial pass (replaces avgpool_bwd_add + bn_act_bwd)
void bn_act_bwd_addpool(Tensor& dy, const Tensor& z, const Tensor& dpooled,
                        const Tensor& mean, const Tensor& invstd,
                        const Tensor& gamma, const Tensor& beta,
                        int64_t act, bool training, Tensor& dgamma,
                        Tensor& dbeta, Tensor& dz) {
  auto zr = rows_of(z, "z");
  TORCH_CHECK(zr.C % 8 == 0, "C must be a multiple of 8");
  auto gr = rows_of(dy, "dy");
  TORCH_CHECK(zr.M == gr.M && zr.C == gr.C, "bn bwd shape mismatch");
  TORCH_CHECK(dpooled.scalar_type() == torch::kBFloat16 &&
                  dpooled.numel() == int64_t(dy.size(0)) * zr.C,
              "dpooled must be bf16 [N, C]");
  const int HW = int(zr.M / dy.size(0));
  auto ws = f32_ws(z, yams::bn_stats_ws_elems(zr.M, zr.C));
  auto coef = f32_ws(z, 4 * zr.C);
  yams::bn_act_bwd_addpool(
      const_cast<void*>(gr.ptr), zr.ptr, dpooled.data_ptr(), HW,
      fp32_req(mean, "mean"), fp32_req(invstd, "invstd"),
      fp32_req(gamma, "gamma"), fp32_req(beta, "beta"), int(act), training,
      dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), dz.data_ptr(),
      ws.data_ptr<float>(), coef.data_ptr<float>(), zr.M, zr.C,
      cur_stream());
}

// reduce-only backward: dgamma/dbeta + the coef[4C] used by LOP_BNBWD
// consumers (no dz materialization)
void bn_bwd_stats(const Tensor& dy, const Tensor& z, const Tensor& mean,
                  const Tensor& invstd, const Tensor& gamma,
                  const Tensor& beta, int64_t act, bool training,
                  Tensor& dgamma, Tensor& dbeta, Tensor& coef) {
  auto zr = rows_of(z, "z");
  TORCH_CHECK(zr.C % 8 == 0, "C must be a multiple of 8");
  auto gr = rows_of(dy, "dy");
  TORCH_CHECK(zr.M == gr.M && zr.C == gr.C, "bn bwd shape mismatch");
  TORCH_CHECK(coef.numel() == 4 * zr.C, "coef must be [4C]");
  auto ws = f32_ws(z, yams::bn_stats_ws_elems(zr.M, zr.C));
  yams::bn_act_bwd(gr.ptr, zr.ptr, fp32_req(mean, "mean"),
                   fp32_req(invstd, "invstd"), fp32_req(gamma, "gamma"),
                   fp32_req(beta, "beta"), int(act), training,
                   dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                   nullptr, ws.data_ptr<float>(), coef.data_ptr<float>(),
                   zr.M, zr.C, false, 0, cur_stream());
}

void bn_bwd_apply(const Tensor& dy, const Tensor& z, const Tensor& coef,
                  int64_t act, Tensor& dz) {
  auto zr = rows_of(z, "z");
  auto gr = rows_of(dy, "dy");
  TORCH_CHECK(zr.M == gr.M && zr.C == gr.C && zr.C % 8 == 0,
              "bn_bwd_apply shape mismatch");
  TORCH_CHECK(coef.numel() >= 4 * zr.C, "coef must be [4C]");
  yams::bn_bwd_apply(gr.ptr, zr.ptr, fp32_req(coef, "coef"), int(act),
                     dz.data_ptr(), zr.M, zr.C, cur_stream());
}

// SyncBN: per-rank {sum, sumsq} [2, C] for cross-rank all-reduce
void bn_sums(const Tensor& z, Tensor& out) {
  auto zr = rows_of(z, "z");
  TORCH_CHECK(zr.C % 8 == 0, "C must be a multiple of 8");
  TORCH_CHECK(out.numel() == 2 * zr.C &&
                  out.scalar_type() == torch::kFloat32,
              "out must be fp32 [2C]");
  auto ws = f32_ws(z, yams::bn_stats_ws_elems(zr.M, zr.C));
  yams::bn_sums(zr.ptr, ws.data_ptr<float>(), out.data_ptr<float>(),
                zr.M, zr.C, cur_stream());
}

// fold conv-epilogue slabs [2, splits, C] into [2, C] (SyncBN with
// epilogue-produced statistics)
void bn_fold(const Tensor& ws, int64_t splits, Tensor& out) {
  const int C = int(out.numel() / 2);
  TORCH_CHECK(out.scalar_type() == torch::kFloat32 &&
                  ws.numel() >= 2 * splits * C,
              "bn_fold shapes");
  yams::bn_fold_sums(fp32_req(ws, "ws"), int(splits), out.data_ptr<float>(),
                     C, cur_stream());
}

// combine conv-epilogue stats slabs [2, splits, C] into mean/invstd/ab
void bn_finalize(const Tensor& ws, int64_t splits, Tensor& mean,
                 Tensor& invstd, Tensor& ab, const Tensor& gamma,
                 const Tensor& beta, c10::optional<Tensor> running_mean,
                 c10::optional<Tensor> running_var, double momentum,
                 double eps, int64_t M) {
  const int C = int(mean.numel());
  float* rm = nullptr;
  float* rv = nullptr;
  if (running_mean.has_value()) {
    rm = running_mean->data_ptr<float>();
    rv = running_var->data_ptr<float>();
  }
  yams::bn_finalize(fp32_req(ws, "ws"), int(splits),
                    mean.data_ptr<float>(), invstd.data_ptr<float>(),
                    ab.data_ptr<float>(), fp32_req(gamma, "gamma"),
                    fp32_req(beta, "beta"), rm, rv, M, C, float(momentum),
                    float(eps), cur_stream());
}

// ---- misc --------------------------------------------------------------

void bias_act_fwd(const Tensor& z, c10::optional<Tensor> bias, int64_t act,
                  Tensor& y) {
  auto zr = rows_of(z, "z");
  yams::bias_act_fwd(zr.ptr, fp32_ptr(bias, "bias"), int(act), y.data_ptr(),
                     zr.M, zr.C, cur_stream());
}

void bias_act_bwd(const Tensor& dy, const Tensor& z,
                  c10::optional<Tensor> bias, int64_t act, Tensor& dz,
                  c10::optional<Tensor> dbias) {
  auto zr = rows_of(z, "z");
  auto gr = rows_of(dy, "dy");
  TORCH_CHECK(zr.M == gr.M && zr.C == gr.C, "bias bwd shape mismatch");
  float* dbp = nullptr;
  Tensor ws;
  if (dbias.has_value()) {
    dbp = dbias->data_ptr<float>();
    ws = f32_ws(z, yams::bias_bwd_ws_elems(zr.M, zr.C));
  }
  yams::bias_act_bwd(gr.ptr, zr.ptr, fp32_ptr(bias, "bias"), int(act),
                     dz.data_ptr(), dbp,
                     dbias.has_value() ? ws.data_ptr<float>() : nullptr,
                     zr.M, zr.C, cur_stream());
}

void rowscale_fwd(const Tensor& x, const Tensor& scale, Tensor& y,
                  int64_t lop_kind, int64_t lop_act,
                  c10::optional<Tensor> lop_coef,
                  c10::optional<Tensor> lop_z) {
  auto xr = rows_of(x, "x");
  yams::rowscale_fwd(xr.ptr, scale.data_ptr(), y.data_ptr(),
                     int(x.size(0)), int(x.size(2) * x.size(3)), xr.C,
                     make_lop(lop_kind, lop_act, lop_coef, lop_z, xr.C),
                     cur_stream());
}

void rowscale_bwd(const Tensor& dy, const Tensor& x, const Tensor& scale,
                  Tensor& dx, Tensor& dscale, int64_t lop_kind,
                  int64_t lop_act, c10::optional<Tensor> lop_coef,
                  c10::optional<Tensor> lop_z) {
  auto xr = rows_of(x, "x");
  yams::rowscale_bwd(dy.data_ptr(), xr.ptr, scale.data_ptr(), dx.data_ptr(),
                     dscale.data_ptr(), int(x.size(0)),
                     int(x.size(2) * x.size(3)), xr.C,
                     make_lop(lop_kind, lop_act, lop_coef, lop_z, xr.C),
                     cur_stream());
}

void avgpool_fwd(const Tensor& x, Tensor& out, int64_t lop_kind,
                 int64_t lop_act, c10::optional<Tensor> lop_coef,
                 c10::optional<Tensor> lop_z) {
  auto xr = rows_of(x, "x");
  yams::avgpool_fwd(xr.ptr, out.data_ptr(), int(x.size(0)),
                    int(x.size(2) * x.size(3)), xr.C,
                    make_lop(lop_kind, lop_act, lop_coef, lop_z, xr.C),
                    cur_stream());
}

void avgpool_bwd(const Tensor& dpooled, Tensor& dx) {
  auto xr = rows_of(dx, "dx");
  yams::avgpool_bwd(dpooled.data_ptr(), dx.data_ptr(), int(dx.size(0)),
                    int(dx.size(2) * dx.size(3)), xr.C, false, cur_stream());
}

void avgpool_bwd_add(const Tensor& dpooled, Tensor& dx) {
  auto xr = rows_of(dx, "dx");
  yams::avgpool_bwd(dpooled.data_ptr(), dx.data_ptr(), int(dx.size(0)),
                    int(dx.size(2) * dx.size(3)), xr.C, true, cur_stream());
}

void label_smooth_ce(const Tensor& logits, const Tensor& target,
                     double smoothing, Tensor& loss, Tensor& dlogits) {
  auto lr = rows_of(logits, "logits");
  TORCH_CHECK(target.scalar_type() == torch::kInt64 && target.is_contiguous(),
              "target must be int64");
  auto rowloss = f32_ws(loss, lr.M);
  yams::label_smooth_ce(lr.ptr, target.data_ptr<int64_t>(),
                        float(smoothing), loss.data_ptr<float>(),
                        dlogits.data_ptr<float>(), rowloss.data_ptr<float>(),
                        int(lr.M), lr.C, cur_stream());
}

// ---- fused multi-tensor updates ----------------------------------------

void check_f32(const Tensor& t) {
  // dense (possibly channels_last) fp32: the flat update only needs every
  // element covered once; the Python side guarantees grads/buffers share
  // the param's memory layout
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_non_overlapping_and_dense(),
              "update tensors must be fp32 cuda dense");
}

void fused_repack(const std::vector<Tensor>& srcs,
                  const std::vector<Tensor>& dsts,
                  const std::vector<int64_t>& adims) {
  const int maxc = yams::update_max_chunks();
  const int n = int(srcs.size());
  TORCH_CHECK(int(dsts.size()) == n && int(adims.size()) == n,
              "fused_repack list length mismatch");
  for (int base = 0; base < n; base += maxc) {
    const int cnt = std::min(maxc, n - base);
    const float* src[32];
    void* dst[32];
    int numel[32];
    int adim[32];
    for (int c = 0; c < cnt; ++c) {
      const Tensor& s = srcs[base + c];
      const Tensor& d = dsts[base + c];
      TORCH_CHECK(s.scalar_type() == torch::kFloat32 &&
                      d.scalar_type() == torch::kBFloat16 &&
                      d.is_contiguous() && s.numel() == d.numel(),
                  "fused_repack src/dst mismatch");
      const int a = int(adims[base + c]);
      TORCH_CHECK(a == 0 || (a > 0 && s.numel() % a == 0),
                  "fused_repack bad adim");
      src[c] = s.data_ptr<float>();
      dst[c] = d.data_ptr();
      numel[c] = int(s.numel());
      adim[c] = a;
    }
    yams::fused_repack(src, dst, numel, adim, cnt, cur_stream());
  }
}

void fused_sgd(const std::vector<Tensor>& params,
               const std::vector<Tensor>& grads,
               const std::vector<Tensor>& moms,
               const std::vector<double>& wds, double lr, double momentum,
               bool nesterov) {
  const int maxc = yams::update_max_chunks();
  const int n = int(params.size());
  for (int base = 0; base < n; base += maxc) {
    const int cnt = std::min(maxc, n - base);
    const float* g[32];
    float* p[32];
    float* m[32];
    float wd[32];
    int numel[32];
    for (int c = 0; c < cnt; ++c) {
      check_f32(params[base + c]);
      g[c] = grads[base + c].data_ptr<float>();
      p[c] = params[base + c].data_ptr<float>();
      m[c] = moms[base + c].data_ptr<float>();
      wd[c] = float(wds[base + c]);
      numel[c] = int(params[base + c].numel());
    }
    yams::fused_sgd(g, p, m, wd, numel, cnt, float(lr), float(momentum),
                    nesterov, cur_stream());
  }
}

void fused_rmsprop_tf(const std::vector<Tensor>& params,
                      const std::vector<Tensor>& grads,
                      const std::vector<Tensor>& mss,
                      const std::vector<Tensor>& moms,
                      const std::vector<double>& wds, double lr,
                      double alpha, double eps, double momentum) {
  const int maxc = yams::update_max_chunks();
  const int n = int(params.size());
  for (int base = 0; base < n; base += maxc) {
    const int cnt = std::min(maxc, n - base);
    const float* g[32];
    float* p[32];
    float* ms[32];
    float* mm[32];
    float wd[32];
    int numel[32];
    for (int c = 0; c < cnt; ++c) {
      check_f32(params[base + c]);
      g[c] = grads[base + c].data_ptr<float>();
      p[c] = params[base + c].data_ptr<float>();
      ms[c] = mss[base + c].data_ptr<float>();
      mm[c] = moms[base + c].data_ptr<float>();
      wd[c] = float(wds[base + c]);
      numel[c] = int(params[base + c].numel());
    }
    yams::fused_rmsprop_tf(g, p, ms, mm, wd, numel, cnt, float(lr),
                           float(alpha), float(eps), float(momentum),
                           cur_stream());
  }
}

void fused_ema(const std::vector<Tensor>& values,
               const std::vector<Tensor>& shadows, double decay) {
  const int maxc = yams::update_max_chunks();
  const int n = int(values.size());
  for (int base = 0; base < n; base += maxc) {
    const int cnt = std::min(maxc, n - base);
    const float* v[32];
    float* s[32];
    int numel[32];
    for (int c = 0; c < cnt; ++c) {
      v[c] = values[base + c].data_ptr<float>();
      s[c] = shadows[base + c].data_ptr<float>();
      numel[c] = int(values[base + c].numel());
    }
    yams::fused_ema(v, s, numel, cnt, float(decay), cur_stream());
  }
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X (gfx950) CDNA4 kernel engine for the MobileNet zoo";
  using namespace pybind11::literals;
  auto nolop = py::arg_v("lop_kind", 0);
  m.def("pw_conv", &pw_conv, "y[M,N] = x[M,K] @ w[N,K]^T (MFMA bf16)",
        "x"_a, "w"_a, "y"_a, "lop_kind"_a = 0, "lop_act"_a = 0,
        "lop_coef"_a = c10::nullopt, "lop_z"_a = c10::nullopt,
        "stats_ws"_a = c10::nullopt, "bstats_act"_a = 0,
        "bstats_coef"_a = c10::nullopt, "bstats_z"_a = c10::nullopt,
        "residual"_a = c10::nullopt);
  m.def("pw_stats_mblocks", &pw_stats_mblocks);
  m.def("pw_conv_wgrad", &pw_conv_wgrad, "dw[N1,N2] = a^T @ b (split-M)",
        "a"_a, "b"_a, "dw"_a, "lopa_kind"_a = 0, "lopa_act"_a = 0,
        "lopa_coef"_a = c10::nullopt, "lopa_z"_a = c10::nullopt,
        "lopb_kind"_a = 0, "lopb_act"_a = 0, "lopb_coef"_a = c10::nullopt,
        "lopb_z"_a = c10::nullopt);
  m.def("dw_conv_fwd", &dw_conv_fwd, "x"_a, "w"_a, "y"_a, "k"_a, "stride"_a,
        "pad"_a, "lop_kind"_a = 0, "lop_act"_a = 0,
        "lop_coef"_a = c10::nullopt, "lop_z"_a = c10::nullopt,
        "stats_ws"_a = c10::nullopt);
  m.def("dw_stats_chunks", &dw_stats_chunks);
  m.def("dw_conv_dgrad", &dw_conv_dgrad, "dy"_a, "w"_a, "dx"_a, "k"_a,
        "stride"_a, "pad"_a, "lop_kind"_a = 0, "lop_act"_a = 0,
        "lop_coef"_a = c10::nullopt, "lop_z"_a = c10::nullopt,
        "stats_ws"_a = c10::nullopt, "bstats_act"_a = 0,
        "bstats_coef"_a = c10::nullopt, "bstats_z"_a = c10::nullopt);
  m.def("dw_conv_wgrad", &dw_conv_wgrad, "x"_a, "dy"_a, "dw"_a, "k"_a,
        "stride"_a, "pad"_a, "lopx_kind"_a = 0, "lopx_act"_a = 0,
        "lopx_coef"_a = c10::nullopt, "lopx_z"_a = c10::nullopt,
        "lopdy_kind"_a = 0, "lopdy_act"_a = 0,
        "lopdy_coef"_a = c10::nullopt, "lopdy_z"_a = c10::nullopt);
  m.def("se_fwd", &se_fwd, "fused squeeze-excite forward (one kernel)",
        "x"_a, "w1"_a, "b1"_a, "w2"_a, "b2"_a, "y"_a, "pooled"_a, "ha"_a,
        "sz"_a, "lop_kind"_a = 0, "lop_act"_a = 0,
        "lop_coef"_a = c10::nullopt, "ymat"_a = c10::nullopt);
  m.def("se_bwd", &se_bwd, "dy"_a, "x"_a, "w1"_a, "w2"_a, "ha"_a, "sz"_a,
        "dx"_a, "dsz"_a, "dhz"_a, "bn_z"_a = c10::nullopt,
        "bn_ab"_a = c10::nullopt, "bn_act"_a = 0,
        "bn_stats_ws"_a = c10::nullopt);
  m.def("se_wgrad", &se_wgrad, "dsz"_a, "ha"_a, "dhz"_a, "pooled"_a,
        "dw1"_a, "dw2"_a, "db1"_a, "db2"_a);
  m.def("dropout", &dropout, "x"_a, "y"_a, "seed"_a, "p"_a,
        "forward"_a = true);
  m.def("im2row", &im2row);
  m.def("bn_stats", &bn_stats);
  m.def("bn_finalize", &bn_finalize, "ws"_a, "splits"_a, "mean"_a,
        "invstd"_a, "ab"_a, "gamma"_a, "beta"_a,
        "running_mean"_a = c10::nullopt, "running_var"_a = c10::nullopt,
        "momentum"_a = 0.1, "eps"_a = 1e-5, "M"_a = 1);
  m.def("bn_act_fwd", &bn_act_fwd);
  m.def("bn_act_bwd", &bn_act_bwd, "dy"_a, "z"_a, "mean"_a, "invstd"_a,
        "gamma"_a, "beta"_a, "act"_a, "training"_a, "dgamma"_a, "dbeta"_a,
        "dz"_a, "partial_ws"_a = c10::nullopt, "partial_splits"_a = 0);
  m.def("bn_bwd_stats", &bn_bwd_stats);
  m.def("bn_sums", &bn_sums);
  m.def("bn_fold", &bn_fold);
  m.def("bn_bwd_apply", &bn_bwd_apply);
  m.def("bn_act_bwd_addpool", &bn_act_bwd_addpool);
  m.def("bias_act_fwd", &bias_act_fwd);
  m.def("bias_act_bwd", &bias_act_bwd);
  m.def("rowscale_fwd", &rowscale_fwd, "x"_a, "scale"_a, "y"_a,
        "lop_kind"_a = 0, "lop_act"_a = 0, "lop_coef"_a = c10::nullopt,
        "lop_z"_a = c10::nullopt);
  m.def("rowscale_bwd", &rowscale_bwd, "dy"_a, "x"_a, "scale"_a, "dx"_a,
        "dscale"_a, "lop_kind"_a = 0, "lop_act"_a = 0,
        "lop_coef"_a = c10::nullopt, "lop_z"_a = c10::nullopt);
  m.def("avgpool_fwd", &avgpool_fwd, "x"_a, "out"_a, "lop_kind"_a = 0,
        "lop_act"_a = 0, "lop_coef"_a = c10::nullopt,
        "lop_z"_a = c10::nullopt);
  m.def("avgpool_bwd", &avgpool_bwd);
  m.def("avgpool_bwd_add", &avgpool_bwd_add);
  m.def("label_smooth_ce", &label_smooth_ce);
  m.def("fused_repack", &fused_repack);
  m.def("fused_sgd", &fused_sgd);
  m.def("fused_rmsprop_tf", &fused_rmsprop_tf);
  m.def("fused_ema", &fused_ema);
}
