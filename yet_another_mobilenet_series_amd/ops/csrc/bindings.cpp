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

// ---- GEMM family -------------------------------------------------------

void pw_conv(const Tensor& x, const Tensor& w, Tensor& y) {
  auto xr = rows_of(x, "x");
  auto yr = rows_of(y, "y");
  check_w(w);
  TORCH_CHECK(w.dim() == 2 && w.size(1) == xr.C && w.size(0) == yr.C,
              "pw_conv weight shape mismatch");
  TORCH_CHECK(xr.M == yr.M, "pw_conv row mismatch");
  TORCH_CHECK(xr.C % 8 == 0, "pw_conv K must be a multiple of 8");
  yams::pw_conv(xr.ptr, w.data_ptr(), y.data_ptr(), xr.M, xr.C, yr.C,
                cur_stream());
}

void pw_conv_wgrad(const Tensor& a, const Tensor& b, Tensor& dw) {
  auto ar = rows_of(a, "a");
  auto br = rows_of(b, "b");
  TORCH_CHECK(ar.M == br.M, "wgrad row mismatch");
  TORCH_CHECK(dw.scalar_type() == torch::kFloat32 && dw.is_contiguous() &&
                  dw.dim() == 2 && dw.size(0) == ar.C && dw.size(1) == br.C,
              "wgrad out must be fp32 [N1,N2]");
  auto ws = f32_ws(a, yams::pw_wgrad_ws_elems(ar.M, ar.C, br.C));
  yams::pw_conv_wgrad(ar.ptr, br.ptr, dw.data_ptr<float>(),
                      ws.data_ptr<float>(), ar.M, ar.C, br.C, cur_stream());
}

// ---- depthwise ---------------------------------------------------------

void dw_conv_fwd(const Tensor& x, const Tensor& w, Tensor& y, int64_t k,
                 int64_t stride, int64_t pad) {
  auto xr = rows_of(x, "x");
  auto yr = rows_of(y, "y");
  check_w(w);
  TORCH_CHECK(xr.C == yr.C && xr.C % 8 == 0, "dw channel mismatch");
  yams::dw_conv_fwd(xr.ptr, w.data_ptr(), y.data_ptr(), int(x.size(0)),
                    int(x.size(2)), int(x.size(3)), xr.C, int(y.size(2)),
                    int(y.size(3)), int(k), int(stride), int(pad),
                    cur_stream());
}

void dw_conv_dgrad(const Tensor& dy, const Tensor& w, Tensor& dx, int64_t k,
                   int64_t stride, int64_t pad) {
  auto gr = rows_of(dy, "dy");
  auto xr = rows_of(dx, "dx");
  check_w(w);
  TORCH_CHECK(gr.C == xr.C, "dw channel mismatch");
  yams::dw_conv_dgrad(gr.ptr, w.data_ptr(), dx.data_ptr(), int(dx.size(0)),
                      int(dx.size(2)), int(dx.size(3)), xr.C,
                      int(dy.size(2)), int(dy.size(3)), int(k), int(stride),
                      int(pad), cur_stream());
}

void dw_conv_wgrad(const Tensor& x, const Tensor& dy, Tensor& dwgt,
                   int64_t k, int64_t stride, int64_t pad) {
  auto xr = rows_of(x, "x");
  auto gr = rows_of(dy, "dy");
  TORCH_CHECK(dwgt.scalar_type() == torch::kFloat32 && dwgt.is_contiguous(),
              "dw wgrad out must be fp32");
  const int N = int(x.size(0)), H = int(x.size(2)), W = int(x.size(3));
  const int Ho = int(dy.size(2)), Wo = int(dy.size(3));
  auto ws = f32_ws(x, yams::dw_wgrad_ws_elems(N, Ho, Wo, xr.C, int(k)));
  yams::dw_conv_wgrad(xr.ptr, gr.ptr, dwgt.data_ptr<float>(),
                      ws.data_ptr<float>(), N, H, W, xr.C, Ho, Wo, int(k),
                      int(stride), int(pad), cur_stream());
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
  const void* res = nullptr;
  if (residual.has_value()) res = rows_of(*residual, "residual").ptr;
  yams::bn_act_fwd(zr.ptr, fp32_req(ab, "ab"), int(act), y.data_ptr(), res,
                   zr.M, zr.C, cur_stream());
}

void bn_act_bwd(const Tensor& dy, const Tensor& z, const Tensor& mean,
                const Tensor& invstd, const Tensor& gamma,
                const Tensor& beta, int64_t act, bool training,
                Tensor& dgamma, Tensor& dbeta, Tensor& dz) {
  auto zr = rows_of(z, "z");
  auto gr = rows_of(dy, "dy");
  TORCH_CHECK(zr.M == gr.M && zr.C == gr.C, "bn bwd shape mismatch");
  auto ws = f32_ws(z, yams::bn_stats_ws_elems(zr.M, zr.C));
  auto coef = f32_ws(z, 4 * zr.C);
  yams::bn_act_bwd(gr.ptr, zr.ptr, fp32_req(mean, "mean"),
                   fp32_req(invstd, "invstd"), fp32_req(gamma, "gamma"),
                   fp32_req(beta, "beta"), int(act), training,
                   dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                   dz.data_ptr(), ws.data_ptr<float>(),
                   coef.data_ptr<float>(), zr.M, zr.C, cur_stream());
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

void rowscale_fwd(const Tensor& x, const Tensor& scale, Tensor& y) {
  auto xr = rows_of(x, "x");
  yams::rowscale_fwd(xr.ptr, scale.data_ptr(), y.data_ptr(),
                     int(x.size(0)), int(x.size(2) * x.size(3)), xr.C,
                     cur_stream());
}

void rowscale_bwd(const Tensor& dy, const Tensor& x, const Tensor& scale,
                  Tensor& dx, Tensor& dscale) {
  auto xr = rows_of(x, "x");
  yams::rowscale_bwd(dy.data_ptr(), xr.ptr, scale.data_ptr(), dx.data_ptr(),
                     dscale.data_ptr(), int(x.size(0)),
                     int(x.size(2) * x.size(3)), xr.C, cur_stream());
}

void avgpool_fwd(const Tensor& x, Tensor& out) {
  auto xr = rows_of(x, "x");
  yams::avgpool_fwd(xr.ptr, out.data_ptr(), int(x.size(0)),
                    int(x.size(2) * x.size(3)), xr.C, cur_stream());
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

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X (gfx950) CDNA4 kernel engine for the MobileNet zoo";
  m.def("pw_conv", &pw_conv, "y[M,N] = x[M,K] @ w[N,K]^T (MFMA bf16)");
  m.def("pw_conv_wgrad", &pw_conv_wgrad, "dw[N1,N2] = a^T @ b (split-M)");
  m.def("dw_conv_fwd", &dw_conv_fwd);
  m.def("dw_conv_dgrad", &dw_conv_dgrad);
  m.def("dw_conv_wgrad", &dw_conv_wgrad);
  m.def("im2row", &im2row);
  m.def("bn_stats", &bn_stats);
  m.def("bn_act_fwd", &bn_act_fwd);
  m.def("bn_act_bwd", &bn_act_bwd);
  m.def("bias_act_fwd", &bias_act_fwd);
  m.def("bias_act_bwd", &bias_act_bwd);
  m.def("rowscale_fwd", &rowscale_fwd);
  m.def("rowscale_bwd", &rowscale_bwd);
  m.def("avgpool_fwd", &avgpool_fwd);
  m.def("avgpool_bwd", &avgpool_bwd);
  m.def("avgpool_bwd_add", &avgpool_bwd_add);
  m.def("label_smooth_ce", &label_smooth_ce);
}
