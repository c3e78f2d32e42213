// Torch-facing launchers for the fused NHWC BatchNorm kernels
// (kernels/batchnorm.hip) — replaces ATen's channels_last batch-norm
// kernels, the largest single GPU-time consumer in the ResNet-50 config
// (profiles/README.md). Runs on the CURRENT torch stream so it composes
// with autograd like any other op.
#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "kernels/kernels.h"

namespace axon {
namespace {

bool has(const at::Tensor& t) { return t.defined() && t.numel() > 0; }
void* optr(const at::Tensor& t) { return has(t) ? t.data_ptr() : nullptr; }

void check_nhwc(const at::Tensor& t, const char* what) {
  TORCH_CHECK(t.is_cuda(), what, " must be a GPU tensor");
  TORCH_CHECK(t.dim() == 4, what, " must be 4-D (NCHW logical)");
  TORCH_CHECK(t.is_contiguous(at::MemoryFormat::ChannelsLast),
              what, " must be channels_last contiguous");
}

std::vector<at::Tensor> bn_fwd(const at::Tensor& x, const at::Tensor& gamma,
                               const at::Tensor& beta,
                               at::Tensor running_mean, at::Tensor running_var,
                               double momentum, double eps, bool training,
                               bool relu, const at::Tensor& add) {
  check_nhwc(x, "bn input");
  if (has(add)) {
    check_nhwc(add, "bn residual");
    TORCH_CHECK(add.sizes() == x.sizes() && add.scalar_type() == x.scalar_type(),
                "bn residual shape/dtype mismatch");
  }
  const int C = static_cast<int>(x.size(1));
  const int64_t rows = x.numel() / C;
  TORCH_CHECK(rows > 0 && C > 0, "bn: empty input");
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor save_mean = at::empty({C}, fopt);
  at::Tensor save_rstd = at::empty({C}, fopt);
  at::Tensor ab = at::empty({2 * C}, fopt);
  at::Tensor y = at::empty_like(x);
  at::Tensor scratch, partial;
  if (training) {
    scratch = at::empty({2 * C}, fopt);
    partial = at::empty({static_cast<int64_t>(bn_partial_rows(rows, C)) * 2 * C},
                        fopt);
  }

  int pdt = static_cast<int>(has(gamma) ? gamma.scalar_type()
                             : has(running_mean) ? running_mean.scalar_type()
                                                 : x.scalar_type());
  TORCH_CHECK(training || (has(running_mean) && has(running_var)),
              "bn eval mode needs running stats");
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(x.get_device());
  launch_bn_fwd(static_cast<int>(x.scalar_type()), pdt, x.data_ptr(),
                y.data_ptr(), optr(add), rows, C, optr(gamma), optr(beta),
                optr(running_mean), optr(running_var),
                static_cast<float>(momentum), static_cast<float>(eps),
                training, relu,
                training ? partial.data_ptr<float>() : nullptr,
                training ? scratch.data_ptr<float>() : nullptr,
                save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
                ab.data_ptr<float>(), stream.stream());
  return {y, save_mean, save_rstd};
}

std::vector<at::Tensor> bn_bwd(const at::Tensor& x, const at::Tensor& dy_in,
                               const at::Tensor& y, const at::Tensor& gamma,
                               const at::Tensor& save_mean,
                               const at::Tensor& save_rstd, bool relu,
                               bool want_dadd) {
  check_nhwc(x, "bn input");
  at::Tensor dy = dy_in.is_contiguous(at::MemoryFormat::ChannelsLast)
                      ? dy_in
                      : dy_in.contiguous(at::MemoryFormat::ChannelsLast);
  const int C = static_cast<int>(x.size(1));
  const int64_t rows = x.numel() / C;
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor partial =
      at::empty({static_cast<int64_t>(bn_partial_rows(rows, C)) * 2 * C}, fopt);
  at::Tensor scratch = at::empty({2 * C}, fopt);
  at::Tensor dgamma = at::empty({C}, fopt);
  at::Tensor dbeta = at::empty({C}, fopt);
  at::Tensor coef = at::empty({3 * C}, fopt);
  at::Tensor dx = at::empty_like(x);
  at::Tensor dadd;
  if (want_dadd) dadd = at::empty_like(x);
  if (relu) TORCH_CHECK(has(y), "bn relu backward needs the fwd output");

  int pdt = static_cast<int>(has(gamma) ? gamma.scalar_type()
                                        : x.scalar_type());
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(x.get_device());
  launch_bn_bwd(static_cast<int>(x.scalar_type()), pdt, x.data_ptr(),
                dy.data_ptr(), relu ? y.data_ptr() : nullptr, dx.data_ptr(),
                want_dadd ? dadd.data_ptr() : nullptr, rows, C, optr(gamma),
                save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(), relu,
                partial.data_ptr<float>(), scratch.data_ptr<float>(),
                dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                coef.data_ptr<float>(), stream.stream());
  if (!want_dadd) dadd = at::Tensor();
  return {dx, dgamma, dbeta, dadd};
}

// ------------------------------------------------------------- RMSNorm

std::vector<at::Tensor> rms_fwd(const at::Tensor& x, const at::Tensor& w,
                                double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "rmsnorm input must be "
              "contiguous GPU tensor");
  const int H = static_cast<int>(x.size(-1));
  TORCH_CHECK(w.is_contiguous() && w.numel() == H &&
                  w.scalar_type() == x.scalar_type(),
              "rmsnorm weight mismatch");
  int vc_ = 0, ch_ = 0;
  TORCH_CHECK(rms_bwd_pick(H, &vc_, &ch_),
              "rmsnorm fast path supports H in {64,128,256,512,1024,2048,"
              "4096} (fallback in Python)");
  const int64_t rows = x.numel() / H;
  TORCH_CHECK(rows > 0, "rmsnorm: empty input");
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor rstd = at::empty({rows}, fopt);
  at::Tensor y = at::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(x.get_device());
  launch_rms_fwd(static_cast<int>(x.scalar_type()), x.data_ptr(), w.data_ptr(),
                 y.data_ptr(), rstd.data_ptr<float>(), rows, H,
                 static_cast<float>(eps), stream.stream());
  return {y, rstd};
}

std::vector<at::Tensor> rms_bwd(const at::Tensor& x, const at::Tensor& dy_in,
                                const at::Tensor& w, const at::Tensor& rstd) {
  const int H = static_cast<int>(x.size(-1));
  const int64_t rows = x.numel() / H;
  at::Tensor dy = dy_in.is_contiguous() ? dy_in : dy_in.contiguous();
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor dw = at::empty({H}, fopt);
  at::Tensor dwpart = at::empty(
      {static_cast<int64_t>(H) * rms_bwd_blocks(rows, nullptr)}, fopt);
  at::Tensor dx = at::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(x.get_device());
  launch_rms_bwd(static_cast<int>(x.scalar_type()), x.data_ptr(), dy.data_ptr(),
                 w.data_ptr(), rstd.data_ptr<float>(), dx.data_ptr(),
                 dwpart.data_ptr<float>(), dw.data_ptr<float>(), rows, H,
                 stream.stream());
  return {dx, dw};
}

std::vector<at::Tensor> ln_fwd(const at::Tensor& x, const at::Tensor& w,
                               const at::Tensor& b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(),
              "layernorm input must be contiguous GPU tensor");
  const int H = static_cast<int>(x.size(-1));
  TORCH_CHECK(w.is_contiguous() && b.is_contiguous() && w.numel() == H &&
                  b.numel() == H && w.scalar_type() == x.scalar_type() &&
                  b.scalar_type() == x.scalar_type(),
              "layernorm weight/bias mismatch");
  int vc_ = 0, ch_ = 0;
  TORCH_CHECK(rms_bwd_pick(H, &vc_, &ch_),
              "layernorm fast path supports H in {64,128,256,512,1024,2048,"
              "4096} (fallback in Python)");
  const int64_t rows = x.numel() / H;
  TORCH_CHECK(rows > 0, "layernorm: empty input");
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor mean = at::empty({rows}, fopt);
  at::Tensor rstd = at::empty({rows}, fopt);
  at::Tensor y = at::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(x.get_device());
  launch_ln_fwd(static_cast<int>(x.scalar_type()), x.data_ptr(), w.data_ptr(),
                b.data_ptr(), y.data_ptr(), mean.data_ptr<float>(),
                rstd.data_ptr<float>(), rows, H, static_cast<float>(eps),
                stream.stream());
  return {y, mean, rstd};
}

std::vector<at::Tensor> ln_bwd(const at::Tensor& x, const at::Tensor& dy_in,
                               const at::Tensor& w, const at::Tensor& mean,
                               const at::Tensor& rstd) {
  const int H = static_cast<int>(x.size(-1));
  const int64_t rows = x.numel() / H;
  at::Tensor dy = dy_in.is_contiguous() ? dy_in : dy_in.contiguous();
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  int nblocks = rms_bwd_blocks(rows, nullptr);
  at::Tensor dw = at::empty({H}, fopt);
  at::Tensor db = at::empty({H}, fopt);
  at::Tensor dwpart = at::empty({static_cast<int64_t>(H) * nblocks}, fopt);
  at::Tensor dbpart = at::empty({static_cast<int64_t>(H) * nblocks}, fopt);
  at::Tensor dx = at::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(x.get_device());
  launch_ln_bwd(static_cast<int>(x.scalar_type()), x.data_ptr(), dy.data_ptr(),
                w.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                dx.data_ptr(), dwpart.data_ptr<float>(),
                dbpart.data_ptr<float>(), dw.data_ptr<float>(),
                db.data_ptr<float>(), rows, H, stream.stream());
  return {dx, dw, db};
}

at::Tensor rope_apply(const at::Tensor& x, const at::Tensor& cos_t,
                      const at::Tensor& sin_t, bool backward) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4,
              "rope input must be contiguous [B,S,H,D] GPU tensor");
  const int D = static_cast<int>(x.size(3));
  const int H = static_cast<int>(x.size(2));
  const int S = static_cast<int>(x.size(1));
  TORCH_CHECK(D % 2 == 0, "rope head dim must be even");
  TORCH_CHECK(cos_t.is_contiguous() && sin_t.is_contiguous() &&
                  cos_t.numel() == static_cast<int64_t>(S) * (D / 2) &&
                  cos_t.scalar_type() == x.scalar_type() &&
                  sin_t.scalar_type() == x.scalar_type(),
              "rope cos/sin must be [S, D/2] in the input dtype");
  at::Tensor y = at::empty_like(x);
  if (x.numel() == 0) return y;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(x.get_device());
  launch_rope(static_cast<int>(x.scalar_type()), x.data_ptr(), y.data_ptr(),
              cos_t.data_ptr(), sin_t.data_ptr(), x.numel(), D / 2, H, S,
              backward, stream.stream());
  return y;
}

at::Tensor swiglu_fwd(const at::Tensor& a, const at::Tensor& g) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && g.is_contiguous() &&
                  a.sizes() == g.sizes() && a.scalar_type() == g.scalar_type(),
              "swiglu inputs must be matching contiguous GPU tensors");
  at::Tensor z = at::empty_like(a);
  if (a.numel() == 0) return z;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(a.get_device());
  launch_swiglu_fwd(static_cast<int>(a.scalar_type()), a.data_ptr(),
                    g.data_ptr(), z.data_ptr(), a.numel(), stream.stream());
  return z;
}

std::vector<at::Tensor> swiglu_bwd(const at::Tensor& dz_in, const at::Tensor& a,
                                   const at::Tensor& g) {
  at::Tensor dz = dz_in.is_contiguous() ? dz_in : dz_in.contiguous();
  at::Tensor da = at::empty_like(a);
  at::Tensor dg = at::empty_like(g);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(a.get_device());
  launch_swiglu_bwd(static_cast<int>(a.scalar_type()), dz.data_ptr(),
                    a.data_ptr(), g.data_ptr(), da.data_ptr(), dg.data_ptr(),
                    a.numel(), stream.stream());
  return {da, dg};
}

// EXPERIMENTAL (docs/FA_PLAN.md): flash-attention forward, v0.
std::vector<at::Tensor> fa_fwd(const at::Tensor& q, const at::Tensor& k,
                               const at::Tensor& v, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.is_contiguous() &&
                  k.is_contiguous() && v.is_contiguous(),
              "fa_fwd: contiguous [B,H,S,D] GPU tensors required");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16 &&
                  k.scalar_type() == at::kBFloat16 &&
                  v.scalar_type() == at::kBFloat16,
              "fa_fwd: bf16 only (v0)");
  const int B = q.size(0), HQ = q.size(1), S = q.size(2), D = q.size(3);
  const int HK = k.size(1);
  TORCH_CHECK(D == 128, "fa_fwd: head dim 128 only (v0)");
  TORCH_CHECK(k.size(2) == S && v.size(2) == S && HQ % HK == 0,
              "fa_fwd: shape mismatch");
  at::Tensor o = at::empty_like(q);
  at::Tensor lse = at::empty({B, HQ, S},
                             at::TensorOptions().dtype(at::kFloat)
                                 .device(q.device()));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(q.get_device());
  launch_fa_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                lse.data_ptr<float>(), B, HQ, HK, S, D, causal,
                stream.stream());
  return {o, lse};
}

// EXPERIMENTAL: FA backward v0 (probe-only; docs/FA_PLAN.md).
std::vector<at::Tensor> fa_bwd(const at::Tensor& q, const at::Tensor& k,
                               const at::Tensor& v, const at::Tensor& o,
                               const at::Tensor& dout, const at::Tensor& lse,
                               bool causal) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.is_contiguous() &&
                  k.is_contiguous() && v.is_contiguous() &&
                  o.is_contiguous() && lse.is_contiguous(),
              "fa_bwd: contiguous GPU tensors required");
  at::Tensor dy = dout.is_contiguous() ? dout : dout.contiguous();
  const int B = q.size(0), HQ = q.size(1), S = q.size(2), D = q.size(3);
  const int HK = k.size(1);
  TORCH_CHECK(D == 128 && q.scalar_type() == at::kBFloat16,
              "fa_bwd: bf16 head dim 128 only (v0)");
  at::Tensor dq = at::empty_like(q);
  at::Tensor dk = at::empty_like(k);
  at::Tensor dv = at::empty_like(v);
  at::Tensor delta = at::empty({static_cast<int64_t>(B) * HQ * S},
                               at::TensorOptions().dtype(at::kFloat)
                                   .device(q.device()));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(q.get_device());
  launch_fa_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), dy.data_ptr(),
                o.data_ptr(), lse.data_ptr<float>(), delta.data_ptr<float>(),
                dq.data_ptr(), dk.data_ptr(), dv.data_ptr(), B, HQ, HK, S, D,
                causal, stream.stream());
  return {dq, dk, dv};
}

}  // namespace

void register_bn(pybind11::module_& m) {
  m.def("fa_fwd", &fa_fwd, py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("causal") = true);
  m.def("fa_bwd", &fa_bwd, py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("o"), py::arg("dout"), py::arg("lse"),
        py::arg("causal") = true);
  m.def("ln_fwd", &ln_fwd, py::arg("x"), py::arg("w"), py::arg("b"),
        py::arg("eps"));
  m.def("ln_bwd", &ln_bwd, py::arg("x"), py::arg("dy"), py::arg("w"),
        py::arg("mean"), py::arg("rstd"));
  m.def("rope", &rope_apply, py::arg("x"), py::arg("cos"), py::arg("sin"),
        py::arg("backward") = false);
  m.def("swiglu_fwd", &swiglu_fwd, py::arg("a"), py::arg("g"));
  m.def("swiglu_bwd", &swiglu_bwd, py::arg("dz"), py::arg("a"), py::arg("g"));
  m.def("rms_fwd", &rms_fwd, py::arg("x"), py::arg("w"), py::arg("eps"));
  m.def("rms_bwd", &rms_bwd, py::arg("x"), py::arg("dy"), py::arg("w"),
        py::arg("rstd"));
  m.def("bn_fwd", &bn_fwd, py::arg("x"), py::arg("gamma"), py::arg("beta"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("momentum"),
        py::arg("eps"), py::arg("training"), py::arg("relu"),
        py::arg("add"));
  m.def("bn_bwd", &bn_bwd, py::arg("x"), py::arg("dy"), py::arg("y"),
        py::arg("gamma"), py::arg("save_mean"), py::arg("save_rstd"),
        py::arg("relu") = false, py::arg("want_dadd") = false);
}

}  // namespace axon
