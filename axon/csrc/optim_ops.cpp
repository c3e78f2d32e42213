// Torch-facing launchers for the fused optimizer kernels (K6): build
// multi-tensor chunk descriptors, stage them through pinned memory, launch
// on the current torch stream. One kernel launch per parameter group.
#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "kernels/optimizers.h"

namespace axon {
namespace {

int64_t pick_chunk_elems(int64_t total) {
  // Target ~4096 chunks so 256 CUs stay filled; clamp to [16K, 256K].
  int64_t c = total / 4096;
  int64_t v = 16384;
  while (v < c && v < (1 << 18)) v <<= 1;
  return v;
}

struct ChunkPlan {
  at::Tensor host;  // pinned byte buffer of OptChunk[]
  at::Tensor dev;
  int nchunks = 0;
};

ChunkPlan build_chunks(const std::vector<at::Tensor>& ps,
                       const std::vector<at::Tensor>& gs,
                       const std::vector<at::Tensor>& ms,
                       const std::vector<at::Tensor>& vs,
                       float* u_base /* fp32 scratch or null */) {
  int64_t total = 0;
  for (auto& p : ps) total += p.numel();
  int64_t ce = pick_chunk_elems(total);
  std::vector<OptChunk> chunks;
  int64_t uoff = 0;
  for (size_t t = 0; t < ps.size(); ++t) {
    auto& p = ps[t];
    // Elementwise over dense storage: any non-overlapping-and-dense layout
    // (incl. channels_last) works as long as param/grad/state agree on
    // strides — same storage index = same logical element on all of them.
    TORCH_CHECK(p.is_non_overlapping_and_dense() &&
                    gs[t].is_non_overlapping_and_dense(),
                "fused optimizer requires dense params/grads");
    TORCH_CHECK(p.numel() <= 1 || gs[t].strides() == p.strides(),
                "fused optimizer: grad layout must match param layout");
    TORCH_CHECK(ms.empty() || p.numel() <= 1 || ms[t].strides() == p.strides(),
                "fused optimizer: state layout must match param layout");
    TORCH_CHECK(vs.empty() || p.numel() <= 1 || vs[t].strides() == p.strides(),
                "fused optimizer: state layout must match param layout");
    size_t esize = p.element_size();
    char* pp = static_cast<char*>(p.data_ptr());
    char* gg = static_cast<char*>(gs[t].data_ptr());
    char* mm = ms.empty() ? nullptr : static_cast<char*>(ms[t].data_ptr());
    char* vv = vs.empty() ? nullptr : static_cast<char*>(vs[t].data_ptr());
    int64_t n = p.numel();
    for (int64_t off = 0; off < n; off += ce) {
      OptChunk c;
      c.p = pp + off * esize;
      c.g = gg + off * esize;
      c.m = mm ? mm + off * esize : nullptr;
      c.v = vv ? vv + off * esize : nullptr;
      c.u = u_base ? u_base + uoff + off : nullptr;
      c.n = std::min<int64_t>(ce, n - off);
      c.tidx = static_cast<int64_t>(t);
      chunks.push_back(c);
    }
    uoff += n;
  }
  ChunkPlan plan;
  plan.nchunks = static_cast<int>(chunks.size());
  int64_t bytes = static_cast<int64_t>(chunks.size() * sizeof(OptChunk));
  plan.host = at::empty({bytes}, at::TensorOptions()
                                     .dtype(at::kByte)
                                     .pinned_memory(true));
  std::memcpy(plan.host.data_ptr(), chunks.data(), bytes);
  plan.dev = at::empty({bytes}, at::TensorOptions()
                                    .dtype(at::kByte)
                                    .device(ps[0].device()));
  plan.dev.copy_(plan.host, /*non_blocking=*/true);
  return plan;
}

int dtype_code(const at::Tensor& t) {
  return static_cast<int>(t.scalar_type());
}

hipStream_t cur_stream(const at::Tensor& t) {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(
             static_cast<c10::DeviceIndex>(t.get_device()))
      .stream();
}

void fused_sgd(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
               std::vector<at::Tensor> ms, double lr, double momentum,
               double dampening, double wd, bool nesterov, bool maximize) {
  if (ps.empty()) return;
  auto plan = build_chunks(ps, gs, momentum != 0 ? ms : std::vector<at::Tensor>{},
                           {}, nullptr);
  // momentum==0 passes ms pointers as null via empty list; kernel ignores m.
  if (momentum != 0) {
    TORCH_CHECK(ms.size() == ps.size(), "momentum buffers missing");
  }
  launch_fused_sgd(reinterpret_cast<const OptChunk*>(plan.dev.data_ptr()),
                   plan.nchunks, dtype_code(ps[0]), static_cast<float>(lr),
                   static_cast<float>(momentum), static_cast<float>(dampening),
                   static_cast<float>(wd), nesterov, maximize, cur_stream(ps[0]));
}

void fused_adam(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                double lr, double beta1, double beta2, double eps, double wd,
                int64_t step, bool adamw) {
  if (ps.empty()) return;
  auto plan = build_chunks(ps, gs, ms, vs, nullptr);
  float bc1 = 1.0f - std::pow(static_cast<float>(beta1), static_cast<float>(step));
  float bc2_sqrt =
      std::sqrt(1.0f - std::pow(static_cast<float>(beta2), static_cast<float>(step)));
  launch_fused_adam(reinterpret_cast<const OptChunk*>(plan.dev.data_ptr()),
                    plan.nchunks, dtype_code(ps[0]), static_cast<float>(lr),
                    static_cast<float>(beta1), static_cast<float>(beta2),
                    static_cast<float>(eps), static_cast<float>(wd), bc1,
                    bc2_sqrt, adamw, cur_stream(ps[0]));
}

void fused_lamb(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                double lr, double beta1, double beta2, double eps, double wd,
                int64_t step) {
  if (ps.empty()) return;
  int64_t total = 0;
  for (auto& p : ps) total += p.numel();
  at::Tensor u = at::empty({total}, ps[0].options().dtype(at::kFloat));
  at::Tensor norms = at::zeros({2 * static_cast<int64_t>(ps.size())},
                               ps[0].options().dtype(at::kFloat));
  auto plan = build_chunks(ps, gs, ms, vs, static_cast<float*>(u.data_ptr()));
  float bc1 = 1.0f - std::pow(static_cast<float>(beta1), static_cast<float>(step));
  float bc2_sqrt =
      std::sqrt(1.0f - std::pow(static_cast<float>(beta2), static_cast<float>(step)));
  auto* descs = reinterpret_cast<const OptChunk*>(plan.dev.data_ptr());
  auto stream = cur_stream(ps[0]);
  launch_fused_lamb_phase1(descs, plan.nchunks, dtype_code(ps[0]),
                           static_cast<float>(beta1), static_cast<float>(beta2),
                           static_cast<float>(eps), static_cast<float>(wd), bc1,
                           bc2_sqrt, static_cast<float*>(norms.data_ptr()),
                           stream);
  launch_fused_lamb_phase2(descs, plan.nchunks, dtype_code(ps[0]),
                           static_cast<float>(lr),
                           static_cast<const float*>(norms.data_ptr()), stream);
}

}  // namespace

void register_optim(pybind11::module_& m) {
  m.def("fused_sgd", &fused_sgd, pybind11::arg("params"), pybind11::arg("grads"),
        pybind11::arg("momentum_buffers"), pybind11::arg("lr"),
        pybind11::arg("momentum"), pybind11::arg("dampening"),
        pybind11::arg("weight_decay"), pybind11::arg("nesterov"),
        pybind11::arg("maximize"));
  m.def("fused_adam", &fused_adam, pybind11::arg("params"), pybind11::arg("grads"),
        pybind11::arg("exp_avgs"), pybind11::arg("exp_avg_sqs"),
        pybind11::arg("lr"), pybind11::arg("beta1"), pybind11::arg("beta2"),
        pybind11::arg("eps"), pybind11::arg("weight_decay"), pybind11::arg("step"),
        pybind11::arg("adamw"));
  m.def("fused_lamb", &fused_lamb, pybind11::arg("params"), pybind11::arg("grads"),
        pybind11::arg("exp_avgs"), pybind11::arg("exp_avg_sqs"),
        pybind11::arg("lr"), pybind11::arg("beta1"), pybind11::arg("beta2"),
        pybind11::arg("eps"), pybind11::arg("weight_decay"), pybind11::arg("step"));
}

}  // namespace axon
