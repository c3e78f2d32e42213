// pybind11 surface of the axon core — the analog of the reference's torch
// binding (horovod/torch/mpi_ops_v2.cc + handle_manager, SURVEY.md §2.1
// "Torch C++ binding"): converts at::Tensor args into core TensorEntry ops,
// returns integer handles, synchronize(handle) blocks GIL-free.
#include <torch/extension.h>

#include <condition_variable>
#include <mutex>
#include <unordered_map>

#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "core/gpu_util.h"
#include "core/message.h"
#include "core/operations.h"
#include "core/parameter_manager.h"

namespace axon {
namespace {

struct HandleState {
  std::mutex mu;
  std::condition_variable cv;
  bool done = false;
  Status status;
};

class HandleManager {
 public:
  static HandleManager& instance() {
    static HandleManager m;
    return m;
  }

  std::pair<uint64_t, StatusCallback> create() {
    auto st = std::make_shared<HandleState>();
    uint64_t id;
    {
      std::lock_guard<std::mutex> lk(mu_);
      id = next_++;
      handles_[id] = st;
    }
    StatusCallback cb = [st](const Status& s) {
      std::lock_guard<std::mutex> lk(st->mu);
      st->status = s;
      st->done = true;
      st->cv.notify_all();
    };
    return {id, cb};
  }

  std::shared_ptr<HandleState> get(uint64_t id) {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = handles_.find(id);
    return it == handles_.end() ? nullptr : it->second;
  }

  void erase(uint64_t id) {
    std::lock_guard<std::mutex> lk(mu_);
    handles_.erase(id);
  }

 private:
  std::mutex mu_;
  uint64_t next_ = 1;
  std::unordered_map<uint64_t, std::shared_ptr<HandleState>> handles_;
};

// Record a ready event on the producer (current torch) stream so the comm
// stream can fence against it (SURVEY.md §7 "Overlap vs. caching allocator").
uint64_t maybe_record_ready(const at::Tensor& t) {
  if (!t.defined() || !t.is_cuda()) return 0;
  hipEvent_t ev = EventPool::instance().get();
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(
      static_cast<c10::DeviceIndex>(t.get_device()));
  hip_check(hipEventRecord(ev, stream.stream()), "hipEventRecord(ready)");
  return reinterpret_cast<uint64_t>(ev);
}

int32_t dev_index_of(const at::Tensor& t) {
  return t.is_cuda() ? static_cast<int32_t>(t.get_device()) : -1;
}

uint64_t submit(TensorEntry e) {
  auto [id, cb] = HandleManager::instance().create();
  e.handle = id;
  e.callback = std::move(cb);
  try {
    enqueue(std::move(e));
  } catch (...) {
    HandleManager::instance().erase(id);
    throw;
  }
  return id;
}

uint64_t do_allreduce(at::Tensor in, at::Tensor out, const std::string& name,
                      int red, double prescale, double postscale,
                      int wire_dtype, int process_set) {
  TensorEntry e;
  e.name = name;
  e.op = OpType::ALLREDUCE;
  e.red = static_cast<ReduceOp>(red);
  e.in = in;
  e.out = out;
  e.prescale = prescale;
  e.postscale = postscale;
  e.wire_dtype = wire_dtype < 0 ? at::ScalarType::Undefined
                                : static_cast<at::ScalarType>(wire_dtype);
  e.process_set = process_set;
  e.device = dev_index_of(in);
  e.ready_event = maybe_record_ready(in);
  return submit(std::move(e));
}

uint64_t do_broadcast(at::Tensor in, at::Tensor out, int root,
                      const std::string& name, int process_set) {
  TensorEntry e;
  e.name = name;
  e.op = OpType::BROADCAST;
  e.in = in;
  e.out = out;
  e.root_rank = root;
  e.process_set = process_set;
  e.device = dev_index_of(in);
  e.ready_event = maybe_record_ready(in);
  return submit(std::move(e));
}

uint64_t do_allgather(at::Tensor in, at::Tensor out, const std::string& name,
                      int process_set) {
  TensorEntry e;
  e.name = name;
  e.op = OpType::ALLGATHER;
  e.in = in;
  e.out = out;
  e.process_set = process_set;
  e.device = dev_index_of(in);
  e.ready_event = maybe_record_ready(in);
  return submit(std::move(e));
}

uint64_t do_alltoall(at::Tensor in, at::Tensor out, std::vector<int64_t> splits,
                     at::Tensor received_splits, const std::string& name,
                     int process_set) {
  TensorEntry e;
  e.name = name;
  e.op = OpType::ALLTOALL;
  e.in = in;
  e.out = out;
  e.splits = std::move(splits);
  e.received_splits = received_splits;
  e.process_set = process_set;
  e.device = dev_index_of(in);
  e.ready_event = maybe_record_ready(in);
  return submit(std::move(e));
}

uint64_t do_reducescatter(at::Tensor in, at::Tensor out, const std::string& name,
                          int red, double prescale, double postscale,
                          int process_set) {
  TensorEntry e;
  e.name = name;
  e.op = OpType::REDUCESCATTER;
  e.red = static_cast<ReduceOp>(red);
  e.in = in;
  e.out = out;
  e.prescale = prescale;
  e.postscale = postscale;
  e.process_set = process_set;
  e.device = dev_index_of(in);
  e.ready_event = maybe_record_ready(in);
  return submit(std::move(e));
}

uint64_t do_join(int device, at::Tensor out) {
  TensorEntry e;
  e.name = "__axon.join__";
  e.op = OpType::JOIN;
  e.out = out;  // CPU int64 scalar: receives last_joined rank
  e.device = device;
  return submit(std::move(e));
}

uint64_t do_barrier(const std::string& name) {
  TensorEntry e;
  e.name = name;
  e.op = OpType::BARRIER;
  return submit(std::move(e));
}

void do_synchronize(uint64_t handle) {
  auto st = HandleManager::instance().get(handle);
  if (!st) throw std::runtime_error("axon: unknown handle");
  {
    py::gil_scoped_release release;
    std::unique_lock<std::mutex> lk(st->mu);
    st->cv.wait(lk, [&] { return st->done; });
  }
  HandleManager::instance().erase(handle);
  if (!st->status.ok()) throw std::runtime_error(st->status.reason);
}

bool do_poll(uint64_t handle) {
  auto st = HandleManager::instance().get(handle);
  if (!st) throw std::runtime_error("axon: unknown handle");
  std::lock_guard<std::mutex> lk(st->mu);
  return st->done;
}

}  // namespace

void register_optim(pybind11::module_& m);  // optim_ops.cpp
void register_bn(pybind11::module_& m);     // bn_ops.cpp

}  // namespace axon

PYBIND11_MODULE(_core, m) {
  axon::register_optim(m);
  axon::register_bn(m);
  m.doc() = "axon core: MI355X-native ring-allreduce data-parallel engine";
  m.def("init", &axon::initialize, py::arg("rank"), py::arg("size"),
        py::arg("local_rank"), py::arg("control_path"),
        py::call_guard<py::gil_scoped_release>());
  m.def("shutdown", &axon::shutdown, py::call_guard<py::gil_scoped_release>());
  m.def("is_initialized", &axon::is_initialized);
  m.def("rank", &axon::rank);
  m.def("size", &axon::size);
  m.def("local_rank", &axon::local_rank);
  m.def("fatal_error", &axon::fatal_error);

  m.def("allreduce_async", &axon::do_allreduce, py::arg("in"), py::arg("out"),
        py::arg("name"), py::arg("red"), py::arg("prescale") = 1.0,
        py::arg("postscale") = 1.0, py::arg("wire_dtype") = -1,
        py::arg("process_set") = 0);
  m.def("broadcast_async", &axon::do_broadcast, py::arg("in"), py::arg("out"),
        py::arg("root"), py::arg("name"), py::arg("process_set") = 0);
  m.def("allgather_async", &axon::do_allgather, py::arg("in"), py::arg("out"),
        py::arg("name"), py::arg("process_set") = 0);
  m.def("alltoall_async", &axon::do_alltoall, py::arg("in"), py::arg("out"),
        py::arg("splits"), py::arg("received_splits"), py::arg("name"),
        py::arg("process_set") = 0);
  m.def("reducescatter_async", &axon::do_reducescatter, py::arg("in"),
        py::arg("out"), py::arg("name"), py::arg("red"),
        py::arg("prescale") = 1.0, py::arg("postscale") = 1.0,
        py::arg("process_set") = 0);
  m.def("autotune_step_mark", &axon::pm_step_mark);
  m.def("_autotune_bo_propose",
        [](const std::vector<std::tuple<int64_t, double, double>>& obs)
            -> pybind11::object {
          // (threshold_bytes, cycle_ms, score) triples -> EI-max proposal
          // (deterministic; exposed for the unit test of the GP/EI math).
          std::vector<axon::ParameterManager::Sample> v;
          for (auto& t : obs)
            v.push_back({std::get<0>(t), std::get<1>(t), std::get<2>(t)});
          int64_t thr;
          double cyc;
          if (!axon::autotune_bo_propose(v, &thr, &cyc))
            return pybind11::none();
          return pybind11::make_tuple(thr, cyc);
        });
  m.def("start_timeline", &axon::timeline_start, pybind11::arg("path"));
  m.def("stop_timeline", &axon::timeline_stop);
  m.def("ps_register", &axon::ps_register, py::arg("ranks"));
  m.def("ps_remove", &axon::ps_remove, py::arg("id"));
  m.def("ps_count", &axon::ps_count);
  m.def("ps_ranks", [](int id) {
    auto psi = axon::ps_get(id);
    if (!psi.valid) throw std::runtime_error("axon: unknown/removed process set");
    return psi.ranks;
  }, py::arg("id"));
  m.def("join_async", &axon::do_join, py::arg("device"), py::arg("out"));
  m.def("barrier_async", &axon::do_barrier, py::arg("name"));
  m.def("synchronize", &axon::do_synchronize, py::arg("handle"));
  m.def("poll", &axon::do_poll, py::arg("handle"));
}
