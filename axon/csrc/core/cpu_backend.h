// CPU collective backend over the unix-socket control plane (star topology
// through rank 0). This is the engine's CPU-tensor data plane — the analog
// of the reference's MPI/Gloo CPU ops (SURVEY.md §2.1 "MPI ops / Gloo ops /
// CPU ops") with zero external backends: the MNIST/CPU config
// (BASELINE.json:7) and all multi-process CPU tests run through this.
// Throughput is irrelevant here (correctness config); the GPU data plane is
// rccl_backend.*.
#pragma once

#include "controller.h"

namespace axon {

class CpuBackend {
 public:
  CpuBackend(int rank, int size, ControlPlane* cp)
      : rank_(rank), size_(size), cp_(cp) {}

  // Execute one response synchronously. Entries are this rank's matching
  // TensorEntry list (already zero-filled stand-ins for joined ranks).
  Status execute(const Response& r, std::vector<TensorEntry>& entries);

 private:
  Status allreduce(const Response& r, std::vector<TensorEntry>& entries);
  Status adasum(const Response& r, std::vector<TensorEntry>& entries);
  Status broadcast(const Response& r, std::vector<TensorEntry>& entries);
  Status allgather(const Response& r, std::vector<TensorEntry>& entries);
  Status alltoall(const Response& r, std::vector<TensorEntry>& entries);
  Status reducescatter(const Response& r, std::vector<TensorEntry>& entries);

  void send_tensor(int peer, const at::Tensor& t);
  at::Tensor recv_tensor(int peer, at::ScalarType dtype, int64_t numel);
  at::Tensor recv_tensor_any(int peer, at::ScalarType dtype);

  int rank_, size_;
  ControlPlane* cp_;
};

// Shared helpers (also used by tests/oracles).
// Adasum pairwise combine: a' = (1 - a.b/(2|a|^2)) a + (1 - a.b/(2|b|^2)) b,
// computed in fp64 on CPU (SURVEY.md §2.1 "Adasum ops").
at::Tensor adasum_combine_cpu(const at::Tensor& a, const at::Tensor& b);

}  // namespace axon
