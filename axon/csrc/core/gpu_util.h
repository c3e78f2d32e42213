// Small HIP utilities shared by the torch binding (ready-event recording at
// enqueue time) and the RCCL backend (comm stream + completion events).
// Native PyTorch-ROCm stream plumbing — SURVEY.md §2.1 "GPU context &
// stream/event pool".
#pragma once

#include <hip/hip_runtime.h>

#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

namespace axon {

inline void hip_check(hipError_t e, const char* what) {
  if (e != hipSuccess)
    throw std::runtime_error(std::string("axon HIP error in ") + what + ": " +
                             hipGetErrorString(e));
}

// Reusable pool of timing-disabled events (hipEventCreate is ~µs but the hot
// path records one event per enqueued gradient — pool them).
class EventPool {
 public:
  static EventPool& instance() {
    static EventPool p;
    return p;
  }
  hipEvent_t get() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      if (!free_.empty()) {
        hipEvent_t e = free_.back();
        free_.pop_back();
        return e;
      }
    }
    hipEvent_t e;
    hip_check(hipEventCreateWithFlags(&e, hipEventDisableTiming), "hipEventCreate");
    return e;
  }
  void put(hipEvent_t e) {
    std::lock_guard<std::mutex> lk(mu_);
    free_.push_back(e);
  }

 private:
  std::mutex mu_;
  std::vector<hipEvent_t> free_;
};

}  // namespace axon
