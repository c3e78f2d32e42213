// Fused multi-tensor optimizer kernel launchers (K6 — SURVEY.md §2.4).
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace axon {

// One contiguous chunk of one parameter tensor. Pointers pre-offset to the
// chunk start; n = elements; tidx = tensor index (LAMB norm slots);
// u = fp32 scratch for the LAMB update vector (pre-offset).
struct OptChunk {
  void* p;
  void* g;
  void* m;
  void* v;
  void* u;
  int64_t n;
  int64_t tidx;
};

void launch_fused_sgd(const OptChunk* chunks, int nchunks, int dtype, float lr,
                      float momentum, float dampening, float wd, bool nesterov,
                      bool maximize, hipStream_t stream);

void launch_fused_adam(const OptChunk* chunks, int nchunks, int dtype, float lr,
                       float beta1, float beta2, float eps, float wd, float bc1,
                       float bc2_sqrt, bool adamw, hipStream_t stream);

void launch_fused_lamb_phase1(const OptChunk* chunks, int nchunks, int dtype,
                              float beta1, float beta2, float eps, float wd,
                              float bc1, float bc2_sqrt, float* norms,
                              hipStream_t stream);
void launch_fused_lamb_phase2(const OptChunk* chunks, int nchunks, int dtype,
                              float lr, const float* norms, hipStream_t stream);

}  // namespace axon
