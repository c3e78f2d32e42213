// K6: fused multi-tensor optimizer steps — SGD / Adam(W) /
// LAMB — one kernel launch per parameter group instead of per-tensor torch
// kernels. CDNA4 (gfx950): wave64, 256-thread blocks, 16-byte vector IO;
// these are bandwidth-bound (read p,g,m[,v] + write p,m[,v]), so wide IO
// and a single fused pass are the levers. The LAMB per-tensor norm
// reduction uses an f32 MFMA (v_mfma_f32_16x16x4_f32) as a 64->16 lane
// tree step (dot-product-shaped work on the matrix core) + LDS for the
// cross-wave tree.
//
// State-dtype contract: state tensors have the SAME dtype
// and meaning as stock torch optimizers (momentum_buffer / exp_avg /
// exp_avg_sq in param dtype), so checkpoints interchange with vanilla
// PyTorch; arithmetic is fp32 internally.
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include "kernels.h"
#include "optimizers.h"

namespace axon {

namespace {

constexpr int kThreads = 256;
constexpr int kHalf = 5, kFloat = 6, kBF16 = 15;

template <typename T>
__device__ __forceinline__ float f_of(T x);
template <>
__device__ __forceinline__ float f_of(float x) { return x; }
template <>
__device__ __forceinline__ float f_of(__half x) { return __half2float(x); }
template <>
__device__ __forceinline__ float f_of(__hip_bfloat16 x) { return __bfloat162float(x); }

template <typename T>
__device__ __forceinline__ T t_of(float x);
template <>
__device__ __forceinline__ float t_of(float x) { return x; }
template <>
__device__ __forceinline__ __half t_of(float x) { return __float2half(x); }
template <>
__device__ __forceinline__ __hip_bfloat16 t_of(float x) { return __float2bfloat16(x); }

template <typename T, int V>
struct alignas(sizeof(T) * V) TVec {
  T v[V];
};

// ------------------------------------------------------------------ SGD

template <typename T>
__global__ void sgd_kernel(const OptChunk* __restrict__ chunks, float lr,
                           float momentum, float dampening, float wd,
                           int nesterov, int maximize) {
  const OptChunk c = chunks[blockIdx.x];
  T* __restrict__ p = static_cast<T*>(c.p);
  const T* __restrict__ g = static_cast<const T*>(c.g);
  T* __restrict__ m = static_cast<T*>(c.m);
  constexpr int V = 8;
  const bool vec = c.n % V == 0 &&
      (reinterpret_cast<uintptr_t>(p) | reinterpret_cast<uintptr_t>(g) |
       reinterpret_cast<uintptr_t>(m)) % (sizeof(T) * V) == 0;
  auto update = [&](float pv, float gv, float mv, float* pm) {
    float d = maximize ? -gv : gv;
    d += wd * pv;
    if (momentum != 0.f) {
      mv = momentum * mv + (1.f - dampening) * d;
      *pm = mv;
      d = nesterov ? d + momentum * mv : mv;
    }
    return pv - lr * d;
  };
  if (vec) {
    auto* pv8 = reinterpret_cast<TVec<T, V>*>(p);
    auto* gv8 = reinterpret_cast<const TVec<T, V>*>(g);
    auto* mv8 = reinterpret_cast<TVec<T, V>*>(m);
    for (int64_t i = threadIdx.x; i < c.n / V; i += kThreads) {
      TVec<T, V> pp = pv8[i], gg = gv8[i], mm;
      if (momentum != 0.f) mm = mv8[i];
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float pm = momentum != 0.f ? f_of(mm.v[k]) : 0.f;
        float out = update(f_of(pp.v[k]), f_of(gg.v[k]), pm, &pm);
        pp.v[k] = t_of<T>(out);
        if (momentum != 0.f) mm.v[k] = t_of<T>(pm);
      }
      pv8[i] = pp;
      if (momentum != 0.f) mv8[i] = mm;
    }
  } else {
    for (int64_t i = threadIdx.x; i < c.n; i += kThreads) {
      float pm = momentum != 0.f ? f_of(m[i]) : 0.f;
      float out = update(f_of(p[i]), f_of(g[i]), pm, &pm);
      p[i] = t_of<T>(out);
      if (momentum != 0.f) m[i] = t_of<T>(pm);
    }
  }
}

// ---------------------------------------------------------------- Adam(W)

template <typename T>
__global__ void adam_kernel(const OptChunk* __restrict__ chunks, float lr,
                            float beta1, float beta2, float eps, float wd,
                            float bc1, float bc2_sqrt, int adamw) {
  const OptChunk c = chunks[blockIdx.x];
  T* __restrict__ p = static_cast<T*>(c.p);
  const T* __restrict__ g = static_cast<const T*>(c.g);
  T* __restrict__ m = static_cast<T*>(c.m);
  T* __restrict__ v = static_cast<T*>(c.v);
  constexpr int V = 8;
  const bool vec = c.n % V == 0 &&
      (reinterpret_cast<uintptr_t>(p) | reinterpret_cast<uintptr_t>(g) |
       reinterpret_cast<uintptr_t>(m) | reinterpret_cast<uintptr_t>(v)) %
              (sizeof(T) * V) == 0;
  auto update = [&](float pv, float gv, float* mv, float* vv) {
    if (adamw) {
      pv *= (1.f - lr * wd);
    } else {
      gv += wd * pv;
    }
    *mv = beta1 * *mv + (1.f - beta1) * gv;
    *vv = beta2 * *vv + (1.f - beta2) * gv * gv;
    float mh = *mv / bc1;
    float vh = sqrtf(*vv) / bc2_sqrt;
    return pv - lr * mh / (vh + eps);
  };
  if (vec) {
    auto* pv8 = reinterpret_cast<TVec<T, V>*>(p);
    auto* gv8 = reinterpret_cast<const TVec<T, V>*>(g);
    auto* mv8 = reinterpret_cast<TVec<T, V>*>(m);
    auto* vv8 = reinterpret_cast<TVec<T, V>*>(v);
    for (int64_t i = threadIdx.x; i < c.n / V; i += kThreads) {
      TVec<T, V> pp = pv8[i], gg = gv8[i], mm = mv8[i], vv_ = vv8[i];
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float mf = f_of(mm.v[k]), vf = f_of(vv_.v[k]);
        float out = update(f_of(pp.v[k]), f_of(gg.v[k]), &mf, &vf);
        pp.v[k] = t_of<T>(out);
        mm.v[k] = t_of<T>(mf);
        vv_.v[k] = t_of<T>(vf);
      }
      pv8[i] = pp;
      mv8[i] = mm;
      vv8[i] = vv_;
    }
  } else {
    for (int64_t i = threadIdx.x; i < c.n; i += kThreads) {
      float mf = f_of(m[i]), vf = f_of(v[i]);
      float out = update(f_of(p[i]), f_of(g[i]), &mf, &vf);
      p[i] = t_of<T>(out);
      m[i] = t_of<T>(mf);
      v[i] = t_of<T>(vf);
    }
  }
}

// ------------------------------------------------------------------ LAMB
//
// Phase 1: compute the Adam-style update u (with decoupled weight decay),
// store u to scratch, and accumulate per-TENSOR ||p||^2 and ||u||^2.
// Phase 2: p -= lr * trust_ratio(t) * u with trust_ratio = ||p||/||u||
// (1.0 when either norm is zero). You et al., "Large Batch Optimization
// for Deep Learning" (LAMB).

// 64-lane sum via one f32 MFMA (v_mfma_f32_16x16x4_f32): lane l holds
// A[l&15][l>>4] and B[l>>4][l&15]; with B=1 the D row sums A over k, i.e.
// lane groups {i, i+16, i+32, i+48} — a 64->16 reduction on the matrix
// core — then 4 DS swizzles finish 16->1.
__device__ __forceinline__ float wave_sum_mfma(float x) {
  using f32x4 = __attribute__((ext_vector_type(4))) float;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x4f32(x, 1.0f, acc, 0, 0, 0);
  // Lane l now holds D[row=(l>>4)*4+reg][col=l&15]; every row of D equals
  // the same 16 column sums, so take reg 0 and butterfly the 16 columns.
  float s = acc[0];
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) s += __shfl_xor(s, off, 64);
  return s;  // every lane: total across the wave
}

template <typename T>
__global__ void lamb_phase1_kernel(const OptChunk* __restrict__ chunks,
                                   float beta1, float beta2, float eps,
                                   float wd, float bc1, float bc2_sqrt,
                                   float* __restrict__ norms /* [ntensor][2] */) {
  const OptChunk c = chunks[blockIdx.x];
  T* __restrict__ p = static_cast<T*>(c.p);
  const T* __restrict__ g = static_cast<const T*>(c.g);
  T* __restrict__ m = static_cast<T*>(c.m);
  T* __restrict__ v = static_cast<T*>(c.v);
  float* __restrict__ u = static_cast<float*>(c.u);
  float pn = 0.f, un = 0.f;
  for (int64_t i = threadIdx.x; i < c.n; i += kThreads) {
    float pv = f_of(p[i]), gv = f_of(g[i]);
    float mf = beta1 * f_of(m[i]) + (1.f - beta1) * gv;
    float vf = beta2 * f_of(v[i]) + (1.f - beta2) * gv * gv;
    m[i] = t_of<T>(mf);
    v[i] = t_of<T>(vf);
    float mh = mf / bc1;
    float vh = sqrtf(vf) / bc2_sqrt;
    float uu = mh / (vh + eps) + wd * pv;
    u[i] = uu;
    pn += pv * pv;
    un += uu * uu;
  }
  __shared__ float lds[2][kThreads / 64];
  pn = wave_sum_mfma(pn);
  un = wave_sum_mfma(un);
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  if (lane == 0) {
    lds[0][wave] = pn;
    lds[1][wave] = un;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float sp = 0.f, su = 0.f;
#pragma unroll
    for (int w = 0; w < kThreads / 64; ++w) {
      sp += lds[0][w];
      su += lds[1][w];
    }
    atomicAdd(&norms[2 * c.tidx], sp);
    atomicAdd(&norms[2 * c.tidx + 1], su);
  }
}

template <typename T>
__global__ void lamb_phase2_kernel(const OptChunk* __restrict__ chunks, float lr,
                                   const float* __restrict__ norms) {
  const OptChunk c = chunks[blockIdx.x];
  T* __restrict__ p = static_cast<T*>(c.p);
  const float* __restrict__ u = static_cast<const float*>(c.u);
  const float pn = sqrtf(norms[2 * c.tidx]);
  const float un = sqrtf(norms[2 * c.tidx + 1]);
  const float ratio = (pn > 0.f && un > 0.f) ? pn / un : 1.0f;
  const float step = lr * ratio;
  for (int64_t i = threadIdx.x; i < c.n; i += kThreads) {
    p[i] = t_of<T>(f_of(p[i]) - step * u[i]);
  }
}

}  // namespace

#define DISPATCH(fn, ...)                                        \
  do {                                                           \
    if (dtype == kFloat) {                                       \
      hipLaunchKernelGGL((fn<float>), grid, block, 0, stream,    \
                         __VA_ARGS__);                           \
      return;                                                    \
    }                                                            \
    if (dtype == kBF16) {                                        \
      hipLaunchKernelGGL((fn<__hip_bfloat16>), grid, block, 0,   \
                         stream, __VA_ARGS__);                   \
      return;                                                    \
    }                                                            \
    if (dtype == kHalf) {                                        \
      hipLaunchKernelGGL((fn<__half>), grid, block, 0, stream,   \
                         __VA_ARGS__);                           \
      return;                                                    \
    }                                                            \
    abort();                                                     \
  } while (0)

void launch_fused_sgd(const OptChunk* chunks, int nchunks, int dtype, float lr,
                      float momentum, float dampening, float wd, bool nesterov,
                      bool maximize, hipStream_t stream) {
  if (nchunks <= 0) return;
  dim3 grid(nchunks), block(kThreads);
  DISPATCH(sgd_kernel, chunks, lr, momentum, dampening, wd, nesterov ? 1 : 0,
           maximize ? 1 : 0);
}

void launch_fused_adam(const OptChunk* chunks, int nchunks, int dtype, float lr,
                       float beta1, float beta2, float eps, float wd, float bc1,
                       float bc2_sqrt, bool adamw, hipStream_t stream) {
  if (nchunks <= 0) return;
  dim3 grid(nchunks), block(kThreads);
  DISPATCH(adam_kernel, chunks, lr, beta1, beta2, eps, wd, bc1, bc2_sqrt,
           adamw ? 1 : 0);
}

void launch_fused_lamb_phase1(const OptChunk* chunks, int nchunks, int dtype,
                              float beta1, float beta2, float eps, float wd,
                              float bc1, float bc2_sqrt, float* norms,
                              hipStream_t stream) {
  if (nchunks <= 0) return;
  dim3 grid(nchunks), block(kThreads);
  DISPATCH(lamb_phase1_kernel, chunks, beta1, beta2, eps, wd, bc1, bc2_sqrt,
           norms);
}

void launch_fused_lamb_phase2(const OptChunk* chunks, int nchunks, int dtype,
                              float lr, const float* norms, hipStream_t stream) {
  if (nchunks <= 0) return;
  dim3 grid(nchunks), block(kThreads);
  DISPATCH(lamb_phase2_kernel, chunks, lr, norms);
}

#undef DISPATCH

}  // namespace axon
