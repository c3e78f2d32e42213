// K5: Adasum inner kernels — fused triple dot product
// (a.b, a.a, b.b) in one pass over the pair's segment, and the scaled-add
// apply. Double accumulation for parity with the fp64 CPU oracle
// (cpu_backend.cpp adasum_combine_cpu). CDNA4: wave64 shfl_down butterfly +
// LDS cross-wave tree, one fp64 atomicAdd per block per component.
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include "kernels.h"

namespace axon {

namespace {

constexpr int kThreads = 256;
constexpr int kHalf = 5, kFloat = 6, kDouble = 7, kBF16 = 15;

template <typename T>
__device__ __forceinline__ double d_of(T x);
template <>
__device__ __forceinline__ double d_of(float x) { return x; }
template <>
__device__ __forceinline__ double d_of(double x) { return x; }
template <>
__device__ __forceinline__ double d_of(__half x) { return __half2float(x); }
template <>
__device__ __forceinline__ double d_of(__hip_bfloat16 x) { return __bfloat162float(x); }

template <typename T>
__device__ __forceinline__ T t_of(double x);
template <>
__device__ __forceinline__ float t_of(double x) { return static_cast<float>(x); }
template <>
__device__ __forceinline__ double t_of(double x) { return x; }
template <>
__device__ __forceinline__ __half t_of(double x) {
  return __float2half(static_cast<float>(x));
}
template <>
__device__ __forceinline__ __hip_bfloat16 t_of(double x) {
  return __float2bfloat16(static_cast<float>(x));
}

__device__ __forceinline__ double wave_sum(double v) {
  // 64-lane wavefront butterfly (CDNA wave = 64, not 32).
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

template <typename T>
__global__ void adasum_dots_kernel(const T* __restrict__ kept,
                                   const T* __restrict__ recv, int64_t n,
                                   double* __restrict__ out3, int a_side) {
  double ab = 0, kk = 0, rr = 0;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * kThreads;
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * kThreads + threadIdx.x;
       i < n; i += stride) {
    double k = d_of(kept[i]), r = d_of(recv[i]);
    ab += k * r;
    kk += k * k;
    rr += r * r;
  }
  __shared__ double lds[3][kThreads / 64];
  ab = wave_sum(ab);
  kk = wave_sum(kk);
  rr = wave_sum(rr);
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  if (lane == 0) {
    lds[0][wave] = ab;
    lds[1][wave] = kk;
    lds[2][wave] = rr;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    double sab = 0, skk = 0, srr = 0;
#pragma unroll
    for (int w = 0; w < kThreads / 64; ++w) {
      sab += lds[0][w];
      skk += lds[1][w];
      srr += lds[2][w];
    }
    // Canonical (ab, aa, bb) order: 'a' is the lower subtree's vector.
    atomicAdd(&out3[0], sab);
    atomicAdd(&out3[a_side ? 1 : 2], skk);
    atomicAdd(&out3[a_side ? 2 : 1], srr);
  }
}

template <typename T>
__global__ void adasum_apply_kernel(T* __restrict__ kept,
                                    const T* __restrict__ recv, int64_t n,
                                    const double* __restrict__ dots3, int a_side) {
  const double ab = dots3[0], aa = dots3[1], bb = dots3[2];
  const double ca = aa > 0 ? 1.0 - ab / (2.0 * aa) : 1.0;
  const double cb = bb > 0 ? 1.0 - ab / (2.0 * bb) : 1.0;
  const double ck = a_side ? ca : cb;  // coefficient for my kept half
  const double cr = a_side ? cb : ca;  // coefficient for the received half
  const int64_t stride = static_cast<int64_t>(gridDim.x) * kThreads;
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * kThreads + threadIdx.x;
       i < n; i += stride) {
    kept[i] = t_of<T>(ck * d_of(kept[i]) + cr * d_of(recv[i]));
  }
}

int grid_for(int64_t n) {
  int64_t b = (n + kThreads - 1) / kThreads;
  // >=2048 workgroups fills 256 CUs; cap so tiny tensors stay cheap.
  return static_cast<int>(b < 2048 ? (b ? b : 1) : 2048);
}

__global__ void add3_kernel(double* __restrict__ mine,
                            const double* __restrict__ peer) {
  if (threadIdx.x < 3) mine[threadIdx.x] += peer[threadIdx.x];
}

}  // namespace

// mine[0..2] += peer[0..2] — merges canonical partial-dot triples during the
// per-level butterfly reduction (rccl_backend.cpp adasum path).
void launch_add3(double* mine, const double* peer, hipStream_t s) {
  hipLaunchKernelGGL(add3_kernel, dim3(1), dim3(64), 0, s, mine, peer);
}

void launch_adasum_dots(const void* kept, const void* recv, int64_t n,
                        int dtype, double* out3, bool a_side, hipStream_t s) {
  dim3 grid(grid_for(n)), block(kThreads);
#define CASE(T, C)                                                             \
  if (dtype == C) {                                                            \
    hipLaunchKernelGGL((adasum_dots_kernel<T>), grid, block, 0, s,             \
                       static_cast<const T*>(kept), static_cast<const T*>(recv), \
                       n, out3, a_side ? 1 : 0);                               \
    return;                                                                    \
  }
  CASE(float, kFloat)
  CASE(double, kDouble)
  CASE(__half, kHalf)
  CASE(__hip_bfloat16, kBF16)
#undef CASE
  abort();
}

void launch_adasum_apply(void* kept, const void* recv, int64_t n, int dtype,
                         const double* dots3, bool a_side, hipStream_t s) {
  dim3 grid(grid_for(n)), block(kThreads);
#define CASE(T, C)                                                            \
  if (dtype == C) {                                                           \
    hipLaunchKernelGGL((adasum_apply_kernel<T>), grid, block, 0, s,           \
                       static_cast<T*>(kept), static_cast<const T*>(recv), n, \
                       dots3, a_side ? 1 : 0);                                \
    return;                                                                   \
  }
  CASE(float, kFloat)
  CASE(double, kDouble)
  CASE(__half, kHalf)
  CASE(__hip_bfloat16, kBF16)
#undef CASE
  abort();
}

}  // namespace axon
