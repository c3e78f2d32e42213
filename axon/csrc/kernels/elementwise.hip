// Fused elementwise kernels for the Llama path (CDNA4): rotary position
// embedding and SwiGLU. Pure bandwidth ops — the win is kernel-count and
// pass-count reduction vs eager (rope on strided 0::2/1::2 views costs
// several half-efficiency kernels per call; silu+mul is two passes).
//
// rope: x is [B, S, H, D] contiguous (rope applied BEFORE the attention
// transpose so the stream stays coalesced); cos/sin are [S, D/2] in the
// compute dtype. Interleaved pairs: y[2i] = x[2i]*c - x[2i+1]*s,
// y[2i+1] = x[2i+1]*c + x[2i]*s. The backward is the transposed rotation
// (-s) — same kernel, sign flag.
//
// swiglu: z = silu(a) * g in one pass; backward recomputes silu(a) in
// registers: da = dz*g*sig*(1 + a*(1-sig)), dg = dz*a*sig.
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include <algorithm>

#include "kernels.h"

namespace axon {

namespace {

constexpr int kThreads = 256;
constexpr int kHalfId = 5, kFloatId = 6, kBF16Id = 15;

template <typename T>
__device__ __forceinline__ float ef_of(T x);
template <>
__device__ __forceinline__ float ef_of(float x) { return x; }
template <>
__device__ __forceinline__ float ef_of(__half x) { return __half2float(x); }
template <>
__device__ __forceinline__ float ef_of(__hip_bfloat16 x) { return __bfloat162float(x); }

template <typename T>
__device__ __forceinline__ T et_of(float x);
template <>
__device__ __forceinline__ float et_of(float x) { return x; }
template <>
__device__ __forceinline__ __half et_of(float x) { return __float2half(x); }
template <>
__device__ __forceinline__ __hip_bfloat16 et_of(float x) { return __float2bfloat16(x); }

template <typename T, int V>
struct alignas(sizeof(T) * V) EVec {
  T v[V];
};

// ------------------------------------------------------------------- rope
// One thread handles VP interleaved pairs (2*VP consecutive elements), all
// within one head row (VP divides D/2), so one (s, d0) lookup covers them.

template <typename T, int VP>
__global__ void rope_kernel(const T* __restrict__ x, T* __restrict__ y,
                            const T* __restrict__ cs, const T* __restrict__ sn,
                            int64_t nvec, int half_d, int heads, int S,
                            float sign) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int64_t p0 = i * VP;                     // first pair index
    int d0 = static_cast<int>(p0 % half_d);  // pair pos within head
    int64_t hrow = p0 / half_d;              // b*S*H + s*H + h
    int s = static_cast<int>((hrow / heads) % S);
    EVec<T, 2 * VP> xv = reinterpret_cast<const EVec<T, 2 * VP>*>(x)[i];
    EVec<T, VP> cv = *reinterpret_cast<const EVec<T, VP>*>(cs + static_cast<int64_t>(s) * half_d + d0);
    EVec<T, VP> sv = *reinterpret_cast<const EVec<T, VP>*>(sn + static_cast<int64_t>(s) * half_d + d0);
    EVec<T, 2 * VP> yv;
#pragma unroll
    for (int k = 0; k < VP; ++k) {
      float x1 = ef_of(xv.v[2 * k]), x2 = ef_of(xv.v[2 * k + 1]);
      float c = ef_of(cv.v[k]), ss = sign * ef_of(sv.v[k]);
      yv.v[2 * k] = et_of<T>(x1 * c - x2 * ss);
      yv.v[2 * k + 1] = et_of<T>(x2 * c + x1 * ss);
    }
    reinterpret_cast<EVec<T, 2 * VP>*>(y)[i] = yv;
  }
}

// ----------------------------------------------------------------- swiglu

template <typename T, int V>
__global__ void swiglu_fwd_kernel(const T* __restrict__ a,
                                  const T* __restrict__ g,
                                  T* __restrict__ z, int64_t nvec) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    EVec<T, V> av = reinterpret_cast<const EVec<T, V>*>(a)[i];
    EVec<T, V> gv = reinterpret_cast<const EVec<T, V>*>(g)[i];
    EVec<T, V> zv;
#pragma unroll
    for (int k = 0; k < V; ++k) {
      float af = ef_of(av.v[k]);
      float sig = 1.f / (1.f + __expf(-af));
      zv.v[k] = et_of<T>(af * sig * ef_of(gv.v[k]));
    }
    reinterpret_cast<EVec<T, V>*>(z)[i] = zv;
  }
}

template <typename T, int V>
__global__ void swiglu_bwd_kernel(const T* __restrict__ dz,
                                  const T* __restrict__ a,
                                  const T* __restrict__ g,
                                  T* __restrict__ da, T* __restrict__ dg,
                                  int64_t nvec) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    EVec<T, V> dv = reinterpret_cast<const EVec<T, V>*>(dz)[i];
    EVec<T, V> av = reinterpret_cast<const EVec<T, V>*>(a)[i];
    EVec<T, V> gv = reinterpret_cast<const EVec<T, V>*>(g)[i];
    EVec<T, V> dav, dgv;
#pragma unroll
    for (int k = 0; k < V; ++k) {
      float d = ef_of(dv.v[k]), af = ef_of(av.v[k]), gf = ef_of(gv.v[k]);
      float sig = 1.f / (1.f + __expf(-af));
      dav.v[k] = et_of<T>(d * gf * sig * (1.f + af * (1.f - sig)));
      dgv.v[k] = et_of<T>(d * af * sig);
    }
    reinterpret_cast<EVec<T, V>*>(da)[i] = dav;
    reinterpret_cast<EVec<T, V>*>(dg)[i] = dgv;
  }
}

int ew_blocks(int64_t nvec) {
  return static_cast<int>(std::min<int64_t>(
      (nvec + kThreads - 1) / kThreads, 16384));
}

template <typename T>
void rope_t(const void* x, void* y, const void* cs, const void* sn, int64_t n,
            int half_d, int heads, int S, float sign, hipStream_t stream) {
  int vp = half_d % 4 == 0 ? 4 : half_d % 2 == 0 ? 2 : 1;
  int64_t nvec = n / (2 * vp);
  switch (vp) {
    case 4: hipLaunchKernelGGL((rope_kernel<T, 4>), ew_blocks(nvec), kThreads, 0, stream, static_cast<const T*>(x), static_cast<T*>(y), static_cast<const T*>(cs), static_cast<const T*>(sn), nvec, half_d, heads, S, sign); break;
    case 2: hipLaunchKernelGGL((rope_kernel<T, 2>), ew_blocks(nvec), kThreads, 0, stream, static_cast<const T*>(x), static_cast<T*>(y), static_cast<const T*>(cs), static_cast<const T*>(sn), nvec, half_d, heads, S, sign); break;
    default: hipLaunchKernelGGL((rope_kernel<T, 1>), ew_blocks(nvec), kThreads, 0, stream, static_cast<const T*>(x), static_cast<T*>(y), static_cast<const T*>(cs), static_cast<const T*>(sn), nvec, half_d, heads, S, sign); break;
  }
}

template <typename T>
void swiglu_fwd_t(const void* a, const void* g, void* z, int64_t n,
                  hipStream_t stream) {
  constexpr int V = 8;
  if (n % V == 0) {
    hipLaunchKernelGGL((swiglu_fwd_kernel<T, V>), ew_blocks(n / V), kThreads, 0,
                       stream, static_cast<const T*>(a), static_cast<const T*>(g),
                       static_cast<T*>(z), n / V);
  } else {
    hipLaunchKernelGGL((swiglu_fwd_kernel<T, 1>), ew_blocks(n), kThreads, 0,
                       stream, static_cast<const T*>(a), static_cast<const T*>(g),
                       static_cast<T*>(z), n);
  }
}

template <typename T>
void swiglu_bwd_t(const void* dz, const void* a, const void* g, void* da,
                  void* dg, int64_t n, hipStream_t stream) {
  constexpr int V = 8;
  if (n % V == 0) {
    hipLaunchKernelGGL((swiglu_bwd_kernel<T, V>), ew_blocks(n / V), kThreads, 0,
                       stream, static_cast<const T*>(dz), static_cast<const T*>(a),
                       static_cast<const T*>(g), static_cast<T*>(da),
                       static_cast<T*>(dg), n / V);
  } else {
    hipLaunchKernelGGL((swiglu_bwd_kernel<T, 1>), ew_blocks(n), kThreads, 0,
                       stream, static_cast<const T*>(dz), static_cast<const T*>(a),
                       static_cast<const T*>(g), static_cast<T*>(da),
                       static_cast<T*>(dg), n);
  }
}

}  // namespace

void launch_rope(int dtype, const void* x, void* y, const void* cos_t,
                 const void* sin_t, int64_t numel, int half_d, int heads,
                 int seq, bool backward, hipStream_t stream) {
  float sign = backward ? -1.f : 1.f;
  switch (dtype) {
    case kFloatId: rope_t<float>(x, y, cos_t, sin_t, numel, half_d, heads, seq, sign, stream); break;
    case kHalfId: rope_t<__half>(x, y, cos_t, sin_t, numel, half_d, heads, seq, sign, stream); break;
    case kBF16Id: rope_t<__hip_bfloat16>(x, y, cos_t, sin_t, numel, half_d, heads, seq, sign, stream); break;
  }
}

void launch_swiglu_fwd(int dtype, const void* a, const void* g, void* z,
                       int64_t numel, hipStream_t stream) {
  switch (dtype) {
    case kFloatId: swiglu_fwd_t<float>(a, g, z, numel, stream); break;
    case kHalfId: swiglu_fwd_t<__half>(a, g, z, numel, stream); break;
    case kBF16Id: swiglu_fwd_t<__hip_bfloat16>(a, g, z, numel, stream); break;
  }
}

void launch_swiglu_bwd(int dtype, const void* dz, const void* a, const void* g,
                       void* da, void* dg, int64_t numel, hipStream_t stream) {
  switch (dtype) {
    case kFloatId: swiglu_bwd_t<float>(dz, a, g, da, dg, numel, stream); break;
    case kHalfId: swiglu_bwd_t<__half>(dz, a, g, da, dg, numel, stream); break;
    case kBF16Id: swiglu_bwd_t<__hip_bfloat16>(dz, a, g, da, dg, numel, stream); break;
  }
}

}  // namespace axon
