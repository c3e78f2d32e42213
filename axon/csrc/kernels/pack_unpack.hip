// K1/K2/K3/K4: batched fusion-buffer pack/unpack with the
// scale (K3) and fp16/bf16 wire compression casts (K4) fused into the same
// pass — one read + one write of each byte instead of separate memcpy +
// scale + cast kernels. CDNA4 (gfx950) notes: wave64, 256-thread blocks,
// 16-byte vector accesses (global_load/store_dwordx4), one workgroup per
// chunk so a full fusion buffer launches >=1024 workgroups across 256 CUs.
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include "kernels.h"

namespace axon {

namespace {

constexpr int kThreads = 256;

// at::ScalarType codes we handle (kept in sync with c10/core/ScalarType.h).
constexpr int kHalf = 5, kFloat = 6, kDouble = 7, kBF16 = 15;

template <typename T, int V>
struct alignas(sizeof(T) * V) Vec {
  T v[V];
};

__device__ __forceinline__ float to_acc(float x) { return x; }
__device__ __forceinline__ float to_acc(__half x) { return __half2float(x); }
__device__ __forceinline__ float to_acc(__hip_bfloat16 x) { return __bfloat162float(x); }
__device__ __forceinline__ double to_acc(double x) { return x; }

template <typename T>
__device__ __forceinline__ T from_acc(float x);
template <>
__device__ __forceinline__ float from_acc<float>(float x) { return x; }
template <>
__device__ __forceinline__ __half from_acc<__half>(float x) { return __float2half(x); }
template <>
__device__ __forceinline__ __hip_bfloat16 from_acc<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}

__global__ void copy_bytes_kernel(const ChunkDesc* __restrict__ descs) {
  const ChunkDesc d = descs[blockIdx.x];
  const char* __restrict__ s = static_cast<const char*>(d.src);
  char* __restrict__ dst = static_cast<char*>(d.dst);
  const int64_t n = d.n;
  if (((reinterpret_cast<uintptr_t>(s) | reinterpret_cast<uintptr_t>(dst)) & 15) == 0) {
    const int64_t nv = n >> 4;
    const uint4* __restrict__ sv = reinterpret_cast<const uint4*>(s);
    uint4* __restrict__ dv = reinterpret_cast<uint4*>(dst);
    for (int64_t i = threadIdx.x; i < nv; i += kThreads) dv[i] = sv[i];
    for (int64_t i = (nv << 4) + threadIdx.x; i < n; i += kThreads) dst[i] = s[i];
  } else {
    for (int64_t i = threadIdx.x; i < n; i += kThreads) dst[i] = s[i];
  }
}

// dst[i] = DstT(SrcT(src[i]) * scale); n = elements. 8 elements per thread
// per iteration through 16B-aligned vector paths where possible.
template <typename SrcT, typename DstT>
__global__ void xform_kernel(const ChunkDesc* __restrict__ descs, float scale) {
  const ChunkDesc d = descs[blockIdx.x];
  const SrcT* __restrict__ s = static_cast<const SrcT*>(d.src);
  DstT* __restrict__ dst = static_cast<DstT*>(d.dst);
  const int64_t n = d.n;
  constexpr int V = 8;
  const bool aligned =
      (reinterpret_cast<uintptr_t>(s) % (sizeof(SrcT) * V) == 0) &&
      (reinterpret_cast<uintptr_t>(dst) % (sizeof(DstT) * V) == 0);
  if (aligned) {
    const int64_t nv = n / V;
    const Vec<SrcT, V>* __restrict__ sv = reinterpret_cast<const Vec<SrcT, V>*>(s);
    Vec<DstT, V>* __restrict__ dv = reinterpret_cast<Vec<DstT, V>*>(dst);
    for (int64_t i = threadIdx.x; i < nv; i += kThreads) {
      Vec<SrcT, V> a = sv[i];
      Vec<DstT, V> b;
#pragma unroll
      for (int k = 0; k < V; ++k) b.v[k] = from_acc<DstT>(to_acc(a.v[k]) * scale);
      dv[i] = b;
    }
    for (int64_t i = nv * V + threadIdx.x; i < n; i += kThreads)
      dst[i] = from_acc<DstT>(to_acc(s[i]) * scale);
  } else {
    for (int64_t i = threadIdx.x; i < n; i += kThreads)
      dst[i] = from_acc<DstT>(to_acc(s[i]) * scale);
  }
}

// double needs a double accumulator (prescale precision).
__global__ void xform_kernel_f64(const ChunkDesc* __restrict__ descs, double scale) {
  const ChunkDesc d = descs[blockIdx.x];
  const double* __restrict__ s = static_cast<const double*>(d.src);
  double* __restrict__ dst = static_cast<double*>(d.dst);
  for (int64_t i = threadIdx.x; i < d.n; i += kThreads) dst[i] = s[i] * scale;
}

}  // namespace

void launch_copy_bytes(const ChunkDesc* descs, int nchunks, hipStream_t stream) {
  if (nchunks <= 0) return;
  hipLaunchKernelGGL(copy_bytes_kernel, dim3(nchunks), dim3(kThreads), 0, stream,
                     descs);
}

bool xform_supported(int s, int t) {
  auto fp = [](int x) { return x == kHalf || x == kFloat || x == kBF16; };
  return (fp(s) && fp(t)) || (s == kDouble && t == kDouble);
}

void launch_xform(const ChunkDesc* descs, int nchunks, int src_dtype,
                  int dst_dtype, float scale, hipStream_t stream) {
  if (nchunks <= 0) return;
  dim3 grid(nchunks), block(kThreads);
#define XF(ST, SC, DT, DC)                                                  \
  if (src_dtype == SC && dst_dtype == DC) {                                 \
    hipLaunchKernelGGL((xform_kernel<ST, DT>), grid, block, 0, stream, descs, \
                       scale);                                              \
    return;                                                                 \
  }
  XF(float, kFloat, float, kFloat)
  XF(float, kFloat, __half, kHalf)
  XF(__half, kHalf, float, kFloat)
  XF(float, kFloat, __hip_bfloat16, kBF16)
  XF(__hip_bfloat16, kBF16, float, kFloat)
  XF(__half, kHalf, __half, kHalf)
  XF(__hip_bfloat16, kBF16, __hip_bfloat16, kBF16)
  XF(__half, kHalf, __hip_bfloat16, kBF16)
  XF(__hip_bfloat16, kBF16, __half, kHalf)
#undef XF
  if (src_dtype == kDouble && dst_dtype == kDouble) {
    hipLaunchKernelGGL(xform_kernel_f64, grid, block, 0, stream, descs,
                       static_cast<double>(scale));
    return;
  }
  // Unsupported combination is a caller bug (backend routes through
  // copy_bytes for identity/no-scale integer cases).
  abort();
}

}  // namespace axon
