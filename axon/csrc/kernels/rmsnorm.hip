// Fused RMSNorm for CDNA4 — the Llama elementwise hot spot: the eager
// formula (cast→square→mean→rsqrt→mul→mul→cast) costs ~7 kernels and 4+
// passes over x per call; this is one kernel and ONE global read of x
// (row staged through LDS) per direction. rocprof context:
// profiles/README.md (llama: elementwise ≈12% of step time).
//
//   fwd: y = x * w * rsqrt(mean(x^2) + eps); saves per-row rstd (fp32).
//   bwd: dx = rstd*w*dy - x * rstd^3/H * sum(dy*w*x)
//        dw = sum_rows(dy * x * rstd)  (fp32 scratch, atomics per block)
//
// Layout: x is [rows, H], H innermost and contiguous. One block per row
// (fwd) / per row-batch (bwd); 256 threads, 16-byte vector IO, fp32
// accumulation, LDS tree for the cross-wave reduction.
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include <algorithm>
#include <type_traits>

#include "kernels.h"

namespace axon {

namespace {

constexpr int kThreads = 256;
constexpr int kWaves = 4;
constexpr int kHalfId = 5, kFloatId = 6, kBF16Id = 15;

template <typename T>
__device__ __forceinline__ float rf_of(T x);
template <>
__device__ __forceinline__ float rf_of(float x) { return x; }
template <>
__device__ __forceinline__ float rf_of(__half x) { return __half2float(x); }
template <>
__device__ __forceinline__ float rf_of(__hip_bfloat16 x) { return __bfloat162float(x); }

template <typename T>
__device__ __forceinline__ T rt_of(float x);
template <>
__device__ __forceinline__ float rt_of(float x) { return x; }
template <>
__device__ __forceinline__ __half rt_of(float x) { return __float2half(x); }
template <>
__device__ __forceinline__ __hip_bfloat16 rt_of(float x) { return __float2bfloat16(x); }

template <typename T, int V>
struct alignas(sizeof(T) * V) RVec {
  T v[V];
};

// Block-wide sum of one float per thread; returns the total to all threads.
// red[] must be __shared__ float[kWaves].
__device__ float block_sum(float v, float* red) {
  // wave-level tree via xor shuffles (wave64)
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, 64);
  const int lane = threadIdx.x % 64, wave = threadIdx.x / 64;
  if (lane == 0) red[wave] = v;
  __syncthreads();
  float total = red[0];
#pragma unroll
  for (int w = 1; w < kWaves; ++w) total += red[w];
  return total;
}

// ------------------------------------------------------------------ forward
// One WAVE per row, registers only (same structure as the backward below:
// the row lives in lane registers, the sum-of-squares reduces with 6
// xor-shuffles, no LDS and no barriers).

__device__ __forceinline__ float rms_wave_sum_f(float v) {
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, 64);
  return v;
}

template <typename T, int VC, int CH>
__global__ void rms_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                               T* __restrict__ y, float* __restrict__ rstd_out,
                               int64_t rows, int H, float eps,
                               int rows_per_unit) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int unit = blockIdx.x * kWaves + wave;

  RVec<T, VC> wv[CH];
#pragma unroll
  for (int j = 0; j < CH; ++j)
    wv[j] = reinterpret_cast<const RVec<T, VC>*>(w)[lane + 64 * j];

  const int64_t r0 = static_cast<int64_t>(unit) * rows_per_unit;
  const int64_t r1 = r0 + rows_per_unit > rows ? rows : r0 + rows_per_unit;

  for (int64_t r = r0; r < r1; ++r) {
    const RVec<T, VC>* xr = reinterpret_cast<const RVec<T, VC>*>(x + r * H);
    RVec<T, VC> xv[CH];
    float ss = 0.f;
#pragma unroll
    for (int j = 0; j < CH; ++j) {
      xv[j] = xr[lane + 64 * j];
#pragma unroll
      for (int k = 0; k < VC; ++k) {
        float f = rf_of(xv[j].v[k]);
        ss += f * f;
      }
    }
    const float rstd = rsqrtf(rms_wave_sum_f(ss) / H + eps);
    if (lane == 0 && rstd_out) rstd_out[r] = rstd;

    RVec<T, VC>* yr = reinterpret_cast<RVec<T, VC>*>(y + r * H);
#pragma unroll
    for (int j = 0; j < CH; ++j) {
      RVec<T, VC> yv;
#pragma unroll
      for (int k = 0; k < VC; ++k)
        yv.v[k] = rt_of<T>(rf_of(xv[j].v[k]) * rstd * rf_of(wv[j].v[k]));
      yr[lane + 64 * j] = yv;
    }
  }
}

// ----------------------------------------------------------------- backward
// One WAVE per row (no __syncthreads anywhere): the row's x/dy live in
// REGISTERS (lane l, chunk j owns columns (l + 64j)*VC ..+VC), the row
// dot reduces with 6 xor-shuffles, dx writes straight from the held
// registers — a block-per-row version with LDS staging and two block
// barriers per row measured ~7x off streaming SOL at H=1024. Per-wave dw
// partials land in a COLUMN-major [H][nunits] buffer (no atomics:
// same-address atomicAdd from ~2k blocks measured 40x off SOL); a
// per-column fold kernel produces dw. Host guarantees H % (64*VC) == 0
// and H/(64*VC) <= kMaxDwChunks.

constexpr int kMaxDwChunks = 8;  // bounds register arrays: H <= 16384 at VC=8

__device__ __forceinline__ float wave_sum(float v) {
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, 64);
  return v;
}

template <typename T, int VC, int CH>
__global__ void rms_bwd_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                               const T* __restrict__ w,
                               const float* __restrict__ rstd_in,
                               T* __restrict__ dx, float* __restrict__ dwpart,
                               int64_t rows, int H, int rows_per_unit) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int unit = blockIdx.x * kWaves + wave;
  const int nunits = gridDim.x * kWaves;

  RVec<T, VC> wv[CH];
#pragma unroll
  for (int j = 0; j < CH; ++j)
    wv[j] = reinterpret_cast<const RVec<T, VC>*>(w)[lane + 64 * j];

  float dwacc[CH][VC];
#pragma unroll
  for (int j = 0; j < CH; ++j)
#pragma unroll
    for (int k = 0; k < VC; ++k) dwacc[j][k] = 0.f;

  const int64_t r0 = static_cast<int64_t>(unit) * rows_per_unit;
  const int64_t r1 = r0 + rows_per_unit > rows ? rows : r0 + rows_per_unit;

  for (int64_t r = r0; r < r1; ++r) {
    const RVec<T, VC>* xr = reinterpret_cast<const RVec<T, VC>*>(x + r * H);
    const RVec<T, VC>* dr = reinterpret_cast<const RVec<T, VC>*>(dy + r * H);
    const float rstd = rstd_in[r];
    RVec<T, VC> xv[CH], dv[CH];
    float dot = 0.f;
#pragma unroll
    for (int j = 0; j < CH; ++j) {
      xv[j] = xr[lane + 64 * j];
      dv[j] = dr[lane + 64 * j];
#pragma unroll
      for (int k = 0; k < VC; ++k) {
        float xf = rf_of(xv[j].v[k]), df = rf_of(dv[j].v[k]);
        dot += df * rf_of(wv[j].v[k]) * xf;
        dwacc[j][k] += df * xf * rstd;
      }
    }
    const float c2 = wave_sum(dot) * rstd * rstd * rstd / H;

    RVec<T, VC>* oxr = reinterpret_cast<RVec<T, VC>*>(dx + r * H);
#pragma unroll
    for (int j = 0; j < CH; ++j) {
      RVec<T, VC> ov;
#pragma unroll
      for (int k = 0; k < VC; ++k)
        ov.v[k] = rt_of<T>(rstd * rf_of(wv[j].v[k]) * rf_of(dv[j].v[k]) -
                           rf_of(xv[j].v[k]) * c2);
      oxr[lane + 64 * j] = ov;
    }
  }

#pragma unroll
  for (int j = 0; j < CH; ++j)
#pragma unroll
    for (int k = 0; k < VC; ++k)
      dwpart[static_cast<int64_t>((lane + 64 * j) * VC + k) * nunits + unit] =
          dwacc[j][k];
}

// Fold each column's contiguous nblocks partials into dw[col].
__global__ void rms_fold_kernel(const float* __restrict__ dwpart,
                                float* __restrict__ dw, int nblocks) {
  const float* col = dwpart + static_cast<int64_t>(blockIdx.x) * nblocks;
  float acc = 0.f;
  for (int b = threadIdx.x; b < nblocks; b += static_cast<int>(blockDim.x))
    acc += col[b];
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_xor(acc, off, 64);
  __shared__ float red[kWaves];
  const int lane = threadIdx.x % 64, wave = threadIdx.x / 64;
  if (lane == 0) red[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0)
    dw[blockIdx.x] = red[0] + red[1] + red[2] + red[3];
}

}  // namespace

// Number of per-WAVE fold units the backward uses (callers size the dw
// partial buffer as fp32[H * rms_bwd_blocks(rows, ...)]). ~2048 waves
// fill the chip; idle tail units write zeroed partials.
int rms_bwd_blocks(int64_t rows, int* rows_per_unit) {
  int64_t rpu = std::max<int64_t>(1, (rows + 2047) / 2048);
  int64_t units = (rows + rpu - 1) / rpu;
  int blocks = static_cast<int>((units + 3) / 4);  // kWaves per block
  if (rows_per_unit) *rows_per_unit = static_cast<int>(rpu);
  return blocks * 4;
}

// Supported hidden sizes for the register-staged backward (VC, chunk
// template pairs). Others take the Python eager fallback.
bool rms_bwd_pick(int H, int* vc, int* ch) {
  switch (H) {
    case 64: *vc = 1; *ch = 1; return true;
    case 128: *vc = 2; *ch = 1; return true;
    case 256: *vc = 4; *ch = 1; return true;
    case 512: *vc = 8; *ch = 1; return true;
    case 1024: *vc = 8; *ch = 2; return true;
    case 2048: *vc = 8; *ch = 4; return true;
    case 4096: *vc = 8; *ch = 8; return true;
    default: return false;
  }
}

namespace {

int rms_vc(int H, size_t /*tsize*/) {
  if (H % 8 == 0) return 8;
  if (H % 4 == 0) return 4;
  if (H % 2 == 0) return 2;
  return 1;
}

template <typename T>
void rms_fwd_t(const void* x, const void* w, void* y, float* rstd, int64_t rows,
               int H, float eps, hipStream_t stream) {
  int vc = 0, ch = 0;
  rms_bwd_pick(H, &vc, &ch);  // caller validated H
  int rpu = 0;
  int blocks = rms_bwd_blocks(rows, &rpu) / 4;
  auto L = [&](auto vctag, auto chtag) {
    constexpr int V = decltype(vctag)::value;
    constexpr int C = decltype(chtag)::value;
    hipLaunchKernelGGL((rms_fwd_kernel<T, V, C>), blocks, kThreads, 0, stream,
                       static_cast<const T*>(x), static_cast<const T*>(w),
                       static_cast<T*>(y), rstd, rows, H, eps, rpu);
  };
  using I1 = std::integral_constant<int, 1>;
  using I2 = std::integral_constant<int, 2>;
  using I4 = std::integral_constant<int, 4>;
  using I8 = std::integral_constant<int, 8>;
  switch (vc * 16 + ch) {
    case 1 * 16 + 1: L(I1{}, I1{}); break;
    case 2 * 16 + 1: L(I2{}, I1{}); break;
    case 4 * 16 + 1: L(I4{}, I1{}); break;
    case 8 * 16 + 1: L(I8{}, I1{}); break;
    case 8 * 16 + 2: L(I8{}, I2{}); break;
    case 8 * 16 + 4: L(I8{}, I4{}); break;
    case 8 * 16 + 8: L(I8{}, I8{}); break;
    default: return;
  }
}

template <typename T>
void rms_bwd_t(const void* x, const void* dy, const void* w, const float* rstd,
               void* dx, float* dwpart, float* dw, int64_t rows, int H,
               hipStream_t stream) {
  int vc = 0, ch = 0;
  rms_bwd_pick(H, &vc, &ch);  // caller (bn_ops.cpp) validated H
  int rpu = 0;
  int nunits = rms_bwd_blocks(rows, &rpu);
  int blocks = nunits / 4;
  auto L = [&](auto vctag, auto chtag) {
    constexpr int V = decltype(vctag)::value;
    constexpr int C = decltype(chtag)::value;
    hipLaunchKernelGGL((rms_bwd_kernel<T, V, C>), blocks, kThreads, 0, stream,
                       static_cast<const T*>(x), static_cast<const T*>(dy),
                       static_cast<const T*>(w), rstd, static_cast<T*>(dx),
                       dwpart, rows, H, rpu);
  };
  using I1 = std::integral_constant<int, 1>;
  using I2 = std::integral_constant<int, 2>;
  using I4 = std::integral_constant<int, 4>;
  using I8 = std::integral_constant<int, 8>;
  switch (vc * 16 + ch) {
    case 1 * 16 + 1: L(I1{}, I1{}); break;
    case 2 * 16 + 1: L(I2{}, I1{}); break;
    case 4 * 16 + 1: L(I4{}, I1{}); break;
    case 8 * 16 + 1: L(I8{}, I1{}); break;
    case 8 * 16 + 2: L(I8{}, I2{}); break;
    case 8 * 16 + 4: L(I8{}, I4{}); break;
    case 8 * 16 + 8: L(I8{}, I8{}); break;
    default: return;  // unreachable (validated upstream)
  }
  hipLaunchKernelGGL(rms_fold_kernel, H, kThreads, 0, stream, dwpart, dw,
                     nunits);
}

}  // namespace

void launch_rms_fwd(int dtype, const void* x, const void* w, void* y,
                    float* rstd, int64_t rows, int H, float eps,
                    hipStream_t stream) {
  switch (dtype) {
    case kFloatId: rms_fwd_t<float>(x, w, y, rstd, rows, H, eps, stream); break;
    case kHalfId: rms_fwd_t<__half>(x, w, y, rstd, rows, H, eps, stream); break;
    case kBF16Id: rms_fwd_t<__hip_bfloat16>(x, w, y, rstd, rows, H, eps, stream); break;
  }
}

void launch_rms_bwd(int dtype, const void* x, const void* dy, const void* w,
                    const float* rstd, void* dx, float* dwpart, float* dw,
                    int64_t rows, int H, hipStream_t stream) {
  switch (dtype) {
    case kFloatId: rms_bwd_t<float>(x, dy, w, rstd, dx, dwpart, dw, rows, H, stream); break;
    case kHalfId: rms_bwd_t<__half>(x, dy, w, rstd, dx, dwpart, dw, rows, H, stream); break;
    case kBF16Id: rms_bwd_t<__hip_bfloat16>(x, dy, w, rstd, dx, dwpart, dw, rows, H, stream); break;
  }
}

}  // namespace axon
