// Fused LayerNorm for CDNA4 (the BERT path; same design as rmsnorm.hip):
// one kernel and one global read of x per direction, one WAVE per row
// with the row held in registers (no LDS, no barriers), fp32 arithmetic,
// column-major partial buffers for dgamma/dbeta (no atomics).
//
//   fwd: y = (x - mean) * rstd * w + b ; saves per-row mean/rstd (fp32)
//   bwd: g = dy*w ; dx = rstd*(g - mean_r(g) - xhat*mean_r(g*xhat))
//        dw = sum_rows(dy * xhat) ; db = sum_rows(dy)
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include <algorithm>
#include <type_traits>

#include "kernels.h"

namespace axon {

int rms_bwd_blocks(int64_t rows, int* rows_per_unit);  // rmsnorm.hip
bool rms_bwd_pick(int H, int* vc, int* ch);             // rmsnorm.hip

namespace {

constexpr int kThreads = 256;
constexpr int kWaves = 4;
constexpr int kHalfId = 5, kFloatId = 6, kBF16Id = 15;
constexpr int kMaxLnChunks = 4;  // dgamma/dbeta register arrays: H <= 8192@VC8

template <typename T>
__device__ __forceinline__ float lf_of(T x);
template <>
__device__ __forceinline__ float lf_of(float x) { return x; }
template <>
__device__ __forceinline__ float lf_of(__half x) { return __half2float(x); }
template <>
__device__ __forceinline__ float lf_of(__hip_bfloat16 x) { return __bfloat162float(x); }

template <typename T>
__device__ __forceinline__ T lt_of(float x);
template <>
__device__ __forceinline__ float lt_of(float x) { return x; }
template <>
__device__ __forceinline__ __half lt_of(float x) { return __float2half(x); }
template <>
__device__ __forceinline__ __hip_bfloat16 lt_of(float x) { return __float2bfloat16(x); }

template <typename T, int V>
struct alignas(sizeof(T) * V) LVec {
  T v[V];
};

__device__ __forceinline__ float ln_wave_sum(float v) {
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, 64);
  return v;
}

// ------------------------------------------------------------------ forward
// One WAVE per row, registers only (see the backward below).

template <typename T, int VC, int CH>
__global__ void ln_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                              const T* __restrict__ b, T* __restrict__ y,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out, int64_t rows, int H,
                              float eps, int rows_per_unit) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int unit = blockIdx.x * kWaves + wave;

  LVec<T, VC> wv[CH], bv[CH];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    wv[j] = reinterpret_cast<const LVec<T, VC>*>(w)[lane + 64 * j];
    bv[j] = reinterpret_cast<const LVec<T, VC>*>(b)[lane + 64 * j];
  }

  const int64_t r0 = static_cast<int64_t>(unit) * rows_per_unit;
  const int64_t r1 = r0 + rows_per_unit > rows ? rows : r0 + rows_per_unit;

  for (int64_t r = r0; r < r1; ++r) {
    const LVec<T, VC>* xr = reinterpret_cast<const LVec<T, VC>*>(x + r * H);
    LVec<T, VC> xv[CH];
    float s = 0.f, q = 0.f;
#pragma unroll
    for (int j = 0; j < CH; ++j) {
      xv[j] = xr[lane + 64 * j];
#pragma unroll
      for (int k = 0; k < VC; ++k) {
        float f = lf_of(xv[j].v[k]);
        s += f;
        q += f * f;
      }
    }
    const float mean = ln_wave_sum(s) / H;
    const float var = fmaxf(ln_wave_sum(q) / H - mean * mean, 0.f);
    const float rstd = rsqrtf(var + eps);
    if (lane == 0) {
      if (mean_out) mean_out[r] = mean;
      if (rstd_out) rstd_out[r] = rstd;
    }

    LVec<T, VC>* yr = reinterpret_cast<LVec<T, VC>*>(y + r * H);
#pragma unroll
    for (int j = 0; j < CH; ++j) {
      LVec<T, VC> yv;
#pragma unroll
      for (int k = 0; k < VC; ++k)
        yv.v[k] = lt_of<T>(fmaf((lf_of(xv[j].v[k]) - mean) * rstd,
                                lf_of(wv[j].v[k]), lf_of(bv[j].v[k])));
      yr[lane + 64 * j] = yv;
    }
  }
}

// ----------------------------------------------------------------- backward
// One WAVE per row, registers only (see rmsnorm.hip backward: the
// block-per-row/LDS version measured ~7x off streaming SOL at H=1024).
// Two row sums (s1 = sum g, s2 = sum g*xhat) reduce with xor-shuffles.

template <typename T, int VC, int CH>
__global__ void ln_bwd_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                              const T* __restrict__ w,
                              const float* __restrict__ mean_in,
                              const float* __restrict__ rstd_in,
                              T* __restrict__ dx, float* __restrict__ dwpart,
                              float* __restrict__ dbpart, int64_t rows, int H,
                              int rows_per_unit) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int unit = blockIdx.x * kWaves + wave;
  const int nunits = gridDim.x * kWaves;

  LVec<T, VC> wv[CH];
#pragma unroll
  for (int j = 0; j < CH; ++j)
    wv[j] = reinterpret_cast<const LVec<T, VC>*>(w)[lane + 64 * j];

  float dwacc[CH][VC], dbacc[CH][VC];
#pragma unroll
  for (int j = 0; j < CH; ++j)
#pragma unroll
    for (int k = 0; k < VC; ++k) dwacc[j][k] = dbacc[j][k] = 0.f;

  const int64_t r0 = static_cast<int64_t>(unit) * rows_per_unit;
  const int64_t r1 = r0 + rows_per_unit > rows ? rows : r0 + rows_per_unit;

  for (int64_t r = r0; r < r1; ++r) {
    const LVec<T, VC>* xr = reinterpret_cast<const LVec<T, VC>*>(x + r * H);
    const LVec<T, VC>* dr = reinterpret_cast<const LVec<T, VC>*>(dy + r * H);
    const float mean = mean_in[r], rstd = rstd_in[r];
    LVec<T, VC> xv[CH], dv[CH];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int j = 0; j < CH; ++j) {
      xv[j] = xr[lane + 64 * j];
      dv[j] = dr[lane + 64 * j];
#pragma unroll
      for (int k = 0; k < VC; ++k) {
        float xhat = (lf_of(xv[j].v[k]) - mean) * rstd;
        float df = lf_of(dv[j].v[k]);
        float g = df * lf_of(wv[j].v[k]);
        s1 += g;
        s2 += g * xhat;
        dwacc[j][k] += df * xhat;
        dbacc[j][k] += df;
      }
    }
    const float m1 = ln_wave_sum(s1) / H;
    const float m2 = ln_wave_sum(s2) / H;

    LVec<T, VC>* oxr = reinterpret_cast<LVec<T, VC>*>(dx + r * H);
#pragma unroll
    for (int j = 0; j < CH; ++j) {
      LVec<T, VC> ov;
#pragma unroll
      for (int k = 0; k < VC; ++k) {
        float xhat = (lf_of(xv[j].v[k]) - mean) * rstd;
        float g = lf_of(dv[j].v[k]) * lf_of(wv[j].v[k]);
        ov.v[k] = lt_of<T>(rstd * (g - m1 - xhat * m2));
      }
      oxr[lane + 64 * j] = ov;
    }
  }

#pragma unroll
  for (int j = 0; j < CH; ++j)
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      int64_t col = static_cast<int64_t>((lane + 64 * j) * VC + k);
      dwpart[col * nunits + unit] = dwacc[j][k];
      dbpart[col * nunits + unit] = dbacc[j][k];
    }
}

__global__ void ln_fold_kernel(const float* __restrict__ part,
                               float* __restrict__ out, int nblocks) {
  const float* col = part + static_cast<int64_t>(blockIdx.x) * nblocks;
  float acc = 0.f;
  for (int b = threadIdx.x; b < nblocks; b += static_cast<int>(blockDim.x))
    acc += col[b];
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_xor(acc, off, 64);
  __shared__ float red[kWaves];
  const int lane = threadIdx.x % 64, wave = threadIdx.x / 64;
  if (lane == 0) red[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) out[blockIdx.x] = red[0] + red[1] + red[2] + red[3];
}

int ln_vc(int H) {
  if (H % 8 == 0) return 8;
  if (H % 4 == 0) return 4;
  if (H % 2 == 0) return 2;
  return 1;
}

template <typename T>
void ln_fwd_t(const void* x, const void* w, const void* b, void* y,
              float* mean, float* rstd, int64_t rows, int H, float eps,
              hipStream_t stream) {
  int vc = 0, ch = 0;
  rms_bwd_pick(H, &vc, &ch);  // caller validated H
  int rpu = 0;
  int blocks = rms_bwd_blocks(rows, &rpu) / 4;
  auto L = [&](auto vctag, auto chtag) {
    constexpr int V = decltype(vctag)::value;
    constexpr int C = decltype(chtag)::value;
    hipLaunchKernelGGL((ln_fwd_kernel<T, V, C>), blocks, kThreads, 0, stream,
                       static_cast<const T*>(x), static_cast<const T*>(w),
                       static_cast<const T*>(b), static_cast<T*>(y), mean,
                       rstd, rows, H, eps, rpu);
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
void ln_bwd_t(const void* x, const void* dy, const void* w, const float* mean,
              const float* rstd, void* dx, float* dwpart, float* dbpart,
              float* dw, float* db, int64_t rows, int H, hipStream_t stream) {
  int vc = 0, ch = 0;
  rms_bwd_pick(H, &vc, &ch);  // caller (bn_ops.cpp) validated H
  int rpu = 0;
  int nunits = rms_bwd_blocks(rows, &rpu);
  int blocks = nunits / 4;
  auto L = [&](auto vctag, auto chtag) {
    constexpr int V = decltype(vctag)::value;
    constexpr int C = decltype(chtag)::value;
    hipLaunchKernelGGL((ln_bwd_kernel<T, V, C>), blocks, kThreads, 0, stream,
                       static_cast<const T*>(x), static_cast<const T*>(dy),
                       static_cast<const T*>(w), mean, rstd,
                       static_cast<T*>(dx), dwpart, dbpart, rows, H, rpu);
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
  hipLaunchKernelGGL(ln_fold_kernel, H, kThreads, 0, stream, dwpart, dw, nunits);
  hipLaunchKernelGGL(ln_fold_kernel, H, kThreads, 0, stream, dbpart, db, nunits);
}

}  // namespace

void launch_ln_fwd(int dtype, const void* x, const void* w, const void* b,
                   void* y, float* mean, float* rstd, int64_t rows, int H,
                   float eps, hipStream_t stream) {
  switch (dtype) {
    case kFloatId: ln_fwd_t<float>(x, w, b, y, mean, rstd, rows, H, eps, stream); break;
    case kHalfId: ln_fwd_t<__half>(x, w, b, y, mean, rstd, rows, H, eps, stream); break;
    case kBF16Id: ln_fwd_t<__hip_bfloat16>(x, w, b, y, mean, rstd, rows, H, eps, stream); break;
  }
}

void launch_ln_bwd(int dtype, const void* x, const void* dy, const void* w,
                   const float* mean, const float* rstd, void* dx,
                   float* dwpart, float* dbpart, float* dw, float* db,
                   int64_t rows, int H, hipStream_t stream) {
  switch (dtype) {
    case kFloatId: ln_bwd_t<float>(x, dy, w, mean, rstd, dx, dwpart, dbpart, dw, db, rows, H, stream); break;
    case kHalfId: ln_bwd_t<__half>(x, dy, w, mean, rstd, dx, dwpart, dbpart, dw, db, rows, H, stream); break;
    case kBF16Id: ln_bwd_t<__hip_bfloat16>(x, dy, w, mean, rstd, dx, dwpart, dbpart, dw, db, rows, H, stream); break;
  }
}

}  // namespace axon
