// Fused NHWC (channels_last) BatchNorm for CDNA4 — the ResNet hot spot:
// rocprof shows ATen's channels_last BN kernels at ~62% of ResNet-50 GPU
// time on MI355X (profiles/README.md). These kernels are bandwidth-bound;
// the design keeps every pass a single coalesced stream over x/dy:
//
//   fwd train: stats (1 read of x) -> finalize (C-sized) -> apply
//              (1 read x, 1 write y, optional fused ReLU)
//   bwd:       reduce (read x, dy) -> finalize (C) -> apply
//              (read x, dy, write dx, optional fused ReLU mask from y)
//
// Layout: x is [rows, C] with C innermost (rows = N*H*W). A wavefront's 64
// lanes read 64*VC consecutive channels (VC = 8/4/2/1 bf16 lanes per
// vector load, chosen so 64*VC divides C when possible) — one fully
// coalesced 1 KiB (VC=8) transaction per wave. Four waves per block cover
// interleaved row stripes of the same channel tile; partial sums combine
// through LDS, then one fp32 atomicAdd per (block, channel) into the
// scratch accumulator. All arithmetic fp32.
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include <algorithm>
#include <type_traits>

#include "kernels.h"

namespace axon {

namespace {

constexpr int kThreads = 256;  // 4 waves of 64
constexpr int kWaves = 4;
constexpr int kHalfId = 5, kFloatId = 6, kBF16Id = 15;

template <typename T>
__device__ __forceinline__ float bf_of(T x);
template <>
__device__ __forceinline__ float bf_of(float x) { return x; }
template <>
__device__ __forceinline__ float bf_of(__half x) { return __half2float(x); }
template <>
__device__ __forceinline__ float bf_of(__hip_bfloat16 x) { return __bfloat162float(x); }

template <typename T>
__device__ __forceinline__ T bt_of(float x);
template <>
__device__ __forceinline__ float bt_of(float x) { return x; }
template <>
__device__ __forceinline__ __half bt_of(float x) { return __float2half(x); }
template <>
__device__ __forceinline__ __hip_bfloat16 bt_of(float x) { return __float2bfloat16(x); }

template <typename T, int V>
struct alignas(sizeof(T) * V) BVec {
  T v[V];
};

// ------------------------------------------------------ stats (fwd pass 1)
// scratch[c] += sum(x[:, c]); scratch[C + c] += sum(x[:, c]^2)

template <typename T, int VC>
__global__ void bn_stats_kernel(const T* __restrict__ x, int64_t rows, int C,
                                float* __restrict__ scratch, int rows_per_block) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int c0 = blockIdx.x * 64 * VC + lane * VC;  // my first channel
  const int64_t r0 = static_cast<int64_t>(blockIdx.y) * rows_per_block;
  const int64_t r1 = r0 + rows_per_block > rows ? rows : r0 + rows_per_block;

  float s[VC], q[VC];
#pragma unroll
  for (int k = 0; k < VC; ++k) s[k] = q[k] = 0.f;

  if (c0 + VC <= C) {
    for (int64_t r = r0 + wave; r < r1; r += kWaves) {
      BVec<T, VC> xv = *reinterpret_cast<const BVec<T, VC>*>(x + r * C + c0);
#pragma unroll
      for (int k = 0; k < VC; ++k) {
        float v = bf_of(xv.v[k]);
        s[k] += v;
        q[k] += v * v;
      }
    }
  } else if (c0 < C) {  // ragged tail (C not a multiple of 64*VC)
    for (int64_t r = r0 + wave; r < r1; r += kWaves) {
      for (int k = 0; k < VC && c0 + k < C; ++k) {
        float v = bf_of(x[r * C + c0 + k]);
        s[k] += v;
        q[k] += v * v;
      }
    }
  }

  // Cross-wave combine in LDS, then one atomic per channel per block.
  __shared__ float lds[kWaves][64 * VC * 2];
#pragma unroll
  for (int k = 0; k < VC; ++k) {
    lds[wave][lane * VC + k] = s[k];
    lds[wave][64 * VC + lane * VC + k] = q[k];
  }
  __syncthreads();
  if (wave == 0) {
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      int c = c0 + k;
      if (c >= C) break;
      float ss = 0.f, qq = 0.f;
#pragma unroll
      for (int w = 0; w < kWaves; ++w) {
        ss += lds[w][lane * VC + k];
        qq += lds[w][64 * VC + lane * VC + k];
      }
      atomicAdd(&scratch[c], ss);
      atomicAdd(&scratch[C + c], qq);
    }
  }
}

// -------------------------------------------------- finalize (fwd, C-sized)
// From scratch sums: save_mean/save_rstd, running-stat update, and the
// apply coefficients y = a*x + b.

template <typename PT>
__global__ void bn_fwd_finalize_kernel(const float* __restrict__ scratch, int C,
                                       float inv_n, float unbias, float momentum,
                                       float eps, const PT* __restrict__ gamma,
                                       const PT* __restrict__ beta,
                                       PT* __restrict__ running_mean,
                                       PT* __restrict__ running_var,
                                       float* __restrict__ save_mean,
                                       float* __restrict__ save_rstd,
                                       float* __restrict__ ab) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = scratch[c] * inv_n;
  float var = fmaxf(scratch[C + c] * inv_n - mean * mean, 0.f);
  float rstd = rsqrtf(var + eps);
  save_mean[c] = mean;
  save_rstd[c] = rstd;
  if (running_mean != nullptr) {
    running_mean[c] = bt_of<PT>((1.f - momentum) * bf_of(running_mean[c]) +
                                momentum * mean);
    running_var[c] = bt_of<PT>((1.f - momentum) * bf_of(running_var[c]) +
                               momentum * var * unbias);
  }
  float a = (gamma ? bf_of(gamma[c]) : 1.f) * rstd;
  ab[c] = a;
  ab[C + c] = (beta ? bf_of(beta[c]) : 0.f) - mean * a;
}

// Eval-mode coefficients from running stats.
template <typename PT>
__global__ void bn_eval_coef_kernel(int C, float eps,
                                    const PT* __restrict__ gamma,
                                    const PT* __restrict__ beta,
                                    const PT* __restrict__ running_mean,
                                    const PT* __restrict__ running_var,
                                    float* __restrict__ ab) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float rstd = rsqrtf(bf_of(running_var[c]) + eps);
  float a = (gamma ? bf_of(gamma[c]) : 1.f) * rstd;
  ab[c] = a;
  ab[C + c] = (beta ? bf_of(beta[c]) : 0.f) - bf_of(running_mean[c]) * a;
}

// ------------------------------------------------------- apply (fwd pass 2)
// y = relu?(a[c] * x + b[c]) — one coalesced stream; per-channel coeffs are
// broadcast reads served by L2/L1.

template <typename T, int VC, bool RELU>
__global__ void bn_apply_kernel(const T* __restrict__ x, T* __restrict__ y,
                                int64_t nvec, int C,
                                const float* __restrict__ ab) {
  const int cvec = C / VC;  // vectors per row (VC divides C by construction)
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int c0 = static_cast<int>(i % cvec) * VC;
    BVec<T, VC> xv = reinterpret_cast<const BVec<T, VC>*>(x)[i];
    BVec<T, VC> yv;
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      float out = fmaf(bf_of(xv.v[k]), ab[c0 + k], ab[C + c0 + k]);
      if (RELU) out = fmaxf(out, 0.f);
      yv.v[k] = bt_of<T>(out);
    }
    reinterpret_cast<BVec<T, VC>*>(y)[i] = yv;
  }
}

// -------------------------------------------------- backward reduce (pass 1)
// scratch[c] += sum(dy_eff[:, c]); scratch[C+c] += sum(dy_eff * (x - mean)).
// RELU fusion: dy_eff = (y > 0) ? dy : 0 (y = the module's fused output).

template <typename T, int VC, bool RELU>
__global__ void bn_bwd_reduce_kernel(const T* __restrict__ x,
                                     const T* __restrict__ dy,
                                     const T* __restrict__ y,  // RELU only
                                     int64_t rows, int C,
                                     const float* __restrict__ save_mean,
                                     float* __restrict__ scratch,
                                     int rows_per_block) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int c0 = blockIdx.x * 64 * VC + lane * VC;
  const int64_t r0 = static_cast<int64_t>(blockIdx.y) * rows_per_block;
  const int64_t r1 = r0 + rows_per_block > rows ? rows : r0 + rows_per_block;

  float sdy[VC], sxdy[VC], mean[VC];
#pragma unroll
  for (int k = 0; k < VC; ++k) {
    sdy[k] = sxdy[k] = 0.f;
    mean[k] = (c0 + k < C) ? save_mean[c0 + k] : 0.f;
  }

  if (c0 + VC <= C) {
    for (int64_t r = r0 + wave; r < r1; r += kWaves) {
      BVec<T, VC> xv = *reinterpret_cast<const BVec<T, VC>*>(x + r * C + c0);
      BVec<T, VC> dv = *reinterpret_cast<const BVec<T, VC>*>(dy + r * C + c0);
      BVec<T, VC> yv;
      if (RELU) yv = *reinterpret_cast<const BVec<T, VC>*>(y + r * C + c0);
#pragma unroll
      for (int k = 0; k < VC; ++k) {
        float d = bf_of(dv.v[k]);
        if (RELU && bf_of(yv.v[k]) <= 0.f) d = 0.f;
        sdy[k] += d;
        sxdy[k] += d * (bf_of(xv.v[k]) - mean[k]);
      }
    }
  } else if (c0 < C) {
    for (int64_t r = r0 + wave; r < r1; r += kWaves) {
      for (int k = 0; k < VC && c0 + k < C; ++k) {
        float d = bf_of(dy[r * C + c0 + k]);
        if (RELU && bf_of(y[r * C + c0 + k]) <= 0.f) d = 0.f;
        sdy[k] += d;
        sxdy[k] += d * (bf_of(x[r * C + c0 + k]) - mean[k]);
      }
    }
  }

  __shared__ float lds[kWaves][64 * VC * 2];
#pragma unroll
  for (int k = 0; k < VC; ++k) {
    lds[wave][lane * VC + k] = sdy[k];
    lds[wave][64 * VC + lane * VC + k] = sxdy[k];
  }
  __syncthreads();
  if (wave == 0) {
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      int c = c0 + k;
      if (c >= C) break;
      float a = 0.f, bsum = 0.f;
#pragma unroll
      for (int w = 0; w < kWaves; ++w) {
        a += lds[w][lane * VC + k];
        bsum += lds[w][64 * VC + lane * VC + k];
      }
      atomicAdd(&scratch[c], a);
      atomicAdd(&scratch[C + c], bsum);
    }
  }
}

// ------------------------------------------------ backward finalize (C-sized)
// dgamma = sum(dy*xhat) = sxdy * rstd ; dbeta = sdy ;
// dx = c1[c]*dy + c5[c]*x + c4[c]  with
//   c1 = gamma*rstd, c5 = -gamma*rstd^3*sxdy/N,
//   c4 = gamma*rstd*(mean*rstd^2*sxdy - sdy)/N
// (training-mode chain rule through batch statistics).

template <typename PT>
__global__ void bn_bwd_finalize_kernel(const float* __restrict__ scratch, int C,
                                       float inv_n,
                                       const PT* __restrict__ gamma,
                                       const float* __restrict__ save_mean,
                                       const float* __restrict__ save_rstd,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta,
                                       float* __restrict__ coef /* 3C */) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float sdy = scratch[c], sxdy = scratch[C + c];
  float rstd = save_rstd[c], mean = save_mean[c];
  float g = gamma ? bf_of(gamma[c]) : 1.f;
  dgamma[c] = sxdy * rstd;
  dbeta[c] = sdy;
  float c1 = g * rstd;
  float c5 = -c1 * rstd * rstd * sxdy * inv_n;
  coef[c] = c1;
  coef[C + c] = c5;
  coef[2 * C + c] = -(c1 * sdy * inv_n) - mean * c5;
}

// --------------------------------------------------- backward apply (pass 2)

template <typename T, int VC, bool RELU>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ x,
                                    const T* __restrict__ dy,
                                    const T* __restrict__ y,  // RELU only
                                    T* __restrict__ dx, int64_t nvec, int C,
                                    const float* __restrict__ coef) {
  const int cvec = C / VC;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int c0 = static_cast<int>(i % cvec) * VC;
    BVec<T, VC> xv = reinterpret_cast<const BVec<T, VC>*>(x)[i];
    BVec<T, VC> dv = reinterpret_cast<const BVec<T, VC>*>(dy)[i];
    BVec<T, VC> yv;
    if (RELU) yv = reinterpret_cast<const BVec<T, VC>*>(y)[i];
    BVec<T, VC> ov;
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      int c = c0 + k;
      float d = bf_of(dv.v[k]);
      if (RELU && bf_of(yv.v[k]) <= 0.f) d = 0.f;
      ov.v[k] = bt_of<T>(fmaf(d, coef[c],
                              fmaf(bf_of(xv.v[k]), coef[C + c], coef[2 * C + c])));
    }
    reinterpret_cast<BVec<T, VC>*>(dx)[i] = ov;
  }
}

// ------------------------------------------------------------- host launch

int pick_vc(int C) {
  if (C % 512 == 0) return 8;
  if (C % 256 == 0) return 4;
  if (C % 128 == 0) return 2;
  return 1;  // 64-channel tiles with ragged-tail bounds checks
}

int stats_rows_per_block(int64_t rows, int ctiles) {
  // Aim for >= ~1024 blocks total to fill 256 CUs across 8 XCDs, without
  // making blocks so small the LDS/atomic epilogue dominates.
  int64_t target_blocks = 2048 / (ctiles > 0 ? ctiles : 1) + 1;
  int64_t rpb = (rows + target_blocks - 1) / target_blocks;
  rpb = std::max<int64_t>(rpb, 64);
  return static_cast<int>(std::min<int64_t>(rpb, rows));
}

template <typename T, typename PT>
void bn_fwd_impl(const void* x, void* y, int64_t rows, int C, const void* gamma,
                 const void* beta, void* running_mean, void* running_var,
                 float momentum, float eps, bool training, bool relu,
                 float* scratch, float* save_mean, float* save_rstd, float* ab,
                 hipStream_t stream) {
  int vc = pick_vc(C);
  if (training) {
    int ctiles = (C + 64 * vc - 1) / (64 * vc);
    int rpb = stats_rows_per_block(rows, ctiles);
    dim3 grid(ctiles, static_cast<unsigned>((rows + rpb - 1) / rpb));
    switch (vc) {
      case 8: hipLaunchKernelGGL((bn_stats_kernel<T, 8>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, scratch, rpb); break;
      case 4: hipLaunchKernelGGL((bn_stats_kernel<T, 4>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, scratch, rpb); break;
      case 2: hipLaunchKernelGGL((bn_stats_kernel<T, 2>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, scratch, rpb); break;
      default: hipLaunchKernelGGL((bn_stats_kernel<T, 1>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, scratch, rpb); break;
    }
    float inv_n = 1.f / static_cast<float>(rows);
    float unbias = rows > 1 ? static_cast<float>(rows) / (rows - 1) : 1.f;
    hipLaunchKernelGGL((bn_fwd_finalize_kernel<PT>), (C + 255) / 256, 256, 0,
                       stream, scratch, C, inv_n, unbias, momentum, eps,
                       static_cast<const PT*>(gamma), static_cast<const PT*>(beta),
                       static_cast<PT*>(running_mean), static_cast<PT*>(running_var),
                       save_mean, save_rstd, ab);
  } else {
    hipLaunchKernelGGL((bn_eval_coef_kernel<PT>), (C + 255) / 256, 256, 0,
                       stream, C, eps, static_cast<const PT*>(gamma),
                       static_cast<const PT*>(beta),
                       static_cast<const PT*>(running_mean),
                       static_cast<const PT*>(running_var), ab);
  }
  // Apply: VC must divide C for the flat-index channel math; pick_vc
  // guarantees it except the vc==1 ragged case, which also divides.
  int64_t nvec = rows * C / vc;
  int blocks = static_cast<int>(std::min<int64_t>(
      (nvec + kThreads - 1) / kThreads, 16384));
  auto launch_apply = [&](auto vctag, auto relutag) {
    constexpr int V = decltype(vctag)::value;
    constexpr bool R = decltype(relutag)::value;
    hipLaunchKernelGGL((bn_apply_kernel<T, V, R>), blocks, kThreads, 0, stream,
                       static_cast<const T*>(x), static_cast<T*>(y), nvec, C, ab);
  };
  using K1 = std::integral_constant<int, 1>;
  using K2 = std::integral_constant<int, 2>;
  using K4 = std::integral_constant<int, 4>;
  using K8 = std::integral_constant<int, 8>;
  using RT = std::true_type;
  using RF = std::false_type;
  switch (vc) {
    case 8: relu ? launch_apply(K8{}, RT{}) : launch_apply(K8{}, RF{}); break;
    case 4: relu ? launch_apply(K4{}, RT{}) : launch_apply(K4{}, RF{}); break;
    case 2: relu ? launch_apply(K2{}, RT{}) : launch_apply(K2{}, RF{}); break;
    default: relu ? launch_apply(K1{}, RT{}) : launch_apply(K1{}, RF{}); break;
  }
}

template <typename T, typename PT>
void bn_bwd_impl(const void* x, const void* dy, const void* y, void* dx,
                 int64_t rows, int C, const void* gamma, const float* save_mean,
                 const float* save_rstd, bool relu, float* scratch,
                 float* dgamma, float* dbeta, float* coef, hipStream_t stream) {
  int vc = pick_vc(C);
  int ctiles = (C + 64 * vc - 1) / (64 * vc);
  int rpb = stats_rows_per_block(rows, ctiles);
  dim3 grid(ctiles, static_cast<unsigned>((rows + rpb - 1) / rpb));
  auto launch_reduce = [&](auto vctag, auto relutag) {
    constexpr int V = decltype(vctag)::value;
    constexpr bool R = decltype(relutag)::value;
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, V, R>), grid, kThreads, 0,
                       stream, static_cast<const T*>(x), static_cast<const T*>(dy),
                       static_cast<const T*>(y), rows, C, save_mean, scratch, rpb);
  };
  using K1 = std::integral_constant<int, 1>;
  using K2 = std::integral_constant<int, 2>;
  using K4 = std::integral_constant<int, 4>;
  using K8 = std::integral_constant<int, 8>;
  using RT = std::true_type;
  using RF = std::false_type;
  switch (vc) {
    case 8: relu ? launch_reduce(K8{}, RT{}) : launch_reduce(K8{}, RF{}); break;
    case 4: relu ? launch_reduce(K4{}, RT{}) : launch_reduce(K4{}, RF{}); break;
    case 2: relu ? launch_reduce(K2{}, RT{}) : launch_reduce(K2{}, RF{}); break;
    default: relu ? launch_reduce(K1{}, RT{}) : launch_reduce(K1{}, RF{}); break;
  }
  hipLaunchKernelGGL((bn_bwd_finalize_kernel<PT>), (C + 255) / 256, 256, 0,
                     stream, scratch, C, 1.f / static_cast<float>(rows),
                     static_cast<const PT*>(gamma), save_mean, save_rstd,
                     dgamma, dbeta, coef);
  int64_t nvec = rows * C / vc;
  int blocks = static_cast<int>(std::min<int64_t>(
      (nvec + kThreads - 1) / kThreads, 16384));
  auto launch_apply = [&](auto vctag, auto relutag) {
    constexpr int V = decltype(vctag)::value;
    constexpr bool R = decltype(relutag)::value;
    hipLaunchKernelGGL((bn_bwd_apply_kernel<T, V, R>), blocks, kThreads, 0,
                       stream, static_cast<const T*>(x), static_cast<const T*>(dy),
                       static_cast<const T*>(y), static_cast<T*>(dx), nvec, C, coef);
  };
  switch (vc) {
    case 8: relu ? launch_apply(K8{}, RT{}) : launch_apply(K8{}, RF{}); break;
    case 4: relu ? launch_apply(K4{}, RT{}) : launch_apply(K4{}, RF{}); break;
    case 2: relu ? launch_apply(K2{}, RT{}) : launch_apply(K2{}, RF{}); break;
    default: relu ? launch_apply(K1{}, RT{}) : launch_apply(K1{}, RF{}); break;
  }
}

template <typename T>
void bn_fwd_pt(int param_dtype, const void* x, void* y, int64_t rows, int C,
               const void* gamma, const void* beta, void* rm, void* rv,
               float momentum, float eps, bool training, bool relu,
               float* scratch, float* save_mean, float* save_rstd, float* ab,
               hipStream_t stream) {
  switch (param_dtype) {
    case kFloatId:
      bn_fwd_impl<T, float>(x, y, rows, C, gamma, beta, rm, rv, momentum, eps,
                            training, relu, scratch, save_mean, save_rstd, ab,
                            stream);
      break;
    case kHalfId:
      bn_fwd_impl<T, __half>(x, y, rows, C, gamma, beta, rm, rv, momentum, eps,
                             training, relu, scratch, save_mean, save_rstd, ab,
                             stream);
      break;
    case kBF16Id:
      bn_fwd_impl<T, __hip_bfloat16>(x, y, rows, C, gamma, beta, rm, rv,
                                     momentum, eps, training, relu, scratch,
                                     save_mean, save_rstd, ab, stream);
      break;
  }
}

template <typename T>
void bn_bwd_pt(int param_dtype, const void* x, const void* dy, const void* y,
               void* dx, int64_t rows, int C, const void* gamma,
               const float* save_mean, const float* save_rstd, bool relu,
               float* scratch, float* dgamma, float* dbeta, float* coef,
               hipStream_t stream) {
  switch (param_dtype) {
    case kFloatId:
      bn_bwd_impl<T, float>(x, dy, y, dx, rows, C, gamma, save_mean, save_rstd,
                            relu, scratch, dgamma, dbeta, coef, stream);
      break;
    case kHalfId:
      bn_bwd_impl<T, __half>(x, dy, y, dx, rows, C, gamma, save_mean, save_rstd,
                             relu, scratch, dgamma, dbeta, coef, stream);
      break;
    case kBF16Id:
      bn_bwd_impl<T, __hip_bfloat16>(x, dy, y, dx, rows, C, gamma, save_mean,
                                     save_rstd, relu, scratch, dgamma, dbeta,
                                     coef, stream);
      break;
  }
}

}  // namespace

void launch_bn_fwd(int x_dtype, int param_dtype, const void* x, void* y,
                   int64_t rows, int C, const void* gamma, const void* beta,
                   void* running_mean, void* running_var, float momentum,
                   float eps, bool training, bool relu, float* scratch,
                   float* save_mean, float* save_rstd, float* ab,
                   hipStream_t stream) {
  switch (x_dtype) {
    case kFloatId:
      bn_fwd_pt<float>(param_dtype, x, y, rows, C, gamma, beta, running_mean,
                       running_var, momentum, eps, training, relu, scratch,
                       save_mean, save_rstd, ab, stream);
      break;
    case kHalfId:
      bn_fwd_pt<__half>(param_dtype, x, y, rows, C, gamma, beta, running_mean,
                        running_var, momentum, eps, training, relu, scratch,
                        save_mean, save_rstd, ab, stream);
      break;
    case kBF16Id:
      bn_fwd_pt<__hip_bfloat16>(param_dtype, x, y, rows, C, gamma, beta,
                                running_mean, running_var, momentum, eps,
                                training, relu, scratch, save_mean, save_rstd,
                                ab, stream);
      break;
  }
}

void launch_bn_bwd(int x_dtype, int param_dtype, const void* x, const void* dy,
                   const void* y, void* dx, int64_t rows, int C,
                   const void* gamma, const float* save_mean,
                   const float* save_rstd, bool relu, float* scratch,
                   float* dgamma, float* dbeta, float* coef,
                   hipStream_t stream) {
  switch (x_dtype) {
    case kFloatId:
      bn_bwd_pt<float>(param_dtype, x, dy, y, dx, rows, C, gamma, save_mean,
                       save_rstd, relu, scratch, dgamma, dbeta, coef, stream);
      break;
    case kHalfId:
      bn_bwd_pt<__half>(param_dtype, x, dy, y, dx, rows, C, gamma, save_mean,
                        save_rstd, relu, scratch, dgamma, dbeta, coef, stream);
      break;
    case kBF16Id:
      bn_bwd_pt<__hip_bfloat16>(param_dtype, x, dy, y, dx, rows, C, gamma,
                                save_mean, save_rstd, relu, scratch, dgamma,
                                dbeta, coef, stream);
      break;
  }
}

}  // namespace axon
