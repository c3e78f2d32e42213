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
// partial is COLUMN-major [2C][nby]: partial[c*nby + blockIdx.y] = this
// block's row-range sum of x[:, c]; partial[(C+c)*nby + blockIdx.y] the
// sum of squares. No atomics (same-address atomicAdd measured 3-6x off
// streaming SOL with ~2k blocks per channel), and the per-channel partials
// are CONTIGUOUS so the fold kernel reduces them with coalesced vector
// loads — deterministic both stages.

// W = rows covered by ONE wave load iteration (64*VC = C*W when W > 1):
// for small C the wave's 64 VC-wide lanes span W whole rows, keeping the
// per-wave transaction at 64*VC elements (1 KiB at VC=8/bf16) instead of
// a narrow 2*C-byte strip. Lane layout: flat = lane*VC; its channels are
// flat % C .. +VC-1 (fixed per lane) at row offset flat / C.
template <typename T, int VC, int W>
__global__ void bn_stats_kernel(const T* __restrict__ x, int64_t rows, int C,
                                float* __restrict__ partial, int rows_per_block) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int flat = lane * VC;
  const int row_off = W > 1 ? flat / C : 0;
  const int c0 = W > 1 ? flat % C : blockIdx.x * 64 * VC + flat;
  const int64_t r0 = static_cast<int64_t>(blockIdx.y) * rows_per_block;
  const int64_t r1 = r0 + rows_per_block > rows ? rows : r0 + rows_per_block;

  float s[VC], q[VC];
#pragma unroll
  for (int k = 0; k < VC; ++k) s[k] = q[k] = 0.f;

  if (W > 1) {
    for (int64_t r = r0 + wave * W + row_off; r < r1; r += kWaves * W) {
      BVec<T, VC> xv = *reinterpret_cast<const BVec<T, VC>*>(x + r * C + c0);
#pragma unroll
      for (int k = 0; k < VC; ++k) {
        float v = bf_of(xv.v[k]);
        s[k] += v;
        q[k] += v * v;
      }
    }
  } else if (c0 + VC <= C) {
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

  // Cross-wave (and, for W>1, cross-row-group) combine in LDS, then one
  // coalesced partial store.
  __shared__ float lds[kWaves][64 * VC * 2];
#pragma unroll
  for (int k = 0; k < VC; ++k) {
    lds[wave][flat + k] = s[k];
    lds[wave][64 * VC + flat + k] = q[k];
  }
  __syncthreads();
  const int nby = gridDim.y;
  if (wave == 0 && flat < (W > 1 ? C : 64 * VC)) {
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      int c = c0 + k;
      if (c >= C) break;
      float ss = 0.f, qq = 0.f;
#pragma unroll
      for (int w = 0; w < kWaves; ++w)
#pragma unroll
        for (int m = 0; m < W; ++m) {
          ss += lds[w][m * C + flat + k];
          qq += lds[w][64 * VC + m * C + flat + k];
        }
      partial[static_cast<int64_t>(c) * nby + blockIdx.y] = ss;
      partial[static_cast<int64_t>(C + c) * nby + blockIdx.y] = qq;
    }
  }
}

// Fold each channel's contiguous nby partials into scratch[c]
// (deterministic order). One block per channel: 256 threads stream the
// column with coalesced loads, then a shuffle+LDS tree.
__global__ void bn_reduce_partials_kernel(const float* __restrict__ partial,
                                          float* __restrict__ scratch,
                                          int nby) {
  const float* col = partial + static_cast<int64_t>(blockIdx.x) * nby;
  float acc = 0.f;
  for (int b = threadIdx.x; b < nby; b += static_cast<int>(blockDim.x))
    acc += col[b];
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_xor(acc, off, 64);
  __shared__ float red[kWaves];
  const int lane = threadIdx.x % 64, wave = threadIdx.x / 64;
  if (lane == 0) red[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0)
    scratch[blockIdx.x] = red[0] + red[1] + red[2] + red[3];
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

template <typename T, int VC, bool RELU, bool ADD>
__global__ void bn_apply_kernel(const T* __restrict__ x, T* __restrict__ y,
                                const T* __restrict__ add,  // residual (ADD)
                                int64_t nvec, int C,
                                const float* __restrict__ ab) {
  const int cvec = C / VC;  // vectors per row (VC divides C by construction)
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int c0 = static_cast<int>(i % cvec) * VC;
    BVec<T, VC> xv = reinterpret_cast<const BVec<T, VC>*>(x)[i];
    BVec<T, VC> av;
    if (ADD) av = reinterpret_cast<const BVec<T, VC>*>(add)[i];
    BVec<T, VC> yv;
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      float out = fmaf(bf_of(xv.v[k]), ab[c0 + k], ab[C + c0 + k]);
      if (ADD) out += bf_of(av.v[k]);
      if (RELU) out = fmaxf(out, 0.f);
      yv.v[k] = bt_of<T>(out);
    }
    reinterpret_cast<BVec<T, VC>*>(y)[i] = yv;
  }
}

// -------------------------------------------------- backward reduce (pass 1)
// scratch[c] += sum(dy_eff[:, c]); scratch[C+c] += sum(dy_eff * (x - mean)).
// RELU fusion: dy_eff = (y > 0) ? dy : 0 (y = the module's fused output).

template <typename T, int VC, int W, bool RELU>
__global__ void bn_bwd_reduce_kernel(const T* __restrict__ x,
                                     const T* __restrict__ dy,
                                     const T* __restrict__ y,  // RELU only
                                     int64_t rows, int C,
                                     const float* __restrict__ save_mean,
                                     float* __restrict__ partial,
                                     int rows_per_block) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int flat = lane * VC;
  const int row_off = W > 1 ? flat / C : 0;
  const int c0 = W > 1 ? flat % C : blockIdx.x * 64 * VC + flat;
  const int64_t r0 = static_cast<int64_t>(blockIdx.y) * rows_per_block;
  const int64_t r1 = r0 + rows_per_block > rows ? rows : r0 + rows_per_block;

  float sdy[VC], sxdy[VC], mean[VC];
#pragma unroll
  for (int k = 0; k < VC; ++k) {
    sdy[k] = sxdy[k] = 0.f;
    mean[k] = (c0 + k < C) ? save_mean[c0 + k] : 0.f;
  }

  if (W > 1) {
    for (int64_t r = r0 + wave * W + row_off; r < r1; r += kWaves * W) {
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
  } else if (c0 + VC <= C) {
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
    lds[wave][flat + k] = sdy[k];
    lds[wave][64 * VC + flat + k] = sxdy[k];
  }
  __syncthreads();
  const int nby = gridDim.y;
  if (wave == 0 && flat < (W > 1 ? C : 64 * VC)) {
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      int c = c0 + k;
      if (c >= C) break;
      float a = 0.f, bsum = 0.f;
#pragma unroll
      for (int w = 0; w < kWaves; ++w)
#pragma unroll
        for (int m = 0; m < W; ++m) {
          a += lds[w][m * C + flat + k];
          bsum += lds[w][64 * VC + m * C + flat + k];
        }
      partial[static_cast<int64_t>(c) * nby + blockIdx.y] = a;
      partial[static_cast<int64_t>(C + c) * nby + blockIdx.y] = bsum;
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

template <typename T, int VC, bool RELU, bool DADD>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ x,
                                    const T* __restrict__ dy,
                                    const T* __restrict__ y,  // RELU only
                                    T* __restrict__ dx,
                                    T* __restrict__ dadd,  // DADD: dy_eff out
                                    int64_t nvec, int C,
                                    const float* __restrict__ coef) {
  const int cvec = C / VC;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int c0 = static_cast<int>(i % cvec) * VC;
    BVec<T, VC> xv = reinterpret_cast<const BVec<T, VC>*>(x)[i];
    BVec<T, VC> dv = reinterpret_cast<const BVec<T, VC>*>(dy)[i];
    BVec<T, VC> yv;
    if (RELU) yv = reinterpret_cast<const BVec<T, VC>*>(y)[i];
    BVec<T, VC> ov, dav;
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      int c = c0 + k;
      float d = bf_of(dv.v[k]);
      if (RELU && bf_of(yv.v[k]) <= 0.f) d = 0.f;
      if (DADD) dav.v[k] = bt_of<T>(d);
      ov.v[k] = bt_of<T>(fmaf(d, coef[c],
                              fmaf(bf_of(xv.v[k]), coef[C + c], coef[2 * C + c])));
    }
    reinterpret_cast<BVec<T, VC>*>(dx)[i] = ov;
    if (DADD) reinterpret_cast<BVec<T, VC>*>(dadd)[i] = dav;
  }
}

// ------------------------------------------------------------- host launch

int pick_vc(int C) {
  if (C % 512 == 0) return 8;
  if (C % 256 == 0) return 4;
  if (C % 128 == 0) return 2;
  return 1;  // 64-channel tiles with ragged-tail bounds checks
}

// Rows per wave iteration for the reduce kernels: small C gets W whole
// rows per 1 KiB wave transaction (64*8 == C*W).
int pick_w(int C) {
  if (C % 8 == 0 && C <= 512 && 512 % C == 0) return 512 / C;
  return 1;
}

// Vector width for the APPLY kernels (flat elementwise): only needs VC|C.
int pick_apply_vc(int C) {
  if (C % 8 == 0) return 8;
  if (C % 4 == 0) return 4;
  if (C % 2 == 0) return 2;
  return 1;
}

}  // namespace

int bn_partial_rows(int64_t rows, int C) {
  // grid.y for the reduction kernels: enough blocks to fill 256 CUs / 8
  // XCDs (with grid.x = ctiles), few enough that the partial buffer and
  // its fold stay trivial.
  int vc = pick_vc(C);
  int ctiles = (C + 64 * vc - 1) / (64 * vc);
  int64_t target = 1024 / ctiles + 1;
  int64_t max_rows_blocks = (rows + 63) / 64;
  return static_cast<int>(std::max<int64_t>(1, std::min(target, max_rows_blocks)));
}

namespace {

template <typename T, typename PT>
void bn_fwd_impl(const void* x, void* y, const void* add, int64_t rows, int C,
                 const void* gamma, const void* beta, void* running_mean,
                 void* running_var, float momentum, float eps, bool training,
                 bool relu, float* partial, float* scratch, float* save_mean,
                 float* save_rstd, float* ab, hipStream_t stream) {
  const int w = pick_w(C);
  const int vc = w > 1 ? 8 : pick_vc(C);
  if (training) {
    int ctiles = w > 1 ? 1 : (C + 64 * vc - 1) / (64 * vc);
    int nby = bn_partial_rows(rows, C);
    int rpb = static_cast<int>((rows + nby - 1) / nby);
    nby = static_cast<int>((rows + rpb - 1) / rpb);
    dim3 grid(ctiles, static_cast<unsigned>(nby));
    switch (w > 1 ? 8 + w : vc) {
      case 16: hipLaunchKernelGGL((bn_stats_kernel<T, 8, 8>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, partial, rpb); break;
      case 12: hipLaunchKernelGGL((bn_stats_kernel<T, 8, 4>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, partial, rpb); break;
      case 10: hipLaunchKernelGGL((bn_stats_kernel<T, 8, 2>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, partial, rpb); break;
      case 8: hipLaunchKernelGGL((bn_stats_kernel<T, 8, 1>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, partial, rpb); break;
      case 4: hipLaunchKernelGGL((bn_stats_kernel<T, 4, 1>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, partial, rpb); break;
      case 2: hipLaunchKernelGGL((bn_stats_kernel<T, 2, 1>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, partial, rpb); break;
      default: hipLaunchKernelGGL((bn_stats_kernel<T, 1, 1>), grid, kThreads, 0, stream, static_cast<const T*>(x), rows, C, partial, rpb); break;
    }
    hipLaunchKernelGGL(bn_reduce_partials_kernel, 2 * C, 256, 0,
                       stream, partial, scratch, nby);
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
  const int vca = pick_apply_vc(C);
  int64_t nvec = rows * C / vca;
  int blocks = static_cast<int>(std::min<int64_t>(
      (nvec + kThreads - 1) / kThreads, 16384));
  const bool has_add = add != nullptr;
  auto launch_apply = [&](auto vctag, auto relutag, auto addtag) {
    constexpr int V = decltype(vctag)::value;
    constexpr bool R = decltype(relutag)::value;
    constexpr bool A = decltype(addtag)::value;
    hipLaunchKernelGGL((bn_apply_kernel<T, V, R, A>), blocks, kThreads, 0,
                       stream, static_cast<const T*>(x), static_cast<T*>(y),
                       static_cast<const T*>(add), nvec, C, ab);
  };
  using K1 = std::integral_constant<int, 1>;
  using K2 = std::integral_constant<int, 2>;
  using K4 = std::integral_constant<int, 4>;
  using K8 = std::integral_constant<int, 8>;
  using BT = std::true_type;
  using BF = std::false_type;
  auto dispatch = [&](auto vctag) {
    if (relu) {
      has_add ? launch_apply(vctag, BT{}, BT{}) : launch_apply(vctag, BT{}, BF{});
    } else {
      has_add ? launch_apply(vctag, BF{}, BT{}) : launch_apply(vctag, BF{}, BF{});
    }
  };
  switch (vca) {
    case 8: dispatch(K8{}); break;
    case 4: dispatch(K4{}); break;
    case 2: dispatch(K2{}); break;
    default: dispatch(K1{}); break;
  }
}

template <typename T, typename PT>
void bn_bwd_impl(const void* x, const void* dy, const void* y, void* dx,
                 void* dadd, int64_t rows, int C, const void* gamma,
                 const float* save_mean, const float* save_rstd, bool relu,
                 float* partial, float* scratch, float* dgamma, float* dbeta,
                 float* coef, hipStream_t stream) {
  const int w = pick_w(C);
  const int vc = w > 1 ? 8 : pick_vc(C);
  int ctiles = w > 1 ? 1 : (C + 64 * vc - 1) / (64 * vc);
  int nby = bn_partial_rows(rows, C);
  int rpb = static_cast<int>((rows + nby - 1) / nby);
  nby = static_cast<int>((rows + rpb - 1) / rpb);
  dim3 grid(ctiles, static_cast<unsigned>(nby));
  using K1 = std::integral_constant<int, 1>;
  using K2 = std::integral_constant<int, 2>;
  using K4 = std::integral_constant<int, 4>;
  using K8 = std::integral_constant<int, 8>;
  using BT = std::true_type;
  using BF = std::false_type;
  auto launch_reduce = [&](auto vctag, auto wtag, auto relutag) {
    constexpr int V = decltype(vctag)::value;
    constexpr int WW = decltype(wtag)::value;
    constexpr bool R = decltype(relutag)::value;
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, V, WW, R>), grid, kThreads, 0,
                       stream, static_cast<const T*>(x), static_cast<const T*>(dy),
                       static_cast<const T*>(y), rows, C, save_mean, partial, rpb);
  };
  switch (w > 1 ? 8 + w : vc) {
    case 16: relu ? launch_reduce(K8{}, K8{}, BT{}) : launch_reduce(K8{}, K8{}, BF{}); break;
    case 12: relu ? launch_reduce(K8{}, K4{}, BT{}) : launch_reduce(K8{}, K4{}, BF{}); break;
    case 10: relu ? launch_reduce(K8{}, K2{}, BT{}) : launch_reduce(K8{}, K2{}, BF{}); break;
    case 8: relu ? launch_reduce(K8{}, K1{}, BT{}) : launch_reduce(K8{}, K1{}, BF{}); break;
    case 4: relu ? launch_reduce(K4{}, K1{}, BT{}) : launch_reduce(K4{}, K1{}, BF{}); break;
    case 2: relu ? launch_reduce(K2{}, K1{}, BT{}) : launch_reduce(K2{}, K1{}, BF{}); break;
    default: relu ? launch_reduce(K1{}, K1{}, BT{}) : launch_reduce(K1{}, K1{}, BF{}); break;
  }
  hipLaunchKernelGGL(bn_reduce_partials_kernel, 2 * C, 256, 0,
                     stream, partial, scratch, nby);
  hipLaunchKernelGGL((bn_bwd_finalize_kernel<PT>), (C + 255) / 256, 256, 0,
                     stream, scratch, C, 1.f / static_cast<float>(rows),
                     static_cast<const PT*>(gamma), save_mean, save_rstd,
                     dgamma, dbeta, coef);
  const int vca = pick_apply_vc(C);
  int64_t nvec = rows * C / vca;
  int blocks = static_cast<int>(std::min<int64_t>(
      (nvec + kThreads - 1) / kThreads, 16384));
  const bool has_dadd = dadd != nullptr;
  auto launch_apply = [&](auto vctag, auto relutag, auto dat) {
    constexpr int V = decltype(vctag)::value;
    constexpr bool R = decltype(relutag)::value;
    constexpr bool D = decltype(dat)::value;
    hipLaunchKernelGGL((bn_bwd_apply_kernel<T, V, R, D>), blocks, kThreads, 0,
                       stream, static_cast<const T*>(x), static_cast<const T*>(dy),
                       static_cast<const T*>(y), static_cast<T*>(dx),
                       static_cast<T*>(dadd), nvec, C, coef);
  };
  auto dispatch = [&](auto vctag) {
    if (relu) {
      has_dadd ? launch_apply(vctag, BT{}, BT{}) : launch_apply(vctag, BT{}, BF{});
    } else {
      has_dadd ? launch_apply(vctag, BF{}, BT{}) : launch_apply(vctag, BF{}, BF{});
    }
  };
  switch (vca) {
    case 8: dispatch(K8{}); break;
    case 4: dispatch(K4{}); break;
    case 2: dispatch(K2{}); break;
    default: dispatch(K1{}); break;
  }
}

template <typename T>
void bn_fwd_pt(int param_dtype, const void* x, void* y, const void* add,
               int64_t rows, int C, const void* gamma, const void* beta,
               void* rm, void* rv, float momentum, float eps, bool training,
               bool relu, float* partial, float* scratch, float* save_mean,
               float* save_rstd, float* ab, hipStream_t stream) {
  switch (param_dtype) {
    case kFloatId:
      bn_fwd_impl<T, float>(x, y, add, rows, C, gamma, beta, rm, rv, momentum,
                            eps, training, relu, partial, scratch, save_mean,
                            save_rstd, ab, stream);
      break;
    case kHalfId:
      bn_fwd_impl<T, __half>(x, y, add, rows, C, gamma, beta, rm, rv, momentum,
                             eps, training, relu, partial, scratch, save_mean,
                             save_rstd, ab, stream);
      break;
    case kBF16Id:
      bn_fwd_impl<T, __hip_bfloat16>(x, y, add, rows, C, gamma, beta, rm, rv,
                                     momentum, eps, training, relu, partial,
                                     scratch, save_mean, save_rstd, ab, stream);
      break;
  }
}

template <typename T>
void bn_bwd_pt(int param_dtype, const void* x, const void* dy, const void* y,
               void* dx, void* dadd, int64_t rows, int C, const void* gamma,
               const float* save_mean, const float* save_rstd, bool relu,
               float* partial, float* scratch, float* dgamma, float* dbeta,
               float* coef, hipStream_t stream) {
  switch (param_dtype) {
    case kFloatId:
      bn_bwd_impl<T, float>(x, dy, y, dx, dadd, rows, C, gamma, save_mean,
                            save_rstd, relu, partial, scratch, dgamma, dbeta,
                            coef, stream);
      break;
    case kHalfId:
      bn_bwd_impl<T, __half>(x, dy, y, dx, dadd, rows, C, gamma, save_mean,
                             save_rstd, relu, partial, scratch, dgamma, dbeta,
                             coef, stream);
      break;
    case kBF16Id:
      bn_bwd_impl<T, __hip_bfloat16>(x, dy, y, dx, dadd, rows, C, gamma,
                                     save_mean, save_rstd, relu, partial,
                                     scratch, dgamma, dbeta, coef, stream);
      break;
  }
}

}  // namespace

void launch_bn_fwd(int x_dtype, int param_dtype, const void* x, void* y,
                   const void* add, int64_t rows, int C, const void* gamma,
                   const void* beta, void* running_mean, void* running_var,
                   float momentum, float eps, bool training, bool relu,
                   float* partial, float* scratch, float* save_mean,
                   float* save_rstd, float* ab, hipStream_t stream) {
  switch (x_dtype) {
    case kFloatId:
      bn_fwd_pt<float>(param_dtype, x, y, add, rows, C, gamma, beta,
                       running_mean, running_var, momentum, eps, training, relu,
                       partial, scratch, save_mean, save_rstd, ab, stream);
      break;
    case kHalfId:
      bn_fwd_pt<__half>(param_dtype, x, y, add, rows, C, gamma, beta,
                        running_mean, running_var, momentum, eps, training,
                        relu, partial, scratch, save_mean, save_rstd, ab,
                        stream);
      break;
    case kBF16Id:
      bn_fwd_pt<__hip_bfloat16>(param_dtype, x, y, add, rows, C, gamma, beta,
                                running_mean, running_var, momentum, eps,
                                training, relu, partial, scratch, save_mean,
                                save_rstd, ab, stream);
      break;
  }
}

void launch_bn_bwd(int x_dtype, int param_dtype, const void* x, const void* dy,
                   const void* y, void* dx, void* dadd, int64_t rows, int C,
                   const void* gamma, const float* save_mean,
                   const float* save_rstd, bool relu, float* partial,
                   float* scratch, float* dgamma, float* dbeta, float* coef,
                   hipStream_t stream) {
  switch (x_dtype) {
    case kFloatId:
      bn_bwd_pt<float>(param_dtype, x, dy, y, dx, dadd, rows, C, gamma,
                       save_mean, save_rstd, relu, partial, scratch, dgamma,
                       dbeta, coef, stream);
      break;
    case kHalfId:
      bn_bwd_pt<__half>(param_dtype, x, dy, y, dx, dadd, rows, C, gamma,
                        save_mean, save_rstd, relu, partial, scratch, dgamma,
                        dbeta, coef, stream);
      break;
    case kBF16Id:
      bn_bwd_pt<__hip_bfloat16>(param_dtype, x, dy, y, dx, dadd, rows, C,
                                gamma, save_mean, save_rstd, relu, partial,
                                scratch, dgamma, dbeta, coef, stream);
      break;
  }
}

}  // namespace axon
