// Launch interface of the hand-written CDNA4 HIP kernels (gfx950).
// K1/K2 batched fusion-buffer pack/unpack, K3 scale (fused
// into pack/unpack), K4 fp16 wire compress/decompress (fused likewise),
// K5 Adasum dots + scaled-add. K6 fused optimizers live in optimizers.hip.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace axon {

// One contiguous transfer chunk: pointers are pre-offset to the chunk start.
// n = element count (xform kernels) or byte count (byte-copy kernel).
struct ChunkDesc {
  const void* src;
  void* dst;
  int64_t n;
};

// Max chunks staged per launch (device descriptor buffer size).
constexpr int kMaxChunks = 16384;
// Elements processed per chunk/block (xform); bytes per chunk (copy).
constexpr int64_t kChunkElems = 1 << 16;   // 64K elements
constexpr int64_t kChunkBytes = 1 << 18;   // 256 KiB

// Batched memcpy (dtype-agnostic, n = bytes). descs must be in device
// memory; nchunks <= kMaxChunks.
void launch_copy_bytes(const ChunkDesc* descs, int nchunks, hipStream_t stream);

// Batched elementwise convert+scale: dst[i] = DstT(float(src[i]) * scale).
// src_dtype/dst_dtype are at::ScalarType codes (float/half/bfloat16/double).
// n = elements.
void launch_xform(const ChunkDesc* descs, int nchunks, int src_dtype,
                  int dst_dtype, float scale, hipStream_t stream);
bool xform_supported(int src_dtype, int dst_dtype);

// Adasum (K5): partial dot triple over this rank's segment.
// out3 += {a.b, a.a, b.b} in canonical a/b order (a_side tells which of
// kept/recv is 'a'). Accumulates in double; caller zeroes out3 first.
void launch_adasum_dots(const void* kept, const void* recv, int64_t n,
                        int dtype, double* out3, bool a_side, hipStream_t stream);
// mine[0..2] += peer[0..2] (adasum partial-dot butterfly merge).
void launch_add3(double* mine, const double* peer, hipStream_t stream);
// kept = ca*kept + cb*recv (a_side: kept is 'a') with
// ca = 1 - ab/(2*aa), cb = 1 - ab/(2*bb) read from dots3 on device.
void launch_adasum_apply(void* kept, const void* recv, int64_t n, int dtype,
                         const double* dots3, bool a_side, hipStream_t stream);

// Fused NHWC BatchNorm (batchnorm.hip): x/dy/y/dx are [rows, C] channels-
// last streams. partial is fp32[bn_partial_rows(rows,C) * 2C] (uninit ok),
// scratch fp32[2C]; save_mean/save_rstd/ab/coef are fp32[C]/[C]/[2C]/[3C].
// dtype args are at::ScalarType codes (float/half/bfloat16). relu fuses
// the activation into apply (fwd) and masks dy by y>0 (bwd). add (fwd) /
// dadd (bwd) fuse a residual: y = relu(a*x+b+add), dadd = relu-masked dy.
int bn_partial_rows(int64_t rows, int C);
void launch_bn_fwd(int x_dtype, int param_dtype, const void* x, void* y,
                   const void* add, int64_t rows, int C, const void* gamma,
                   const void* beta, void* running_mean, void* running_var,
                   float momentum, float eps, bool training, bool relu,
                   float* partial, float* scratch, float* save_mean,
                   float* save_rstd, float* ab, hipStream_t stream);
void launch_bn_bwd(int x_dtype, int param_dtype, const void* x, const void* dy,
                   const void* y, void* dx, void* dadd, int64_t rows, int C,
                   const void* gamma, const float* save_mean,
                   const float* save_rstd, bool relu, float* partial,
                   float* scratch, float* dgamma, float* dbeta, float* coef,
                   hipStream_t stream);

// Fused RMSNorm (rmsnorm.hip): x/y/dx are [rows, H] contiguous, w is [H]
// (same dtype as x), rstd fp32[rows]; dwpart is an uninitialized fp32
// [H * rms_bwd_blocks(rows)] partial buffer, dw fp32[H] (overwritten).
// Caller guarantees H % 2 == 0 and H <= 16384 (register/LDS bounds) —
// the Python wrapper falls back to eager otherwise.
void launch_rms_fwd(int dtype, const void* x, const void* w, void* y,
                    float* rstd, int64_t rows, int H, float eps,
                    hipStream_t stream);
int rms_bwd_blocks(int64_t rows, int* rows_per_unit);
// Supported (VC, chunk) pairs for the register-staged LN/RMS backward:
// H in {64,128,256,512,1024,2048,4096}; false => Python eager fallback.
bool rms_bwd_pick(int H, int* vc, int* ch);
void launch_rms_bwd(int dtype, const void* x, const void* dy, const void* w,
                    const float* rstd, void* dx, float* dwpart, float* dw,
                    int64_t rows, int H, hipStream_t stream);

// Fused LayerNorm (layernorm.hip): x/y/dx [rows, H] contiguous; w/b [H]
// in x's dtype; mean/rstd fp32[rows]; dwpart/dbpart uninitialized fp32
// [H * rms_bwd_blocks(rows)], dw/db fp32[H] (overwritten). Same bounds as
// RMSNorm (even H <= 16384; Python falls back otherwise).
void launch_ln_fwd(int dtype, const void* x, const void* w, const void* b,
                   void* y, float* mean, float* rstd, int64_t rows, int H,
                   float eps, hipStream_t stream);
void launch_ln_bwd(int dtype, const void* x, const void* dy, const void* w,
                   const float* mean, const float* rstd, void* dx,
                   float* dwpart, float* dbpart, float* dw, float* db,
                   int64_t rows, int H, hipStream_t stream);

// Rotary embedding (elementwise.hip): x/y are [B,S,H,D] contiguous (numel
// elements), cos/sin [S, D/2] in the same dtype; backward = transposed
// rotation. SwiGLU: z = silu(a)*g, all tensors same-shape contiguous.
void launch_rope(int dtype, const void* x, void* y, const void* cos_t,
                 const void* sin_t, int64_t numel, int half_d, int heads,
                 int seq, bool backward, hipStream_t stream);
void launch_swiglu_fwd(int dtype, const void* a, const void* g, void* z,
                       int64_t numel, hipStream_t stream);
void launch_swiglu_bwd(int dtype, const void* dz, const void* a, const void* g,
                       void* da, void* dg, int64_t numel, hipStream_t stream);

// EXPERIMENTAL flash-attention forward (flash_attn.hip, docs/FA_PLAN.md):
// q/k/v/o are [B, H(Q|K), S, 128] contiguous bf16; lse fp32 [B, HQ, S]
// (nullable). v0 correctness build, reachable only via tools/fa_probe.py.
void launch_fa_fwd(const void* q, const void* k, const void* v, void* o,
                   float* lse, int B, int HQ, int HK, int S, int D,
                   bool causal, hipStream_t stream);
// EXPERIMENTAL FA backward v0 (same layouts; delta fp32[B*HQ*S] scratch).
void launch_fa_bwd(const void* q, const void* k, const void* v,
                   const void* dout, const void* o, const float* lse,
                   float* delta, void* dq, void* dk, void* dv, int B, int HQ,
                   int HK, int S, int D, bool causal, hipStream_t stream);

}  // namespace axon
