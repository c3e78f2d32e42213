// EXPERIMENTAL (round-2 work in progress; see docs/FA_PLAN.md): CDNA4
// flash-attention FORWARD, v0 correctness build. Not wired into any model
// or test suite — reachable only through tools/fa_probe.py via the
// `_core.fa_fwd` binding. The performance idioms (T12-T16: swapped QK^T,
// register-staged V, permlane softmax, defer-max) come after numerics are
// proven; this version optimizes for being verifiably correct.
//
// Tiling: one WAVE per 16 query rows; K/V consumed in 32-row tiles.
//   S(16x32) = Q(16x128) K_tile(32x128)^T  — 4x2 mfma_f32_16x16x32_bf16
//   online softmax per row (16-lane-group shuffle reductions)
//   O(16x128) += P(16x32) V_tile(32x128)   — P restaged via LDS into the
//     A-fragment layout; V read from LDS (element reads, v0)
// Fragment maps (guide §3, by analogy with the documented 16x16x4 f32):
//   A: row i = lane&15, k = (lane>>4)*8 + j      (8 bf16 / lane)
//   B: col j = lane&15, k = (lane>>4)*8 + j
//   C/D: col = lane&15, row = (lane>>4)*4 + reg  (4 f32 / lane)
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <algorithm>

#include "kernels.h"

namespace axon {

namespace {

using bf16 = __hip_bfloat16;
using frag_ab = __attribute__((ext_vector_type(8))) short;   // 8 bf16
using frag_cd = __attribute__((ext_vector_type(4))) float;   // 4 f32

constexpr int kD = 128;        // head dim (Llama-3)
constexpr int kQTile = 16;     // q rows per wave
constexpr int kKTile = 32;     // k/v rows per iteration
constexpr int kWavesFA = 4;    // waves per block

__device__ __forceinline__ float fa_group_max(float v) {
  // max over the 16-lane group (rows live in lane>>4 groups)
  for (int off = 1; off < 16; off <<= 1)
    v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

__device__ __forceinline__ float fa_group_sum(float v) {
  for (int off = 1; off < 16; off <<= 1)
    v += __shfl_xor(v, off, 64);
  return v;
}

// One block: kWavesFA waves, each owning 16 q rows of one (batch, head).
// grid: (ceil(S / (16*4)), HQ, B). K/V head = q_head / (HQ/HK).
__global__ void fa_fwd_kernel(const bf16* __restrict__ q,
                              const bf16* __restrict__ k,
                              const bf16* __restrict__ v,
                              bf16* __restrict__ o, float* __restrict__ lse,
                              int B, int HQ, int HK, int S, float scale,
                              int causal) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int qt = blockIdx.x * kWavesFA + wave;  // q-tile index
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = h / (HQ / HK);
  const int q0 = qt * kQTile;
  if (q0 >= S) return;

  const int row = lane & 15;        // A-frag row / C-frag col
  const int grp = lane >> 4;        // 0..3

  // LDS: per wave, a 32x128 bf16 V tile and a 16x32 bf16 P tile.
  __shared__ bf16 v_lds[kWavesFA][kKTile][kD];
  __shared__ bf16 p_lds[kWavesFA][kQTile][kKTile];

  const int64_t qbase = ((static_cast<int64_t>(b) * HQ + h) * S);
  const int64_t kbase = ((static_cast<int64_t>(b) * HK + hk) * S);

  // Q fragments: chunk c covers k-dims [c*32, c*32+32); this lane holds
  // 8 contiguous dims starting at c*32 + grp*8 of row q0+row.
  frag_ab qf[kD / 32];
  const int qrow = q0 + row < S ? q0 + row : S - 1;  // clamp tail reads
#pragma unroll
  for (int c = 0; c < kD / 32; ++c)
    qf[c] = *reinterpret_cast<const frag_ab*>(
        q + (qbase + qrow) * kD + c * 32 + grp * 8);

  // Accumulators: O as 8 column-chunks of 16 cols (C-layout), row stats.
  frag_cd oacc[kD / 16];
#pragma unroll
  for (int cc = 0; cc < kD / 16; ++cc) oacc[cc] = frag_cd{0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }

  const int kend = causal ? (q0 + kQTile < S ? q0 + kQTile : S) : S;
  for (int k0 = 0; k0 < kend; k0 += kKTile) {
    const int ktail = kKTile < S - k0 ? kKTile : S - k0;

    // Stage V tile to LDS (each lane copies rows round-robin, 16B moves).
    for (int idx = lane; idx < kKTile * (kD / 8); idx += 64) {
      int vr = idx / (kD / 8), vc = (idx % (kD / 8)) * 8;
      if (vr < ktail)
        *reinterpret_cast<float4*>(&v_lds[wave][vr][vc]) =
            *reinterpret_cast<const float4*>(v + (kbase + k0 + vr) * kD + vc);
      else
        *reinterpret_cast<float4*>(&v_lds[wave][vr][vc]) =
            float4{0.f, 0.f, 0.f, 0.f};
    }
    // Cross-LANE (same-wave) LDS handoff: fence explicitly. A block-wide
    // __syncthreads is NOT usable here — waves have divergent k-loop trip
    // counts under the causal bound, so a barrier would deadlock.
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // S tile: two 16x16 halves (k columns [0,16) and [16,32)).
    frag_cd sacc[2] = {frag_cd{0.f, 0.f, 0.f, 0.f}, frag_cd{0.f, 0.f, 0.f, 0.f}};
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int kcol = k0 + half * 16 + row;  // B-frag col = K row index
      const int krow = kcol < S ? kcol : S - 1;
#pragma unroll
      for (int c = 0; c < kD / 32; ++c) {
        frag_ab kf = *reinterpret_cast<const frag_ab*>(
            k + (kbase + krow) * kD + c * 32 + grp * 8);
        sacc[half] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[c], kf,
                                                             sacc[half], 0, 0, 0);
      }
    }

    // Online softmax on the 16x32 score tile.
    float p_val[2][4];
    float tile_max[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) tile_max[r] = -1e30f;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int qi = q0 + grp * 4 + r;        // this value's q row
        int ki = k0 + half * 16 + row;    // this value's k col
        float s = sacc[half][r] * scale;
        if (ki >= S || qi >= S || (causal && ki > qi)) s = -1e30f;
        p_val[half][r] = s;
        tile_max[r] = fmaxf(tile_max[r], s);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) tile_max[r] = fa_group_max(tile_max[r]);

    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m_new = fmaxf(m_run[r], tile_max[r]);
      alpha[r] = __expf(m_run[r] - m_new);
      m_run[r] = m_new;
    }
    float row_sum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int half = 0; half < 2; ++half)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = __expf(p_val[half][r] - m_run[r]);
        p_val[half][r] = p;
        row_sum[r] += p;
      }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      l_run[r] = l_run[r] * alpha[r] + fa_group_sum(row_sum[r]);
    }

    // Restage P (C layout: row = grp*4+r, col = half*16 + row-lane) into
    // LDS, then reload in A-fragment layout.
#pragma unroll
    for (int half = 0; half < 2; ++half)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_lds[wave][grp * 4 + r][half * 16 + row] =
            __float2bfloat16(p_val[half][r]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    frag_ab pf;  // A-frag of P: row = lane&15, k = grp*8 + j
#pragma unroll
    for (int j = 0; j < 8; ++j)
      reinterpret_cast<bf16*>(&pf)[j] = p_lds[wave][row][grp * 8 + j];

    // O chunks: rescale by alpha, then accumulate P @ V.
#pragma unroll
    for (int cc = 0; cc < kD / 16; ++cc) {
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[cc][r] *= alpha[r];
      frag_ab vf;  // B-frag of V chunk cc: col = lane&15, k = grp*8 + j
#pragma unroll
      for (int j = 0; j < 8; ++j)
        reinterpret_cast<bf16*>(&vf)[j] =
            v_lds[wave][grp * 8 + j][cc * 16 + row];
      oacc[cc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, oacc[cc],
                                                         0, 0, 0);
    }
  }

  // Epilogue: O /= l, store; L = m + log(l).
#pragma unroll
  for (int cc = 0; cc < kD / 16; ++cc)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qi = q0 + grp * 4 + r;
      if (qi < S) {
        float denom = l_run[r] > 0.f ? l_run[r] : 1.f;
        o[(qbase + qi) * kD + cc * 16 + row] =
            __float2bfloat16(oacc[cc][r] / denom);
      }
    }
  if (lse != nullptr && row == 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qi = q0 + grp * 4 + r;
      if (qi < S)
        lse[(static_cast<int64_t>(b) * HQ + h) * S + qi] =
            m_run[r] + __logf(l_run[r] > 0.f ? l_run[r] : 1.f);
    }
  }
}

}  // namespace

void launch_fa_fwd(const void* q, const void* k, const void* v, void* o,
                   float* lse, int B, int HQ, int HK, int S, int D,
                   bool causal, hipStream_t stream) {
  if (D != kD) return;  // v0: Llama head dim only (wrapper validates)
  float scale = 1.f / sqrtf(static_cast<float>(D));
  dim3 grid((S + kQTile * kWavesFA - 1) / (kQTile * kWavesFA), HQ, B);
  hipLaunchKernelGGL(fa_fwd_kernel, grid, 256, 0, stream,
                     static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                     static_cast<const bf16*>(v), static_cast<bf16*>(o), lse,
                     B, HQ, HK, S, scale, causal ? 1 : 0);
}

}  // namespace axon
