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

// ---------------------------------------------------------------- backward
// EXPERIMENTAL v0 (probe-only, never run in any model/test path until
// round 2 validates it through tools/fa_probe.py --check-bwd). FA-2 style:
// P recomputed from saved logsumexp; delta = rowsum(dO*O) precomputed.
//
//   dkdv kernel: one wave per 16-row K/V tile; inner loop over the
//     (causal-reachable) 32-row q-tiles and, inside, over the q-heads
//     sharing this kv head (GQA accumulates into the same dK/dV, so no
//     atomics and no workspace).
//       ST(16k x 32q) = K Q^T          P^T = exp(ST*scale - lse[col])
//       dV += P^T dO                   (P^T restaged via LDS into A-frag)
//       S(32q x 16k half-wise), dP = dO V^T, dS = P (dP - delta[row])
//       dK += dS^T Q                   (dS^T restaged via LDS)
//   dq kernel: one wave per 16-row q-tile; loop over k tiles:
//       S, P, dP = dO V^T, dS = P (dP - delta); dQ += dS K.

__global__ void fa_delta_kernel(const bf16* __restrict__ o,
                                const bf16* __restrict__ dout,
                                float* __restrict__ delta, int64_t rows) {
  // delta[r] = sum_d o[r][d] * dout[r][d]; one wave per row, D=128.
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int64_t r = static_cast<int64_t>(blockIdx.x) * kWavesFA + wave;
  if (r >= rows) return;
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    int d = lane + 64 * j;
    acc += __bfloat162float(o[r * kD + d]) * __bfloat162float(dout[r * kD + d]);
  }
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_xor(acc, off, 64);
  if (lane == 0) delta[r] = acc;
}

// Shared helper: S(16q x 16k) C-frag from Q rows [q0+ ] and K rows [kk0+ ].
__device__ __forceinline__ frag_cd fa_score_tile(
    const bf16* __restrict__ q, const bf16* __restrict__ k, int64_t qbase,
    int64_t kbase, int q0, int kk0, int S, int lane) {
  const int row = lane & 15, grp = lane >> 4;
  frag_cd acc = frag_cd{0.f, 0.f, 0.f, 0.f};
  const int qr = q0 + row < S ? q0 + row : S - 1;
  const int kr = kk0 + row < S ? kk0 + row : S - 1;
#pragma unroll
  for (int c = 0; c < kD / 32; ++c) {
    frag_ab qf = *reinterpret_cast<const frag_ab*>(
        q + (qbase + qr) * kD + c * 32 + grp * 8);
    frag_ab kf = *reinterpret_cast<const frag_ab*>(
        k + (kbase + kr) * kD + c * 32 + grp * 8);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf, kf, acc, 0, 0, 0);
  }
  return acc;
}

__global__ void fa_bwd_dkdv_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dk, bf16* __restrict__ dv, int B, int HQ, int HK,
    int S, float scale, int causal) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int kt = blockIdx.x * kWavesFA + wave;  // 16-row k tile
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int kt0 = kt * kQTile;
  if (kt0 >= S) return;
  const int row = lane & 15, grp = lane >> 4;
  const int gqa = HQ / HK;

  __shared__ bf16 stage_lds[kWavesFA][kQTile][kKTile];  // P^T / dS^T restage

  const int64_t kbase = (static_cast<int64_t>(b) * HK + hkv) * S;

  frag_cd dkacc[kD / 16], dvacc[kD / 16];
#pragma unroll
  for (int cc = 0; cc < kD / 16; ++cc) {
    dkacc[cc] = frag_cd{0.f, 0.f, 0.f, 0.f};
    dvacc[cc] = frag_cd{0.f, 0.f, 0.f, 0.f};
  }

  for (int g = 0; g < gqa; ++g) {
    const int h = hkv * gqa + g;
    const int64_t qbase = (static_cast<int64_t>(b) * HQ + h) * S;
    const float* lse_h = lse + (static_cast<int64_t>(b) * HQ + h) * S;
    const float* dlt_h = delta + (static_cast<int64_t>(b) * HQ + h) * S;

    const int q_start = causal ? (kt0 / kKTile) * kKTile : 0;
    for (int q0 = q_start; q0 < S; q0 += kKTile) {
      // ---- P^T (16k x 32q) for dV: ST = K Q^T, exp(ST*scale - lse[q]).
      float pT[2][4];
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        frag_cd st = fa_score_tile(k, q, kbase, qbase, kt0, q0 + half * 16,
                                   S, lane);
        // ST C-frag: row = k index (grp*4+r), col = q index (lane&15)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int ki = kt0 + grp * 4 + r;
          int qi = q0 + half * 16 + row;
          float sval = st[r] * scale;
          bool dead = ki >= S || qi >= S || (causal && ki > qi);
          pT[half][r] = dead ? 0.f : __expf(sval - lse_h[qi < S ? qi : 0]);
        }
      }
      // restage P^T into A-frag layout (rows = k, kdim = q, 32 wide)
#pragma unroll
      for (int half = 0; half < 2; ++half)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          stage_lds[wave][grp * 4 + r][half * 16 + row] =
              __float2bfloat16(pT[half][r]);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      frag_ab ptf;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        reinterpret_cast<bf16*>(&ptf)[j] = stage_lds[wave][row][grp * 8 + j];

      // dV += P^T dO : B-frag of dO chunk cc -> strided global reads
      //   dO[q0 + grp*8 + j][cc*16 + row]
#pragma unroll
      for (int cc = 0; cc < kD / 16; ++cc) {
        frag_ab dof;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int qi = q0 + grp * 8 + j;
          reinterpret_cast<bf16*>(&dof)[j] =
              qi < S ? dout[(qbase + qi) * kD + cc * 16 + row]
                     : __float2bfloat16(0.f);
        }
        dvacc[cc] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(ptf, dof, dvacc[cc], 0, 0, 0);
      }

      // ---- dS (32q x 16k), computed half-wise in (q,k) orientation.
      float dsv[2][4];
#pragma unroll
      for (int qh = 0; qh < 2; ++qh) {
        // S half: rows q, cols k
        frag_cd sq = fa_score_tile(q, k, qbase, kbase, q0 + qh * 16, kt0,
                                   S, lane);
        // dP half: dO(16q x 128) V^T -> A = dO chunks, B = V rows
        frag_cd dp = frag_cd{0.f, 0.f, 0.f, 0.f};
        const int qr = q0 + qh * 16 + row < S ? q0 + qh * 16 + row : S - 1;
        const int vr = kt0 + row < S ? kt0 + row : S - 1;
#pragma unroll
        for (int c = 0; c < kD / 32; ++c) {
          frag_ab dof = *reinterpret_cast<const frag_ab*>(
              dout + (qbase + qr) * kD + c * 32 + grp * 8);
          frag_ab vf = *reinterpret_cast<const frag_ab*>(
              v + (kbase + vr) * kD + c * 32 + grp * 8);
          dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dof, vf, dp, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int qi = q0 + qh * 16 + grp * 4 + r;
          int ki = kt0 + row;
          bool dead = ki >= S || qi >= S || (causal && ki > qi);
          float pv = dead ? 0.f
                          : __expf(sq[r] * scale - lse_h[qi < S ? qi : 0]);
          dsv[qh][r] = pv * (dp[r] - dlt_h[qi < S ? qi : 0]) * scale;
        }
      }
      // restage dS^T: A-frag rows = k (lane&15), kdim = q
#pragma unroll
      for (int qh = 0; qh < 2; ++qh)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          stage_lds[wave][row][qh * 16 + grp * 4 + r] =
              __float2bfloat16(dsv[qh][r]);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      frag_ab dstf;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        reinterpret_cast<bf16*>(&dstf)[j] = stage_lds[wave][row][grp * 8 + j];

      // dK += dS^T Q : B-frag = Q[q0 + grp*8 + j][cc*16 + row]
#pragma unroll
      for (int cc = 0; cc < kD / 16; ++cc) {
        frag_ab qf2;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int qi = q0 + grp * 8 + j;
          reinterpret_cast<bf16*>(&qf2)[j] =
              qi < S ? q[(qbase + qi) * kD + cc * 16 + row]
                     : __float2bfloat16(0.f);
        }
        dkacc[cc] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(dstf, qf2, dkacc[cc], 0, 0, 0);
      }
    }
  }

  // store dK/dV (C layout: col = lane&15, row = grp*4 + r)
#pragma unroll
  for (int cc = 0; cc < kD / 16; ++cc)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int ki = kt0 + grp * 4 + r;
      if (ki < S) {
        dk[(kbase + ki) * kD + cc * 16 + row] = __float2bfloat16(dkacc[cc][r]);
        dv[(kbase + ki) * kD + cc * 16 + row] = __float2bfloat16(dvacc[cc][r]);
      }
    }
}

__global__ void fa_bwd_dq_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dq, int B, int HQ, int HK, int S, float scale,
    int causal) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int qt = blockIdx.x * kWavesFA + wave;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = h / (HQ / HK);
  const int q0 = qt * kQTile;
  if (q0 >= S) return;
  const int row = lane & 15, grp = lane >> 4;

  __shared__ bf16 ds_lds[kWavesFA][kQTile][kKTile];

  const int64_t qbase = (static_cast<int64_t>(b) * HQ + h) * S;
  const int64_t kbase = (static_cast<int64_t>(b) * HK + hk) * S;
  const float* lse_h = lse + (static_cast<int64_t>(b) * HQ + h) * S;
  const float* dlt_h = delta + (static_cast<int64_t>(b) * HQ + h) * S;

  frag_cd dqacc[kD / 16];
#pragma unroll
  for (int cc = 0; cc < kD / 16; ++cc) dqacc[cc] = frag_cd{0.f, 0.f, 0.f, 0.f};

  const int kend = causal ? (q0 + kQTile < S ? q0 + kQTile : S) : S;
  for (int k0 = 0; k0 < kend; k0 += kKTile) {
    // dS (16q x 32k) half-wise in (q,k) orientation.
    float dsv[2][4];
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      frag_cd sq = fa_score_tile(q, k, qbase, kbase, q0, k0 + kh * 16, S, lane);
      frag_cd dp = frag_cd{0.f, 0.f, 0.f, 0.f};
      const int qr = q0 + row < S ? q0 + row : S - 1;
      const int vr = k0 + kh * 16 + row < S ? k0 + kh * 16 + row : S - 1;
#pragma unroll
      for (int c = 0; c < kD / 32; ++c) {
        frag_ab dof = *reinterpret_cast<const frag_ab*>(
            dout + (qbase + qr) * kD + c * 32 + grp * 8);
        frag_ab vf = *reinterpret_cast<const frag_ab*>(
            v + (kbase + vr) * kD + c * 32 + grp * 8);
        dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dof, vf, dp, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int qi = q0 + grp * 4 + r;
        int ki = k0 + kh * 16 + row;
        bool dead = ki >= S || qi >= S || (causal && ki > qi);
        float pv = dead ? 0.f
                        : __expf(sq[r] * scale - lse_h[qi < S ? qi : 0]);
        dsv[kh][r] = pv * (dp[r] - dlt_h[qi < S ? qi : 0]) * scale;
      }
    }
    // restage dS into A-frag layout (rows = q, kdim = k)
#pragma unroll
    for (int kh = 0; kh < 2; ++kh)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        ds_lds[wave][grp * 4 + r][kh * 16 + row] = __float2bfloat16(dsv[kh][r]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    frag_ab dsf;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      reinterpret_cast<bf16*>(&dsf)[j] = ds_lds[wave][row][grp * 8 + j];

    // dQ += dS K : B-frag = K[k0 + grp*8 + j][cc*16 + row]
#pragma unroll
    for (int cc = 0; cc < kD / 16; ++cc) {
      frag_ab kf2;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int ki = k0 + grp * 8 + j;
        reinterpret_cast<bf16*>(&kf2)[j] =
            ki < S ? k[(kbase + ki) * kD + cc * 16 + row]
                   : __float2bfloat16(0.f);
      }
      dqacc[cc] =
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, kf2, dqacc[cc], 0, 0, 0);
    }
  }

#pragma unroll
  for (int cc = 0; cc < kD / 16; ++cc)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qi = q0 + grp * 4 + r;
      if (qi < S)
        dq[(qbase + qi) * kD + cc * 16 + row] = __float2bfloat16(dqacc[cc][r]);
    }
}

}  // namespace

void launch_fa_bwd(const void* q, const void* k, const void* v,
                   const void* dout, const void* o, const float* lse,
                   float* delta, void* dq, void* dk, void* dv, int B, int HQ,
                   int HK, int S, int D, bool causal, hipStream_t stream) {
  if (D != kD) return;  // v0: head dim 128 only (wrapper validates)
  float scale = 1.f / sqrtf(static_cast<float>(D));
  int64_t rows = static_cast<int64_t>(B) * HQ * S;
  hipLaunchKernelGGL(fa_delta_kernel,
                     static_cast<unsigned>((rows + kWavesFA - 1) / kWavesFA),
                     256, 0, stream, static_cast<const bf16*>(o),
                     static_cast<const bf16*>(dout), delta, rows);
  dim3 gkv((S + kQTile * kWavesFA - 1) / (kQTile * kWavesFA), HK, B);
  hipLaunchKernelGGL(fa_bwd_dkdv_kernel, gkv, 256, 0, stream,
                     static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                     static_cast<const bf16*>(v),
                     static_cast<const bf16*>(dout), lse, delta,
                     static_cast<bf16*>(dk), static_cast<bf16*>(dv), B, HQ,
                     HK, S, scale, causal ? 1 : 0);
  dim3 gq((S + kQTile * kWavesFA - 1) / (kQTile * kWavesFA), HQ, B);
  hipLaunchKernelGGL(fa_bwd_dq_kernel, gq, 256, 0, stream,
                     static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                     static_cast<const bf16*>(v),
                     static_cast<const bf16*>(dout), lse, delta,
                     static_cast<bf16*>(dq), B, HQ, HK, S, scale,
                     causal ? 1 : 0);
}

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
