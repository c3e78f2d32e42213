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

// ------------------------------------------------------------- forward v2
// Guide-structure forward (cdna_hip_programming.md T12/T13 + §5 anatomy):
// 8 waves per block over 256 q rows; per wave a 32q x 32k tile via
// mfma_f32_32x32x16_bf16 with SWAPPED QK^T (A=K, B=Q) so each lane owns
// one q row's scores — softmax is per-lane + one permlane32_swap
// cross-half merge, and P is repacked to the PV A-fragment entirely in
// registers (cvt_pk pairs + 2 permlane32_swap per 16-key slice); K is
// staged row-major in LDS, V transposed ([d][key]) so the PV B-fragment
// is one ds_read_b128 of 8 consecutive keys; defer-max (T13, textbook
// order: the rescale decision covers every P of its own tile) skips the
// O-wide rescale unless a row max grows by > THR.
//
// Fragment maps (mfma_f32_32x32x16_bf16, guide §3):
//   A: row = lane&31, k = (lane>>5)*8 + j          (8 bf16 / lane)
//   B: col = lane&31, k = (lane>>5)*8 + j
//   C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)

using frag16 = __attribute__((ext_vector_type(16))) float;  // 32x32 C/D

constexpr int kBq2 = 256;   // q rows per block (8-wave variant)
constexpr int kW2 = 8;      // max waves per block (red_lds sizing)
constexpr int kBr2 = 32;    // q rows per wave
constexpr int kBc2 = 32;    // keys per tile
constexpr int kKPad = 8;    // bf16 pad per K-LDS row
constexpr int kVPad = 8;    // bf16 pad per V-LDS row
constexpr float kFaThr = 8.f;  // defer-max threshold (T13)

__device__ __forceinline__ uint32_t pack_bf16(float lo, float hi) {
  // compiler emits v_cvt_pk_bf16_f32 for this pattern (guide T12 note)
  union { __hip_bfloat16 b; unsigned short u; } a{__float2bfloat16(lo)},
      c{__float2bfloat16(hi)};
  return static_cast<uint32_t>(a.u) | (static_cast<uint32_t>(c.u) << 16);
}

__device__ __forceinline__ float xhalf_swap_f(float x) {
  // value held by lane l^32 (cross-half partner)
  auto r = __builtin_amdgcn_permlane32_swap(__float_as_int(x),
                                            __float_as_int(x), false, false);
  return __int_as_float(threadIdx.x % 64 < 32 ? r[1] : r[0]);
}

// Repack a 32x32 C-fragment column (16 f32, C rows (r&3)+8*(r>>2)+4*half)
// into the two A-fragments of the NEXT mfma whose M-dim equals this C's
// col-dim: pack adjacent rows to bf16 pairs, then 2 permlane32_swap per
// 16-slice moves each half's missing quads across (guide T12).
__device__ __forceinline__ void packs_to_afrags(const uint32_t pk[8],
                                                frag_ab* a1, frag_ab* a2) {
  auto r0 = __builtin_amdgcn_permlane32_swap(pk[0], pk[2], false, false);
  auto r1 = __builtin_amdgcn_permlane32_swap(pk[1], pk[3], false, false);
  auto r2 = __builtin_amdgcn_permlane32_swap(pk[4], pk[6], false, false);
  auto r3 = __builtin_amdgcn_permlane32_swap(pk[5], pk[7], false, false);
  uint32_t* u1 = reinterpret_cast<uint32_t*>(a1);
  uint32_t* u2 = reinterpret_cast<uint32_t*>(a2);
  u1[0] = r0[0]; u1[1] = r1[0]; u1[2] = r0[1]; u1[3] = r1[1];
  u2[0] = r2[0]; u2[1] = r3[0]; u2[2] = r2[1]; u2[3] = r3[1];
}

__device__ __forceinline__ void p16_to_afrags(const float p[16], frag_ab* a1,
                                              frag_ab* a2) {
  uint32_t pk[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) pk[i] = pack_bf16(p[2 * i], p[2 * i + 1]);
  auto r0 = __builtin_amdgcn_permlane32_swap(pk[0], pk[2], false, false);
  auto r1 = __builtin_amdgcn_permlane32_swap(pk[1], pk[3], false, false);
  auto r2 = __builtin_amdgcn_permlane32_swap(pk[4], pk[6], false, false);
  auto r3 = __builtin_amdgcn_permlane32_swap(pk[5], pk[7], false, false);
  uint32_t* u1 = reinterpret_cast<uint32_t*>(a1);
  uint32_t* u2 = reinterpret_cast<uint32_t*>(a2);
  u1[0] = r0[0]; u1[1] = r1[0]; u1[2] = r0[1]; u1[3] = r1[1];
  u2[0] = r2[0]; u2[1] = r3[0]; u2[2] = r2[1]; u2[3] = r3[1];
}

template <int NW>  // waves per block (8 -> 256 q rows, 4 -> 128)
__global__ __launch_bounds__(NW * 64, 1) void fa_fwd_v2_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ o, float* __restrict__ lse,
    int B, int HQ, int HK, int S, float scale, int causal) {
  constexpr int kThreads = NW * 64;
  constexpr int kRep = (kBc2 * kD / 8) / kThreads;  // staged chunks/thread
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int idx = threadIdx.x;
  const int col = lane & 31;             // q row (S^T C col) / d col (O)
  const int half = lane >> 5;
  const int qb = blockIdx.x * (NW * kBr2);
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = h / (HQ / HK);
  const int q0w = qb + wave * kBr2;

  __shared__ bf16 k_lds[2][kBc2][kD + kKPad];     // row-major K tiles (dbuf)
  __shared__ bf16 v_lds[2][kD][kBc2 + kVPad];     // transposed V tiles (dbuf)
  __shared__ float red_lds[kW2][kBr2];            // per-row alpha/l exchange

  const int64_t qbase = (static_cast<int64_t>(b) * HQ + h) * S;
  const int64_t kbase = (static_cast<int64_t>(b) * HK + hk) * S;

  // Q fragments, loaded once: B-frag col = q row (col), k-dim = d.
  const int qi = q0w + col;
  const int qrow = qi < S ? qi : S - 1;
  frag_ab qf[kD / 16];
#pragma unroll
  for (int c = 0; c < kD / 16; ++c)
    qf[c] = *reinterpret_cast<const frag_ab*>(
        q + (qbase + qrow) * kD + c * 16 + half * 8);

  frag16 oacc[kD / 32];
#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) oacc[cc][r] = 0.f;
  float m_run = -1e30f, l_run = 0.f;  // row stats for q row `qi` (raw scores)

  // Staging maps (kRep chunks per thread; chunk id cid = idx + r*kThreads):
  //   K: row cid>>4, d (cid&15)*8 — coalesced loads + b128 LDS writes.
  //   V: key cid&31, d (cid>>5)*8 — keeps the transposing ds_write_b16s
  //      spread across banks (see round-2 notes).
  const int kend = causal ? (qb + NW * kBr2 < S ? qb + NW * kBr2 : S) : S;
  const int ntiles = (kend + kBc2 - 1) / kBc2;

  // T14 issue-early/write-late with double-buffered LDS: tile t+1's global
  // loads are issued before tile t's compute, the LDS write lands after it,
  // and ONE barrier per tile publishes the next buffer.
  auto load_tile = [&](int k0, float4* kv4, float4* vv4) {
#pragma unroll
    for (int r = 0; r < kRep; ++r) {
      int cid = idx + r * kThreads;
      int kr = cid >> 4, kdc = (cid & 15) * 8;
      int vr = cid & 31, vdc = (cid >> 5) * 8;
      kv4[r] = float4{0.f, 0.f, 0.f, 0.f};
      vv4[r] = float4{0.f, 0.f, 0.f, 0.f};
      if (k0 + kr < S)
        kv4[r] =
            *reinterpret_cast<const float4*>(k + (kbase + k0 + kr) * kD + kdc);
      if (k0 + vr < S)
        vv4[r] =
            *reinterpret_cast<const float4*>(v + (kbase + k0 + vr) * kD + vdc);
    }
  };
  auto write_tile = [&](int buf, const float4* kv4, const float4* vv4) {
#pragma unroll
    for (int r = 0; r < kRep; ++r) {
      int cid = idx + r * kThreads;
      int kr = cid >> 4, kdc = (cid & 15) * 8;
      int vr = cid & 31, vdc = (cid >> 5) * 8;
      *reinterpret_cast<float4*>(&k_lds[buf][kr][kdc]) = kv4[r];
      const bf16* vb = reinterpret_cast<const bf16*>(&vv4[r]);
#pragma unroll
      for (int j = 0; j < 8; ++j) v_lds[buf][vdc + j][vr] = vb[j];
    }
  };

  float4 kv4[kRep], vv4[kRep];
  load_tile(0, kv4, vv4);
  write_tile(0, kv4, vv4);
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    const int k0 = t * kBc2;
    const int buf = t & 1;
    if (t + 1 < ntiles) load_tile(k0 + kBc2, kv4, vv4);

    // ---- S^T = K Q^T (A = K rows, B = Q cols) ----
    // (One accumulator chain: a split even/odd two-chain variant measured
    // 356 vs 411 TF — the +16 VGPRs dropped occupancy 3 -> 2.)
    frag16 st;
#pragma unroll
    for (int r = 0; r < 16; ++r) st[r] = 0.f;
#pragma unroll
    for (int c = 0; c < kD / 16; ++c) {
      frag_ab kf = *reinterpret_cast<const frag_ab*>(
          &k_lds[buf][col][c * 16 + half * 8]);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[c], st, 0, 0, 0);
    }

    // ---- masking + tile max (raw-score domain) ----
    const bool full_tile = causal ? (k0 + kBc2 - 1 <= q0w) : (k0 + kBc2 <= S);
    if (!full_tile) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int key = k0 + (r & 3) + 8 * (r >> 2) + 4 * half;
        if (key >= S || (causal && key > qi)) st[r] = -1e30f;
      }
    }
    float tmax = st[0];
#pragma unroll
    for (int r = 1; r < 16; ++r) tmax = fmaxf(tmax, st[r]);
    tmax = fmaxf(tmax, xhalf_swap_f(tmax));

    // ---- defer-max online softmax (textbook order) ----
    float alpha = 1.f;
    bool rescale =
        !__all(scale * (tmax - m_run) <= kFaThr) || t == 0;
    if (rescale) {
      float m_new = fmaxf(m_run, tmax);
      alpha = __expf(scale * (m_run - m_new));
      m_run = m_new;
      // O *= alpha: alpha is per-q-row; O holds 16 q rows per lane —
      // redistribute via this wave's red_lds row (no block barrier:
      // same-wave LDS handoff fenced by lgkmcnt).
      red_lds[wave][col] = alpha;
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      float a16[16];
#pragma unroll
      for (int r = 0; r < 16; ++r)
        a16[r] = red_lds[wave][(r & 3) + 8 * (r >> 2) + 4 * half];
#pragma unroll
      for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
        for (int r = 0; r < 16; ++r) oacc[cc][r] *= a16[r];
    }
    float p[16], rsum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      p[r] = __expf(scale * (st[r] - m_run));
      rsum += p[r];
    }
    l_run = l_run * alpha + rsum + xhalf_swap_f(rsum);

    // ---- pack P to PV A-fragments (in registers, T12) ----
    frag_ab pa1, pa2;
    p16_to_afrags(p, &pa1, &pa2);

    // ---- O += P V (B-frag: 8 consecutive keys from transposed V) ----
    // (write_tile placed AFTER PV: hoisting it before the cluster measured
    // 384 vs 411 TF — hipcc's vmcnt(0) at the ds_write drains the t+1
    // loads mid-compute, the failure mode the guide's T14 notes warn of.
    // s_setprio around the MFMA clusters also measured -5% here — without
    // a T3-style phase split it only perturbs the scheduler.)
#pragma unroll
    for (int cc = 0; cc < kD / 32; ++cc) {
      frag_ab v1 = *reinterpret_cast<const frag_ab*>(
          &v_lds[buf][cc * 32 + col][half * 8]);
      frag_ab v2 = *reinterpret_cast<const frag_ab*>(
          &v_lds[buf][cc * 32 + col][16 + half * 8]);
      oacc[cc] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa1, v1, oacc[cc],
                                                         0, 0, 0);
      oacc[cc] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa2, v2, oacc[cc],
                                                         0, 0, 0);
    }

    if (t + 1 < ntiles) write_tile(buf ^ 1, kv4, vv4);
    __syncthreads();  // publish buf^1; also fences buf re-use next round
  }

  // ---- epilogue: O /= l (per-row, via red_lds), store O + LSE ----
  red_lds[wave][col] = l_run > 0.f ? 1.f / l_run : 0.f;
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  float li16[16];
#pragma unroll
  for (int r = 0; r < 16; ++r)
    li16[r] = red_lds[wave][(r & 3) + 8 * (r >> 2) + 4 * half];
#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int qrow_r = q0w + (r & 3) + 8 * (r >> 2) + 4 * half;
      if (qrow_r < S)
        o[(qbase + qrow_r) * kD + cc * 32 + col] =
            __float2bfloat16(oacc[cc][r] * li16[r]);
    }
  if (lse != nullptr && half == 0 && qi < S)
    lse[(static_cast<int64_t>(b) * HQ + h) * S + qi] =
        scale * m_run + __logf(l_run > 0.f ? l_run : 1.f);
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

// ------------------------------------------------------------ backward v2
// Same structural idioms as the v2 forward: 32x32x16 MFMA tiles, operands
// staged once per tile in LDS (row-major for A/B fragments whose M/N dim
// is the lane index; transposed [d][row] images for the B fragments of
// the accumulating GEMMs), P/dS repacked to A-fragments in registers via
// p16_to_afrags. FA-2 recompute from the forward's logsumexp; GQA
// accumulated inside the k-tile loop (no atomics, no workspace).
//
// dkdv: 4 waves over 128 k rows of one kv head. Per (q-head, 32-q tile):
//   S(32q x 32k)  = Q K^T      A = Q (LDS row),   B = K (LDS row)
//   dP(32q x 32k) = dO V^T     A = dO (LDS row),  B = V (LDS row)
//   P = exp(scale*S - lse); dS = P (dP - delta) scale   [in-register]
//   dV += P^T dO   A = repack(P),  B = dO ([d][q] image)
//   dK += dS^T Q   A = repack(dS), B = Q  ([d][q] image)
// dq: 4 waves over 128 q rows of one q head. Per 32-k tile:
//   S^T = K Q^T, dP^T = V dO^T (Q/dO persistent registers), dS^T
//   in-register (lse/delta are per-lane scalars), dQ += dS K with K in a
//   transposed [d][k] image.

constexpr int kBK = 128;  // k rows per dkdv block / q rows per dq block
constexpr int kW4 = 4;

__global__ __launch_bounds__(256, 2) void fa_bwd_dkdv_v2_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dk, bf16* __restrict__ dv, int B, int HQ, int HK,
    int S, float scale, int causal) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int idx = threadIdx.x;  // 0..255
  const int col = lane & 31;
  const int half = lane >> 5;
  const int kb0 = blockIdx.x * kBK;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int gqa = HQ / HK;
  const int kw0 = kb0 + wave * 32;  // this wave's k-tile base

  __shared__ bf16 q_row[2][32][kD + kKPad];
  __shared__ bf16 q_tr[2][kD][32 + kVPad];
  __shared__ bf16 do_row[2][32][kD + kKPad];
  __shared__ bf16 do_tr[2][kD][32 + kVPad];
  __shared__ float lse_lds[2][32], dlt_lds[2][32];

  const int64_t kbase = (static_cast<int64_t>(b) * HK + hkv) * S;

  // This wave's K/V rows live in registers for the whole kernel: the lane
  // layout of a B fragment (col = lane&31, k-dim = (lane>>5)*8+j) matches a
  // per-lane row load, so no LDS image is needed (keeps LDS under 40 KB ->
  // 2 blocks/CU instead of the 108 KB 1-block version).
  const int kc = kw0 + col;
  const int krow_g = kc < S ? kc : S - 1;
  frag_ab kreg[kD / 16], vreg[kD / 16];
#pragma unroll
  for (int c = 0; c < kD / 16; ++c) {
    kreg[c] = *reinterpret_cast<const frag_ab*>(
        k + (kbase + krow_g) * kD + c * 16 + half * 8);
    vreg[c] = *reinterpret_cast<const frag_ab*>(
        v + (kbase + krow_g) * kD + c * 16 + half * 8);
  }

  frag16 dvacc[kD / 32], dkacc[kD / 32];
#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      dvacc[cc][r] = 0.f;
      dkacc[cc][r] = 0.f;
    }

  const int qsr = idx & 31, qsd = (idx >> 5) * 16;  // staging map
  const int q_lo = causal ? kb0 : 0;
  const int nq = (S - q_lo + 31) / 32;
  const int niter = gqa * nq;

  // Flattened (q-head, q-tile) loop with T14 double-buffered staging.
  // Half-prefetch (T14): the FIRST 16B of each row is held across the
  // compute (hides most of the HBM latency); the adjacent second 16B is
  // loaded at write time from the now-hot cache line. Full prefetch spills
  // 56B/lane at the 256-VGPR cap; this costs 24B/lane (5 dwords), the
  // best-measuring compromise.
  float4 qa, da;
  int64_t pre_row = 0, pre_stat = 0;
  auto load_qdo = [&](int it) {
    int g = it / nq, q0 = q_lo + (it % nq) * 32;
    int64_t qb_ = (static_cast<int64_t>(b) * HQ + hkv * gqa + g) * S;
    int qr = q0 + qsr < S ? q0 + qsr : S - 1;
    pre_row = qb_ + qr;
    pre_stat = qb_ + (q0 + idx < S ? q0 + idx : S - 1);
    qa = *reinterpret_cast<const float4*>(q + pre_row * kD + qsd);
    da = *reinterpret_cast<const float4*>(dout + pre_row * kD + qsd);
  };
  auto write_qdo = [&](int bu) {
    float4 qb2 = *reinterpret_cast<const float4*>(q + pre_row * kD + qsd + 8);
    float4 db = *reinterpret_cast<const float4*>(dout + pre_row * kD + qsd + 8);
    float lv = 0.f, dvl = 0.f;
    if (idx < 32) {
      lv = lse[pre_stat];
      dvl = delta[pre_stat];
    }
    *reinterpret_cast<float4*>(&q_row[bu][qsr][qsd]) = qa;
    *reinterpret_cast<float4*>(&q_row[bu][qsr][qsd + 8]) = qb2;
    *reinterpret_cast<float4*>(&do_row[bu][qsr][qsd]) = da;
    *reinterpret_cast<float4*>(&do_row[bu][qsr][qsd + 8]) = db;
    const bf16* qp = reinterpret_cast<const bf16*>(&qa);
    const bf16* qp2 = reinterpret_cast<const bf16*>(&qb2);
    const bf16* dp_ = reinterpret_cast<const bf16*>(&da);
    const bf16* dp2 = reinterpret_cast<const bf16*>(&db);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      q_tr[bu][qsd + j][qsr] = qp[j];
      q_tr[bu][qsd + 8 + j][qsr] = qp2[j];
      do_tr[bu][qsd + j][qsr] = dp_[j];
      do_tr[bu][qsd + 8 + j][qsr] = dp2[j];
    }
    if (idx < 32) {
      lse_lds[bu][idx] = lv;
      dlt_lds[bu][idx] = dvl;
    }
  };
  if (niter > 0) {
    load_qdo(0);
    write_qdo(0);
  }
  __syncthreads();

  for (int it = 0; it < niter; ++it) {
    {
      const int q0 = q_lo + (it % nq) * 32;
      const int bu = it & 1;
      if (it + 1 < niter) load_qdo(it + 1);

      // ---- S = Q K^T and dP = dO V^T (C: col = k, rows = 16 q) ----
      frag16 st, dp;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        st[r] = 0.f;
        dp[r] = 0.f;
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int c = 0; c < kD / 16; ++c) {
        frag_ab qaf = *reinterpret_cast<const frag_ab*>(
            &q_row[bu][col][c * 16 + half * 8]);
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qaf, kreg[c], st, 0, 0, 0);
        frag_ab daf = *reinterpret_cast<const frag_ab*>(
            &do_row[bu][col][c * 16 + half * 8]);
        dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(daf, vreg[c], dp, 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- P and dS (in-register; lse/delta via LDS broadcast) ----
      float pv[16], ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int qroff = (r & 3) + 8 * (r >> 2) + 4 * half;
        int qi2 = q0 + qroff;
        bool dead = kc >= S || qi2 >= S || (causal && kc > qi2);
        pv[r] = dead ? 0.f
                     : __expf(scale * st[r] - lse_lds[bu][qroff]);
        ds[r] = dead ? 0.f : pv[r] * (dp[r] - dlt_lds[bu][qroff]) * scale;
      }

      // ---- dV += P^T dO, dK += dS^T Q ----
      frag_ab pa1, pa2, sa1, sa2;
      p16_to_afrags(pv, &pa1, &pa2);
      p16_to_afrags(ds, &sa1, &sa2);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int cc = 0; cc < kD / 32; ++cc) {
        frag_ab d1 = *reinterpret_cast<const frag_ab*>(
            &do_tr[bu][cc * 32 + col][half * 8]);
        frag_ab d2 = *reinterpret_cast<const frag_ab*>(
            &do_tr[bu][cc * 32 + col][16 + half * 8]);
        dvacc[cc] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa1, d1, dvacc[cc], 0, 0, 0);
        dvacc[cc] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa2, d2, dvacc[cc], 0, 0, 0);
        frag_ab q1 = *reinterpret_cast<const frag_ab*>(
            &q_tr[bu][cc * 32 + col][half * 8]);
        frag_ab q2 = *reinterpret_cast<const frag_ab*>(
            &q_tr[bu][cc * 32 + col][16 + half * 8]);
        dkacc[cc] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(sa1, q1, dkacc[cc], 0, 0, 0);
        dkacc[cc] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(sa2, q2, dkacc[cc], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      if (it + 1 < niter) write_qdo(bu ^ 1);
      __syncthreads();
    }
  }

  // ---- store dK/dV (rows = k, col = d within cc) ----
#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int ki = kw0 + (r & 3) + 8 * (r >> 2) + 4 * half;
      if (ki < S) {
        dk[(kbase + ki) * kD + cc * 32 + col] = __float2bfloat16(dkacc[cc][r]);
        dv[(kbase + ki) * kD + cc * 32 + col] = __float2bfloat16(dvacc[cc][r]);
      }
    }
}

// Split dV / dK kernels: the fused dkdv kernel sits at the 256-VGPR cap
// with 2.6x worse per-MFMA efficiency than dq (measured 1.71 ms vs an
// expected ~1.3x of dq's 0.49 ms at bs1 h32 s4096). Recomputing S once
// more (+25% MFMA) buys a light dV kernel (2 GEMMs, occupancy 3) and a
// medium dK kernel (3 GEMMs) with half the transposed-image staging each.

__global__ __launch_bounds__(256, 2) void fa_bwd_dv_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ dout, const float* __restrict__ lse,
    bf16* __restrict__ dv, int B, int HQ, int HK, int S, float scale,
    int causal) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int idx = threadIdx.x;
  const int col = lane & 31;
  const int half = lane >> 5;
  const int kb0 = blockIdx.x * kBK;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int gqa = HQ / HK;
  const int kw0 = kb0 + wave * 32;

  __shared__ bf16 q_row[2][32][kD + kKPad];
  __shared__ bf16 do_tr[2][kD][32 + kVPad];
  __shared__ float lse_lds[2][32];

  const int64_t kbase = (static_cast<int64_t>(b) * HK + hkv) * S;
  const int kc = kw0 + col;
  const int krow_g = kc < S ? kc : S - 1;
  frag_ab kreg[kD / 16];
#pragma unroll
  for (int c = 0; c < kD / 16; ++c)
    kreg[c] = *reinterpret_cast<const frag_ab*>(
        k + (kbase + krow_g) * kD + c * 16 + half * 8);

  frag16 dvacc[kD / 32];
#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) dvacc[cc][r] = 0.f;

  const int qsr = idx & 31, qsd = (idx >> 5) * 16;
  const int q_lo = causal ? kb0 : 0;
  const int nq = (S - q_lo + 31) / 32;
  const int niter = gqa * nq;

  auto stage = [&](int it, int bu) {
    int g = it / nq, q0 = q_lo + (it % nq) * 32;
    int64_t qb_ = (static_cast<int64_t>(b) * HQ + hkv * gqa + g) * S;
    int qr = q0 + qsr < S ? q0 + qsr : S - 1;
    float4 qa = *reinterpret_cast<const float4*>(q + (qb_ + qr) * kD + qsd);
    float4 qb2 =
        *reinterpret_cast<const float4*>(q + (qb_ + qr) * kD + qsd + 8);
    float4 da = *reinterpret_cast<const float4*>(dout + (qb_ + qr) * kD + qsd);
    float4 db =
        *reinterpret_cast<const float4*>(dout + (qb_ + qr) * kD + qsd + 8);
    float lv = 0.f;
    if (idx < 32) lv = lse[qb_ + (q0 + idx < S ? q0 + idx : S - 1)];
    *reinterpret_cast<float4*>(&q_row[bu][qsr][qsd]) = qa;
    *reinterpret_cast<float4*>(&q_row[bu][qsr][qsd + 8]) = qb2;
    const bf16* dp_ = reinterpret_cast<const bf16*>(&da);
    const bf16* dp2 = reinterpret_cast<const bf16*>(&db);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      do_tr[bu][qsd + j][qsr] = dp_[j];
      do_tr[bu][qsd + 8 + j][qsr] = dp2[j];
    }
    if (idx < 32) lse_lds[bu][idx] = lv;
  };
  if (niter > 0) stage(0, 0);
  __syncthreads();

  for (int it = 0; it < niter; ++it) {
    const int q0 = q_lo + (it % nq) * 32;
    const int bu = it & 1;

    frag16 st;
#pragma unroll
    for (int r = 0; r < 16; ++r) st[r] = 0.f;
#pragma unroll
    for (int c = 0; c < kD / 16; ++c) {
      frag_ab qaf = *reinterpret_cast<const frag_ab*>(
          &q_row[bu][col][c * 16 + half * 8]);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qaf, kreg[c], st, 0, 0, 0);
    }

    float pv[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int qroff = (r & 3) + 8 * (r >> 2) + 4 * half;
      int qi2 = q0 + qroff;
      bool dead = kc >= S || qi2 >= S || (causal && kc > qi2);
      pv[r] = dead ? 0.f : __expf(scale * st[r] - lse_lds[bu][qroff]);
    }
    frag_ab pa1, pa2;
    p16_to_afrags(pv, &pa1, &pa2);
#pragma unroll
    for (int cc = 0; cc < kD / 32; ++cc) {
      frag_ab d1 = *reinterpret_cast<const frag_ab*>(
          &do_tr[bu][cc * 32 + col][half * 8]);
      frag_ab d2 = *reinterpret_cast<const frag_ab*>(
          &do_tr[bu][cc * 32 + col][16 + half * 8]);
      dvacc[cc] =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa1, d1, dvacc[cc], 0, 0, 0);
      dvacc[cc] =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa2, d2, dvacc[cc], 0, 0, 0);
    }

    if (it + 1 < niter) stage(it + 1, bu ^ 1);
    __syncthreads();
  }

#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int ki = kw0 + (r & 3) + 8 * (r >> 2) + 4 * half;
      if (ki < S)
        dv[(kbase + ki) * kD + cc * 32 + col] = __float2bfloat16(dvacc[cc][r]);
    }
}

__global__ __launch_bounds__(256, 2) void fa_bwd_dk_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dk, int B, int HQ, int HK, int S, float scale,
    int causal) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int idx = threadIdx.x;
  const int col = lane & 31;
  const int half = lane >> 5;
  const int kb0 = blockIdx.x * kBK;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int gqa = HQ / HK;
  const int kw0 = kb0 + wave * 32;

  __shared__ bf16 q_row[2][32][kD + kKPad];
  __shared__ bf16 do_row[2][32][kD + kKPad];
  __shared__ bf16 q_tr[2][kD][32 + kVPad];
  __shared__ float lse_lds[2][32], dlt_lds[2][32];

  const int64_t kbase = (static_cast<int64_t>(b) * HK + hkv) * S;
  const int kc = kw0 + col;
  const int krow_g = kc < S ? kc : S - 1;
  frag_ab kreg[kD / 16], vreg[kD / 16];
#pragma unroll
  for (int c = 0; c < kD / 16; ++c) {
    kreg[c] = *reinterpret_cast<const frag_ab*>(
        k + (kbase + krow_g) * kD + c * 16 + half * 8);
    vreg[c] = *reinterpret_cast<const frag_ab*>(
        v + (kbase + krow_g) * kD + c * 16 + half * 8);
  }

  frag16 dkacc[kD / 32];
#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) dkacc[cc][r] = 0.f;

  const int qsr = idx & 31, qsd = (idx >> 5) * 16;
  const int q_lo = causal ? kb0 : 0;
  const int nq = (S - q_lo + 31) / 32;
  const int niter = gqa * nq;

  auto stage = [&](int it, int bu) {
    int g = it / nq, q0 = q_lo + (it % nq) * 32;
    int64_t qb_ = (static_cast<int64_t>(b) * HQ + hkv * gqa + g) * S;
    int qr = q0 + qsr < S ? q0 + qsr : S - 1;
    float4 qa = *reinterpret_cast<const float4*>(q + (qb_ + qr) * kD + qsd);
    float4 qb2 =
        *reinterpret_cast<const float4*>(q + (qb_ + qr) * kD + qsd + 8);
    float4 da = *reinterpret_cast<const float4*>(dout + (qb_ + qr) * kD + qsd);
    float4 db =
        *reinterpret_cast<const float4*>(dout + (qb_ + qr) * kD + qsd + 8);
    float lv = 0.f, dvl = 0.f;
    if (idx < 32) {
      int64_t st_ = qb_ + (q0 + idx < S ? q0 + idx : S - 1);
      lv = lse[st_];
      dvl = delta[st_];
    }
    *reinterpret_cast<float4*>(&q_row[bu][qsr][qsd]) = qa;
    *reinterpret_cast<float4*>(&q_row[bu][qsr][qsd + 8]) = qb2;
    *reinterpret_cast<float4*>(&do_row[bu][qsr][qsd]) = da;
    *reinterpret_cast<float4*>(&do_row[bu][qsr][qsd + 8]) = db;
    const bf16* qp = reinterpret_cast<const bf16*>(&qa);
    const bf16* qp2 = reinterpret_cast<const bf16*>(&qb2);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      q_tr[bu][qsd + j][qsr] = qp[j];
      q_tr[bu][qsd + 8 + j][qsr] = qp2[j];
    }
    if (idx < 32) {
      lse_lds[bu][idx] = lv;
      dlt_lds[bu][idx] = dvl;
    }
  };
  if (niter > 0) stage(0, 0);
  __syncthreads();

  for (int it = 0; it < niter; ++it) {
    const int q0 = q_lo + (it % nq) * 32;
    const int bu = it & 1;

    frag16 st, dp;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      st[r] = 0.f;
      dp[r] = 0.f;
    }
#pragma unroll
    for (int c = 0; c < kD / 16; ++c) {
      frag_ab qaf = *reinterpret_cast<const frag_ab*>(
          &q_row[bu][col][c * 16 + half * 8]);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qaf, kreg[c], st, 0, 0, 0);
      frag_ab daf = *reinterpret_cast<const frag_ab*>(
          &do_row[bu][col][c * 16 + half * 8]);
      dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(daf, vreg[c], dp, 0, 0, 0);
    }

    float ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int qroff = (r & 3) + 8 * (r >> 2) + 4 * half;
      int qi2 = q0 + qroff;
      bool dead = kc >= S || qi2 >= S || (causal && kc > qi2);
      float pvr = dead ? 0.f : __expf(scale * st[r] - lse_lds[bu][qroff]);
      ds[r] = dead ? 0.f : pvr * (dp[r] - dlt_lds[bu][qroff]) * scale;
    }
    frag_ab sa1, sa2;
    p16_to_afrags(ds, &sa1, &sa2);
#pragma unroll
    for (int cc = 0; cc < kD / 32; ++cc) {
      frag_ab q1 = *reinterpret_cast<const frag_ab*>(
          &q_tr[bu][cc * 32 + col][half * 8]);
      frag_ab q2 = *reinterpret_cast<const frag_ab*>(
          &q_tr[bu][cc * 32 + col][16 + half * 8]);
      dkacc[cc] =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(sa1, q1, dkacc[cc], 0, 0, 0);
      dkacc[cc] =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(sa2, q2, dkacc[cc], 0, 0, 0);
    }

    if (it + 1 < niter) stage(it + 1, bu ^ 1);
    __syncthreads();
  }

#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int ki = kw0 + (r & 3) + 8 * (r >> 2) + 4 * half;
      if (ki < S)
        dk[(kbase + ki) * kD + cc * 32 + col] = __float2bfloat16(dkacc[cc][r]);
    }
}

__global__ __launch_bounds__(256, 2) void fa_bwd_dq_v2_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dq, int B, int HQ, int HK, int S, float scale,
    int causal) {
  const int lane = threadIdx.x % 64;
  const int wave = threadIdx.x / 64;
  const int idx = threadIdx.x;
  const int col = lane & 31;
  const int half = lane >> 5;
  const int qb = blockIdx.x * kBK;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = h / (HQ / HK);
  const int q0w = qb + wave * 32;

  __shared__ bf16 k_row2[2][32][kD + kKPad];
  __shared__ bf16 v_row2[2][32][kD + kKPad];
  __shared__ bf16 k_tr[2][kD][32 + kVPad];

  const int64_t qbase = (static_cast<int64_t>(b) * HQ + h) * S;
  const int64_t kbase = (static_cast<int64_t>(b) * HK + hk) * S;
  const float* lse_h = lse + (static_cast<int64_t>(b) * HQ + h) * S;
  const float* dlt_h = delta + (static_cast<int64_t>(b) * HQ + h) * S;

  const int qi = q0w + col;
  const int qrow = qi < S ? qi : S - 1;
  frag_ab qreg[kD / 16], doreg[kD / 16];
#pragma unroll
  for (int c = 0; c < kD / 16; ++c) {
    qreg[c] = *reinterpret_cast<const frag_ab*>(
        q + (qbase + qrow) * kD + c * 16 + half * 8);
    doreg[c] = *reinterpret_cast<const frag_ab*>(
        dout + (qbase + qrow) * kD + c * 16 + half * 8);
  }
  const float lse_q = qi < S ? lse_h[qi] : 0.f;
  const float dlt_q = qi < S ? dlt_h[qi] : 0.f;

  frag16 dqacc[kD / 32];
#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) dqacc[cc][r] = 0.f;

  const int ksr = idx & 31, ksd = (idx >> 5) * 16;
  const int kend = causal ? (qb + kBK < S ? qb + kBK : S) : S;
  const int ntiles = (kend + 31) / 32;

  // T14: prefetch tile t+1's K/V during tile t's compute; one barrier per
  // tile publishes the other LDS buffer.
  float4 ka, kb2, va, vb;
  auto load_kv = [&](int k0) {
    int kr = k0 + ksr < S ? k0 + ksr : S - 1;
    ka = *reinterpret_cast<const float4*>(k + (kbase + kr) * kD + ksd);
    kb2 = *reinterpret_cast<const float4*>(k + (kbase + kr) * kD + ksd + 8);
    va = *reinterpret_cast<const float4*>(v + (kbase + kr) * kD + ksd);
    vb = *reinterpret_cast<const float4*>(v + (kbase + kr) * kD + ksd + 8);
  };
  auto write_kv = [&](int buf) {
    *reinterpret_cast<float4*>(&k_row2[buf][ksr][ksd]) = ka;
    *reinterpret_cast<float4*>(&k_row2[buf][ksr][ksd + 8]) = kb2;
    *reinterpret_cast<float4*>(&v_row2[buf][ksr][ksd]) = va;
    *reinterpret_cast<float4*>(&v_row2[buf][ksr][ksd + 8]) = vb;
    const bf16* kp = reinterpret_cast<const bf16*>(&ka);
    const bf16* kp2 = reinterpret_cast<const bf16*>(&kb2);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      k_tr[buf][ksd + j][ksr] = kp[j];
      k_tr[buf][ksd + 8 + j][ksr] = kp2[j];
    }
  };
  load_kv(0);
  write_kv(0);
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    const int k0 = t * 32;
    const int buf = t & 1;
    if (t + 1 < ntiles) load_kv(k0 + 32);

    // ---- S^T = K Q^T, dP^T = V dO^T (C: col = q, rows = 16 k) ----
    frag16 st, dpt;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      st[r] = 0.f;
      dpt[r] = 0.f;
    }
#pragma unroll
    for (int c = 0; c < kD / 16; ++c) {
      frag_ab kaf = *reinterpret_cast<const frag_ab*>(
          &k_row2[buf][col][c * 16 + half * 8]);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kaf, qreg[c], st, 0, 0, 0);
      frag_ab vaf = *reinterpret_cast<const frag_ab*>(
          &v_row2[buf][col][c * 16 + half * 8]);
      dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vaf, doreg[c], dpt, 0, 0, 0);
    }

    // ---- dS^T in-register (per-lane lse/delta scalars) ----
    float ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int ki = k0 + (r & 3) + 8 * (r >> 2) + 4 * half;
      bool dead = ki >= S || qi >= S || (causal && ki > qi);
      float pT = dead ? 0.f : __expf(scale * st[r] - lse_q);
      ds[r] = dead ? 0.f : pT * (dpt[r] - dlt_q) * scale;
    }

    if (t + 1 < ntiles) write_kv(buf ^ 1);

    // ---- dQ += dS K (B from transposed K image) ----
    frag_ab a1, a2;
    p16_to_afrags(ds, &a1, &a2);
#pragma unroll
    for (int cc = 0; cc < kD / 32; ++cc) {
      frag_ab k1 = *reinterpret_cast<const frag_ab*>(
          &k_tr[buf][cc * 32 + col][half * 8]);
      frag_ab k2 = *reinterpret_cast<const frag_ab*>(
          &k_tr[buf][cc * 32 + col][16 + half * 8]);
      dqacc[cc] =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, k1, dqacc[cc], 0, 0, 0);
      dqacc[cc] =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(a2, k2, dqacc[cc], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int cc = 0; cc < kD / 32; ++cc)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int qr2 = q0w + (r & 3) + 8 * (r >> 2) + 4 * half;
      if (qr2 < S)
        dq[(qbase + qr2) * kD + cc * 32 + col] =
            __float2bfloat16(dqacc[cc][r]);
    }
}

}  // namespace

void launch_fa_bwd(const void* q, const void* k, const void* v,
                   const void* dout, const void* o, const float* lse,
                   float* delta, void* dq, void* dk, void* dv, int B, int HQ,
                   int HK, int S, int D, bool causal, hipStream_t stream) {
  if (D != kD) return;  // head dim 128 only (wrapper validates)
  float scale = 1.f / sqrtf(static_cast<float>(D));
  int64_t rows = static_cast<int64_t>(B) * HQ * S;
  hipLaunchKernelGGL(fa_delta_kernel,
                     static_cast<unsigned>((rows + kWavesFA - 1) / kWavesFA),
                     256, 0, stream, static_cast<const bf16*>(o),
                     static_cast<const bf16*>(dout), delta, rows);
  const char* fv = getenv("AXON_FA_V");
  if (fv && fv[0] == '0') {  // debug fallback: round-1 v0 kernels
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
    return;
  }
  dim3 gkv((S + kBK - 1) / kBK, HK, B);
  const char* fb = getenv("AXON_FA_BWD");
  if (fb && fb[0] == 's') {
    // Split dV/dK variant: measured 2.99 ms vs fused 2.2 (the staging is
    // amortized across both GEMM sets in the fused kernel) — selectable
    // for re-measurement.
    hipLaunchKernelGGL(fa_bwd_dv_kernel, gkv, 256, 0, stream,
                       static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                       static_cast<const bf16*>(dout), lse,
                       static_cast<bf16*>(dv), B, HQ, HK, S, scale,
                       causal ? 1 : 0);
    hipLaunchKernelGGL(fa_bwd_dk_kernel, gkv, 256, 0, stream,
                       static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                       static_cast<const bf16*>(v),
                       static_cast<const bf16*>(dout), lse, delta,
                       static_cast<bf16*>(dk), B, HQ, HK, S, scale,
                       causal ? 1 : 0);
  } else {
    hipLaunchKernelGGL(fa_bwd_dkdv_v2_kernel, gkv, 256, 0, stream,
                       static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                       static_cast<const bf16*>(v),
                       static_cast<const bf16*>(dout), lse, delta,
                       static_cast<bf16*>(dk), static_cast<bf16*>(dv), B, HQ,
                       HK, S, scale, causal ? 1 : 0);
  }
  dim3 gq((S + kBK - 1) / kBK, HQ, B);
  hipLaunchKernelGGL(fa_bwd_dq_v2_kernel, gq, 256, 0, stream,
                     static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                     static_cast<const bf16*>(v),
                     static_cast<const bf16*>(dout), lse, delta,
                     static_cast<bf16*>(dq), B, HQ, HK, S, scale,
                     causal ? 1 : 0);
}

void launch_fa_fwd(const void* q, const void* k, const void* v, void* o,
                   float* lse, int B, int HQ, int HK, int S, int D,
                   bool causal, hipStream_t stream) {
  if (D != kD) return;  // Llama head dim only (wrapper validates)
  float scale = 1.f / sqrtf(static_cast<float>(D));
  const char* fv = getenv("AXON_FA_V");
  if (fv && fv[0] == '0') {  // debug fallback: round-1 v0 kernel
    dim3 grid((S + kQTile * kWavesFA - 1) / (kQTile * kWavesFA), HQ, B);
    hipLaunchKernelGGL(fa_fwd_kernel, grid, 256, 0, stream,
                       static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                       static_cast<const bf16*>(v), static_cast<bf16*>(o), lse,
                       B, HQ, HK, S, scale, causal ? 1 : 0);
    return;
  }
  const char* fw = getenv("AXON_FA_W");
  if (fw && fw[0] == '4') {
    // 4-wave variant: measured 337 vs 390 TF (the +16 staging VGPRs drop
    // occupancy to 2) — kept selectable for re-measurement.
    dim3 grid((S + 127) / 128, HQ, B);
    hipLaunchKernelGGL((fa_fwd_v2_kernel<4>), grid, 256, 0, stream,
                       static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                       static_cast<const bf16*>(v), static_cast<bf16*>(o), lse,
                       B, HQ, HK, S, scale, causal ? 1 : 0);
    return;
  }
  dim3 grid((S + kBq2 - 1) / kBq2, HQ, B);
  hipLaunchKernelGGL((fa_fwd_v2_kernel<8>), grid, 512, 0, stream,
                     static_cast<const bf16*>(q), static_cast<const bf16*>(k),
                     static_cast<const bf16*>(v), static_cast<bf16*>(o), lse,
                     B, HQ, HK, S, scale, causal ? 1 : 0);
}

}  // namespace axon
