// apex_amd._mfma — flash attention for gfx950 (hardware-validated round 2;
// wired into the transformer models and contrib MHA modules).
//
// One 256-thread workgroup per (batch*head, 64 query rows); each wave64 owns
// 16 query rows. Per 32-wide KV tile:
//   the K/V (fwd: V; dq: K; dkv: Q+dO) tile is staged ONCE per workgroup
//       into a DOUBLE-BUFFERED shared LDS tile (block-cooperative,
//       vectorized both sides; the next tile prefetches during the current
//       tile's MFMA work, block barriers order the buffers),
//   S = Q K^T via mfma_f32_16x16x32_bf16 (Q held as A-fragments in
//       registers for the whole row block),
//   online softmax (row max/sum via 4-step shfl_xor over the 16 column
//       lanes; m/l replicated across those lanes),
//   optional fused philox attention dropout (counter = flat (bh, q, kv)
//       index; the backward kernels regenerate the identical mask),
//   P V via one MFMA per 16 head-dim columns, with P transposed from the
//       D-fragment to the A-fragment layout through a 1 KB per-wave LDS
//       bounce (wave-local: lgkm-only s_waitcnt, so the bounce does not
//       serialize against the prefetch's vector-memory loads).
// Inputs are read through (b, h, s) strides with D contiguous, so BSHD
// views of a packed QKV projection pass with no .contiguous() copies;
// cross-attention (Sq != Skv) is supported (causal requires Sq == Skv).
//
// Fragment layouts (verified on-device by mfma_tile_probe):
//   A[r][k]: lane l -> r = l%16, k = (l/16)*8 + j
//   B[k][c]: lane l -> c = l%16, k = (l/16)*8 + j
//   D[r][c]: lane l reg q -> r = (l>>4)*4 + q, c = l&15
#include "common.h"

#include <vector>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int FM_ROWS = 16;   // q rows per wave
constexpr int FM_WAVES = 4;   // waves per workgroup
constexpr int FM_BN = 32;     // kv tile (one MFMA K step for P@V)
constexpr int FM_PAD = 8;     // LDS tile row padding (shorts) -> 16B-aligned rows

// wave-local LDS drain WITHOUT waiting outstanding vector-memory loads:
// s_waitcnt lgkmcnt(0), vmcnt(max), expcnt(max) — gfx9-family encoding
// (vmcnt bits [3:0]+[15:14], expcnt [6:4], lgkmcnt [13:8]). Used for the
// per-wave P/dS fragment bounces so they do NOT serialize against the
// double-buffered tile prefetch in flight.
#define FM_WAIT_LDS() __builtin_amdgcn_s_waitcnt(0xC07F)

// Philox4x32-10 attention dropout (reference analogue: the philox.cuh-era
// fused-attention dropout). Counter = flat (bh, q, kv) index of the
// attention-probability element, so forward and BOTH backward kernels
// regenerate the identical mask — no stored mask tensor.
__device__ __forceinline__ void fm_philox_round(uint32_t& c0, uint32_t& c1, uint32_t& c2,
                                                uint32_t& c3, uint32_t k0, uint32_t k1) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const uint32_t hi0 = __umulhi(M0, c0), lo0 = M0 * c0;
  const uint32_t hi1 = __umulhi(M1, c2), lo1 = M1 * c2;
  c0 = hi1 ^ c1 ^ k0;
  c1 = lo1;
  c2 = hi0 ^ c3 ^ k1;
  c3 = lo0;
}

__device__ __forceinline__ float fm_philox_uniform(unsigned long long seed, long idx) {
  uint32_t c0 = (uint32_t)((unsigned long)idx >> 2);
  uint32_t c1 = (uint32_t)((unsigned long)idx >> 34);
  uint32_t c2 = 0u, c3 = 0u;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    fm_philox_round(c0, c1, c2, c3, k0, k1);
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  uint32_t res;
  switch (idx & 3) {
    case 0: res = c0; break;
    case 1: res = c1; break;
    case 2: res = c2; break;
    default: res = c3; break;
  }
  return (res >> 8) * (1.0f / 16777216.0f);
}

// post-softmax keep/scale: returns p * mask / (1-prob) for element
// (bh, row=q, col=kv); the softmax denominator stays PRE-dropout.
__device__ __forceinline__ float fm_dropout_p(float p, unsigned long long seed, long bh,
                                              long q_g, long kv_g, long S, long SKV,
                                              float prob, float rinv) {
  const long idx = ((long)bh * S + q_g) * SKV + kv_g;
  return fm_philox_uniform(seed, idx) >= prob ? p * rinv : 0.f;
}

// Stage one 32 x D bf16 tile from HBM into LDS (row-major, padded rows),
// wave-cooperative and fully vectorized (bf16x8 both sides). Replaces the
// per-fragment scalar column-strided HBM gathers that capped the round-1
// kernels at ~190 TF — B-fragments are then read from LDS.
template <int D>
__device__ __forceinline__ void fm_stage_tile(short* __restrict__ dst,
                                              const short* __restrict__ src, long row_stride) {
  constexpr int VPR = D / 8;    // vectors per row
  constexpr int NV = FM_BN * VPR;
  const int lane = threadIdx.x & 63;
#pragma unroll
  for (int v = lane; v < NV; v += 64) {
    const int r = v / VPR;
    const int c0 = (v % VPR) * 8;
    *reinterpret_cast<bf16x8*>(dst + r * (D + FM_PAD) + c0) =
        *reinterpret_cast<const bf16x8*>(src + (long)r * row_stride + c0);
  }
}

// block-cooperative variant: ONE shared tile per workgroup (all 256 threads
// stage; callers barrier). Quarters the staging traffic and LDS footprint of
// the per-wave form — the four waves of a workgroup consume the same tiles.
template <int D>
__device__ __forceinline__ void fm_stage_tile_block(short* __restrict__ dst,
                                                    const short* __restrict__ src,
                                                    long row_stride) {
  constexpr int VPR = D / 8;
  constexpr int NV = FM_BN * VPR;
#pragma unroll
  for (int v = threadIdx.x; v < NV; v += FM_WAVES * 64) {
    const int r = v / VPR;
    const int c0 = (v % VPR) * 8;
    *reinterpret_cast<bf16x8*>(dst + r * (D + FM_PAD) + c0) =
        *reinterpret_cast<const bf16x8*>(src + (long)r * row_stride + c0);
  }
}

__device__ __forceinline__ float row_reduce_max16(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) v = fmaxf(v, __shfl_xor(v, m));
  return v;
}

__device__ __forceinline__ float row_reduce_sum16(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) v += __shfl_xor(v, m);
  return v;
}

template <bool CAUSAL, int D, bool DROPOUT = false>
__global__ void __launch_bounds__(FM_WAVES * 64, 2) fmha_fwd_kernel(
    const short* __restrict__ Q, const short* __restrict__ K, const short* __restrict__ V,
    short* __restrict__ O, float* __restrict__ LSE, int S, int SKV, float scale, int H,
    long qb, long qh, long qs, long kb, long kh, long ks, long vb, long vh, long vs,
    float drop_p = 0.f, float drop_rinv = 1.f, unsigned long long drop_seed = 0) {
  constexpr int NK = D / 32;  // K chunks for Q@K^T
  constexpr int ND = D / 16;  // 16-col output groups for P@V
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int bh = blockIdx.y;
  const int q0 = blockIdx.x * (FM_WAVES * FM_ROWS) + wave * FM_ROWS;
  // waves past Sq stay RESIDENT (they co-stage the shared V tile and hit the
  // block barriers) but skip all math and stores
  const bool active = q0 < S;

  const long bb = bh / H, hh = bh % H;
  const short* q_ptr = Q + bb * qb + hh * qh;
  const short* k_ptr = K + bb * kb + hh * kh;
  const short* v_ptr = V + bb * vb + hh * vh;

  constexpr int FM_TILE_ELEMS = FM_BN * (D + FM_PAD);
  __shared__ short lds_p[FM_WAVES][FM_ROWS * FM_BN];
  __shared__ short lds_v[2 * FM_TILE_ELEMS];  // double-buffered shared tile
  short* pbuf = lds_p[wave];
  short* vbuf_pair = lds_v;

  // Q rows for this wave, as A-fragments, resident for the whole pass
  bf16x8 aq[NK];
  const int a_row = min(q0 + (lane & 15), S - 1);
  if (active) {
#pragma unroll
    for (int c = 0; c < NK; ++c)
      aq[c] = *reinterpret_cast<const bf16x8*>(q_ptr + (long)a_row * qs + c * 32 + (lane >> 4) * 8);
  }

  f32x4 acc[ND];
#pragma unroll
  for (int d = 0; d < ND; ++d) acc[d] = f32x4{0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    m_run[q] = -1e30f;
    l_run[q] = 0.f;
  }

  // rows this lane's D-fragments correspond to (replicated over col lanes)
  const int my_r0 = (lane >> 4) * 4;  // + q
  // block-uniform bound (last wave's rows); a wave's extra diagonal tiles
  // are fully masked, which leaves its online-softmax state unchanged
  const int q0_wg = blockIdx.x * (FM_WAVES * FM_ROWS);
  const int kv_end = CAUSAL
      ? min(SKV, ((q0_wg + FM_WAVES * FM_ROWS - 1) / FM_BN + 1) * FM_BN) : SKV;

  // a wave's own last useful tile (tiles past it are fully causal-masked)
  const int kv_end_wave = CAUSAL ? min(SKV, ((q0 + FM_ROWS - 1) / FM_BN + 1) * FM_BN) : SKV;

  const int ntiles = (kv_end + FM_BN - 1) / FM_BN;
  if (ntiles > 0) fm_stage_tile_block<D>(vbuf_pair, v_ptr, vs);
  __syncthreads();
  int vb_cur = 0;
  for (int ti = 0; ti < ntiles; ++ti) {
    const int kv0 = ti * FM_BN;
    // prefetch the next tile into the other buffer while this one computes
    if (ti + 1 < ntiles)
      fm_stage_tile_block<D>(vbuf_pair + (vb_cur ^ 1) * FM_TILE_ELEMS,
                             v_ptr + (long)(kv0 + FM_BN) * vs, vs);
    short* vbuf = vbuf_pair + vb_cur * FM_TILE_ELEMS;
    if (!active || kv0 >= kv_end_wave) {
      __syncthreads();
      vb_cur ^= 1;
      continue;
    }
    // ---- S = scale * Q K^T for this 16 x 32 tile (two 16x16 halves) ----
    f32x4 s_half[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      f32x4 s = f32x4{0.f, 0.f, 0.f, 0.f};
      const int k_row = kv0 + j * 16 + (lane & 15);
#pragma unroll
      for (int c = 0; c < NK; ++c) {
        const bf16x8 bk = *reinterpret_cast<const bf16x8*>(
            k_ptr + (long)k_row * ks + c * 32 + (lane >> 4) * 8);
        s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[c], bk, s, 0, 0, 0);
      }
      s_half[j] = s;
    }

    // ---- scale + causal mask + online softmax update ----
    float p_val[2][4];
    float alpha[4];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int row_g = q0 + my_r0 + q;
      float mx = -1e30f;
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        float v = s_half[j][q] * scale;
        if (CAUSAL) {
          const int col_g = kv0 + j * 16 + (lane & 15);
          if (col_g > row_g) v = -1e30f;
        }
        s_half[j][q] = v;
        mx = fmaxf(mx, v);
      }
      mx = row_reduce_max16(mx);
      const float m_new = fmaxf(m_run[q], mx);
      alpha[q] = __expf(m_run[q] - m_new);
      float sum = 0.f;
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const float p = __expf(s_half[j][q] - m_new);
        p_val[j][q] = p;
        sum += p;
      }
      sum = row_reduce_sum16(sum);
      l_run[q] = l_run[q] * alpha[q] + sum;
      m_run[q] = m_new;
    }

    // ---- rescale the running output ----
#pragma unroll
    for (int d = 0; d < ND; ++d)
#pragma unroll
      for (int q = 0; q < 4; ++q) acc[d][q] *= alpha[q];

    // ---- P: D-fragment -> A-fragment via the wave-local LDS bounce ----
    // (dropout applies to the P used for P@V; l_run keeps the pre-dropout
    // softmax denominator)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        float pv = p_val[j][q];
        if (DROPOUT)
          pv = fm_dropout_p(pv, drop_seed, bh, q0 + my_r0 + q,
                            kv0 + j * 16 + (lane & 15), S, SKV, drop_p, drop_rinv);
        const __hip_bfloat16 pb = __float2bfloat16(pv);
        pbuf[(my_r0 + q) * FM_BN + j * 16 + (lane & 15)] = *reinterpret_cast<const short*>(&pb);
      }
    FM_WAIT_LDS();  // wave-local: drain LDS stores before reads
    const bf16x8 ap = *reinterpret_cast<const bf16x8*>(
        pbuf + (lane & 15) * FM_BN + (lane >> 4) * 8);

    // ---- acc += P V (one MFMA per 16 head-dim columns) ----
#pragma unroll
    for (int d = 0; d < ND; ++d) {
      // B[k=kv][c=dim] from the staged LDS tile: lane -> c = lane%16,
      // k = (lane/16)*8 + j (8 short reads, conflict-free rows)
      bf16x8 bv;
#pragma unroll
      for (int jj = 0; jj < 8; ++jj) {
        bv[jj] = vbuf[((lane >> 4) * 8 + jj) * (D + FM_PAD) + d * 16 + (lane & 15)];
      }
      acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, acc[d], 0, 0, 0);
    }
    __syncthreads();  // prefetch visible + all reads of the current tile done
    vb_cur ^= 1;
  }

  if (!active) return;
  // ---- epilogue: O = acc / l, LSE = m + log(l) ----
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const long row_g = q0 + my_r0 + q;
    const float inv_l = 1.f / l_run[q];
#pragma unroll
    for (int d = 0; d < ND; ++d) {
      const __hip_bfloat16 o = __float2bfloat16(acc[d][q] * inv_l);
      O[((long)bh * S + row_g) * D + d * 16 + (lane & 15)] =
          *reinterpret_cast<const short*>(&o);
    }
    if ((lane & 15) == 0 && LSE)
      LSE[(long)bh * S + row_g] = m_run[q] + logf(l_run[q]);
  }
}

// ---------------- backward ----------------
// Two kernels, both recomputing P from (Q, K, lse) — index math verified
// lane-for-lane by tests/test_fmha_sim_cpu.py (simulate_dq_wave /
// simulate_dkv_wave):
//   dq:  one wave per 16 q rows, loops kv tiles (forward orientation);
//        dP = dO V^T, dS = P*(dP - delta_row), dQ += dS K * scale.
//   dkv: one wave per 16 kv rows, loops q tiles with TRANSPOSED fragment
//        roles (S^T = K Q^T, lse/delta indexed by the q column);
//        dV += P^T dO, dK += dS^T Q * scale.

template <bool CAUSAL, int D, bool DROPOUT = false>
__global__ void __launch_bounds__(FM_WAVES * 64, 2) fmha_bwd_dq_kernel(
    const short* __restrict__ Q, const short* __restrict__ K, const short* __restrict__ V,
    const short* __restrict__ dO, const float* __restrict__ LSE,
    const float* __restrict__ DELTA, short* __restrict__ dQ, int S, int SKV, float scale, int H,
    long qb, long qh, long qs, long kb, long kh, long ks, long vb, long vh, long vs,
    long ob, long oh, long os, float drop_p = 0.f, float drop_rinv = 1.f,
    unsigned long long drop_seed = 0) {
  constexpr int NK = D / 32;
  constexpr int ND = D / 16;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int bh = blockIdx.y;
  const int q0 = blockIdx.x * (FM_WAVES * FM_ROWS) + wave * FM_ROWS;
  const bool active = q0 < S;  // inactive waves co-stage + barrier only
  const long bb = bh / H, hh = bh % H;
  const short* q_ptr = Q + bb * qb + hh * qh;
  const short* k_ptr = K + bb * kb + hh * kh;
  const short* v_ptr = V + bb * vb + hh * vh;
  const short* do_ptr = dO + bb * ob + hh * oh;
  const float* lse = LSE + (long)bh * S;
  const float* delta = DELTA + (long)bh * S;

  constexpr int FM_TILE_ELEMS = FM_BN * (D + FM_PAD);
  __shared__ short lds_p[FM_WAVES][FM_ROWS * FM_BN];
  __shared__ short lds_k[2 * FM_TILE_ELEMS];  // double-buffered shared K tile
  short* pbuf = lds_p[wave];
  short* kbuf_pair = lds_k;

  bf16x8 aq[NK], ado[NK];
  const int a_row = min(q0 + (lane & 15), S - 1);
  if (active) {
#pragma unroll
    for (int c = 0; c < NK; ++c) {
      aq[c] = *reinterpret_cast<const bf16x8*>(q_ptr + (long)a_row * qs + c * 32 + (lane >> 4) * 8);
      ado[c] = *reinterpret_cast<const bf16x8*>(do_ptr + (long)a_row * os + c * 32 + (lane >> 4) * 8);
    }
  }
  f32x4 dq_acc[ND];
#pragma unroll
  for (int d = 0; d < ND; ++d) dq_acc[d] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int my_r0 = (lane >> 4) * 4;
  // block-uniform bound: extra diagonal tiles give p = 0 -> ds = 0
  const int q0_wg = blockIdx.x * (FM_WAVES * FM_ROWS);
  const int kv_end = CAUSAL
      ? min(SKV, ((q0_wg + FM_WAVES * FM_ROWS - 1) / FM_BN + 1) * FM_BN) : SKV;

  const int kv_end_wave = CAUSAL ? min(SKV, ((q0 + FM_ROWS - 1) / FM_BN + 1) * FM_BN) : SKV;

  const int ntiles = (kv_end + FM_BN - 1) / FM_BN;
  if (ntiles > 0) fm_stage_tile_block<D>(kbuf_pair, k_ptr, ks);
  __syncthreads();
  int kb_cur = 0;
  for (int ti = 0; ti < ntiles; ++ti) {
    const int kv0 = ti * FM_BN;
    if (ti + 1 < ntiles)
      fm_stage_tile_block<D>(kbuf_pair + (kb_cur ^ 1) * FM_TILE_ELEMS,
                             k_ptr + (long)(kv0 + FM_BN) * ks, ks);
    short* kbuf = kbuf_pair + kb_cur * FM_TILE_ELEMS;
    if (!active || kv0 >= kv_end_wave) {
      __syncthreads();
      kb_cur ^= 1;
      continue;
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      f32x4 s = f32x4{0.f, 0.f, 0.f, 0.f};
      f32x4 dp = f32x4{0.f, 0.f, 0.f, 0.f};
      const int k_row = kv0 + j * 16 + (lane & 15);
      const int k_lrow = j * 16 + (lane & 15);
#pragma unroll
      for (int c = 0; c < NK; ++c) {
        const bf16x8 bk = *reinterpret_cast<const bf16x8*>(
            kbuf + k_lrow * (D + FM_PAD) + c * 32 + (lane >> 4) * 8);
        s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[c], bk, s, 0, 0, 0);
        const bf16x8 bvt = *reinterpret_cast<const bf16x8*>(
            v_ptr + (long)k_row * vs + c * 32 + (lane >> 4) * 8);
        dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado[c], bvt, dp, 0, 0, 0);
      }
#pragma unroll
      for (int qi = 0; qi < 4; ++qi) {
        const int row_g = q0 + my_r0 + qi;
        const int col_g = kv0 + j * 16 + (lane & 15);
        float p = (CAUSAL && col_g > row_g) ? 0.f : __expf(s[qi] * scale - lse[row_g]);
        // with dropout: dS = P * (M*rinv*dPd - delta) — the Sum term folds
        // to the SAME delta = rowsum(dO*O) as the no-dropout case
        float dpe = dp[qi];
        if (DROPOUT)
          dpe = fm_philox_uniform(drop_seed, ((long)bh * S + row_g) * SKV + col_g) >= drop_p
                    ? dpe * drop_rinv : 0.f;
        const float ds = p * (dpe - delta[row_g]);
        const __hip_bfloat16 db = __float2bfloat16(ds);
        pbuf[(my_r0 + qi) * FM_BN + j * 16 + (lane & 15)] = *reinterpret_cast<const short*>(&db);
      }
    }
    FM_WAIT_LDS();
    const bf16x8 a_ds = *reinterpret_cast<const bf16x8*>(
        pbuf + (lane & 15) * FM_BN + (lane >> 4) * 8);
#pragma unroll
    for (int d = 0; d < ND; ++d) {
      bf16x8 bK;
#pragma unroll
      for (int jj = 0; jj < 8; ++jj)
        bK[jj] = kbuf[((lane >> 4) * 8 + jj) * (D + FM_PAD) + d * 16 + (lane & 15)];
      dq_acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_ds, bK, dq_acc[d], 0, 0, 0);
    }
    __syncthreads();  // prefetch visible + all reads of the current tile done
    kb_cur ^= 1;
  }
  if (!active) return;
#pragma unroll
  for (int qi = 0; qi < 4; ++qi) {
    const long row_g = q0 + my_r0 + qi;
#pragma unroll
    for (int d = 0; d < ND; ++d) {
      const __hip_bfloat16 o = __float2bfloat16(dq_acc[d][qi] * scale);
      dQ[((long)bh * S + row_g) * D + d * 16 + (lane & 15)] =
          *reinterpret_cast<const short*>(&o);
    }
  }
}

template <bool CAUSAL, int D, bool DROPOUT = false>
__global__ void __launch_bounds__(FM_WAVES * 64, 2) fmha_bwd_dkv_kernel(
    const short* __restrict__ Q, const short* __restrict__ K, const short* __restrict__ V,
    const short* __restrict__ dO, const float* __restrict__ LSE,
    const float* __restrict__ DELTA, short* __restrict__ dK, short* __restrict__ dV,
    int S, int SKV, float scale, int H,
    long qb, long qh, long qs, long kb, long kh, long ks, long vb, long vh, long vs,
    long ob, long oh, long os, float drop_p = 0.f, float drop_rinv = 1.f,
    unsigned long long drop_seed = 0) {
  constexpr int NK = D / 32;
  constexpr int ND = D / 16;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int bh = blockIdx.y;
  const int kv0 = blockIdx.x * (FM_WAVES * FM_ROWS) + wave * FM_ROWS;
  const bool active = kv0 < SKV;  // inactive waves co-stage + barrier only
  const long bb = bh / H, hh = bh % H;
  const short* q_ptr = Q + bb * qb + hh * qh;
  const short* k_ptr = K + bb * kb + hh * kh;
  const short* v_ptr = V + bb * vb + hh * vh;
  const short* do_ptr = dO + bb * ob + hh * oh;
  const float* lse = LSE + (long)bh * S;
  const float* delta = DELTA + (long)bh * S;

  constexpr int FM_TILE_ELEMS = FM_BN * (D + FM_PAD);
  __shared__ short lds_p[FM_WAVES][FM_ROWS * FM_BN];
  __shared__ short lds_ds[FM_WAVES][FM_ROWS * FM_BN];
  __shared__ short lds_q[2 * FM_TILE_ELEMS];   // double-buffered shared tiles
  __shared__ short lds_do[2 * FM_TILE_ELEMS];
  short* pbuf = lds_p[wave];
  short* dsbuf = lds_ds[wave];
  short* qbuf_pair = lds_q;
  short* dobuf_pair = lds_do;

  bf16x8 ak[NK], av[NK];
  const int a_row = min(kv0 + (lane & 15), SKV - 1);
  if (active) {
#pragma unroll
    for (int c = 0; c < NK; ++c) {
      ak[c] = *reinterpret_cast<const bf16x8*>(k_ptr + (long)a_row * ks + c * 32 + (lane >> 4) * 8);
      av[c] = *reinterpret_cast<const bf16x8*>(v_ptr + (long)a_row * vs + c * 32 + (lane >> 4) * 8);
    }
  }
  f32x4 dv_acc[ND], dk_acc[ND];
#pragma unroll
  for (int d = 0; d < ND; ++d) {
    dv_acc[d] = f32x4{0.f, 0.f, 0.f, 0.f};
    dk_acc[d] = f32x4{0.f, 0.f, 0.f, 0.f};
  }
  const int my_r0 = (lane >> 4) * 4;
  // block-uniform start (first wave's tile); later waves' extra early tiles
  // are causal-masked to p = 0 -> no contribution
  const int q_start = CAUSAL
      ? ((blockIdx.x * (FM_WAVES * FM_ROWS)) / FM_BN) * FM_BN : 0;

  const int ntiles = (S - q_start + FM_BN - 1) / FM_BN;
  if (ntiles > 0) {
    fm_stage_tile_block<D>(qbuf_pair, q_ptr + (long)q_start * qs, qs);
    fm_stage_tile_block<D>(dobuf_pair, do_ptr + (long)q_start * os, os);
  }
  __syncthreads();
  int tb_cur = 0;
  for (int ti = 0; ti < ntiles; ++ti) {
    const int q0 = q_start + ti * FM_BN;
    if (ti + 1 < ntiles) {
      fm_stage_tile_block<D>(qbuf_pair + (tb_cur ^ 1) * FM_TILE_ELEMS,
                             q_ptr + (long)(q0 + FM_BN) * qs, qs);
      fm_stage_tile_block<D>(dobuf_pair + (tb_cur ^ 1) * FM_TILE_ELEMS,
                             do_ptr + (long)(q0 + FM_BN) * os, os);
    }
    short* qbuf = qbuf_pair + tb_cur * FM_TILE_ELEMS;
    short* dobuf = dobuf_pair + tb_cur * FM_TILE_ELEMS;
    // skip tiles whose q rows all precede this wave's kv rows (fully masked)
    if (!active || (CAUSAL && q0 + FM_BN <= kv0)) {
      __syncthreads();
      tb_cur ^= 1;
      continue;
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      f32x4 sT = f32x4{0.f, 0.f, 0.f, 0.f};
      f32x4 dpT = f32x4{0.f, 0.f, 0.f, 0.f};
      const int q_lrow = j * 16 + (lane & 15);
#pragma unroll
      for (int c = 0; c < NK; ++c) {
        const bf16x8 bq = *reinterpret_cast<const bf16x8*>(
            qbuf + q_lrow * (D + FM_PAD) + c * 32 + (lane >> 4) * 8);
        sT = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak[c], bq, sT, 0, 0, 0);
        const bf16x8 bdo = *reinterpret_cast<const bf16x8*>(
            dobuf + q_lrow * (D + FM_PAD) + c * 32 + (lane >> 4) * 8);
        dpT = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av[c], bdo, dpT, 0, 0, 0);
      }
#pragma unroll
      for (int qi = 0; qi < 4; ++qi) {
        const int kv_g = kv0 + my_r0 + qi;
        const int q_g = q0 + j * 16 + (lane & 15);
        float p = (CAUSAL && kv_g > q_g) ? 0.f : __expf(sT[qi] * scale - lse[q_g]);
        // dV consumes the POST-dropout P-hat; dS uses pre-dropout P with
        // the masked/rescaled dPd (same identities as the dq kernel)
        float p_hat = p, dpe = dpT[qi];
        if (DROPOUT) {
          const bool keep =
              fm_philox_uniform(drop_seed, ((long)bh * S + q_g) * SKV + kv_g) >= drop_p;
          p_hat = keep ? p * drop_rinv : 0.f;
          dpe = keep ? dpe * drop_rinv : 0.f;
        }
        const float ds = p * (dpe - delta[q_g]);
        const int idx = (my_r0 + qi) * FM_BN + j * 16 + (lane & 15);
        const __hip_bfloat16 pb = __float2bfloat16(p_hat);
        const __hip_bfloat16 db = __float2bfloat16(ds);
        pbuf[idx] = *reinterpret_cast<const short*>(&pb);
        dsbuf[idx] = *reinterpret_cast<const short*>(&db);
      }
    }
    FM_WAIT_LDS();
    const bf16x8 a_p = *reinterpret_cast<const bf16x8*>(
        pbuf + (lane & 15) * FM_BN + (lane >> 4) * 8);
    const bf16x8 a_ds = *reinterpret_cast<const bf16x8*>(
        dsbuf + (lane & 15) * FM_BN + (lane >> 4) * 8);
#pragma unroll
    for (int d = 0; d < ND; ++d) {
      bf16x8 b_do, b_q;
#pragma unroll
      for (int jj = 0; jj < 8; ++jj) {
        const int qlr = (lane >> 4) * 8 + jj;
        b_do[jj] = dobuf[qlr * (D + FM_PAD) + d * 16 + (lane & 15)];
        b_q[jj] = qbuf[qlr * (D + FM_PAD) + d * 16 + (lane & 15)];
      }
      dv_acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_p, b_do, dv_acc[d], 0, 0, 0);
      dk_acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_ds, b_q, dk_acc[d], 0, 0, 0);
    }
    __syncthreads();  // prefetch visible + all reads of the current tiles done
    tb_cur ^= 1;
  }
  if (!active) return;
#pragma unroll
  for (int qi = 0; qi < 4; ++qi) {
    const long kv_g = kv0 + my_r0 + qi;
#pragma unroll
    for (int d = 0; d < ND; ++d) {
      // dK/dV are packed [B, H, SKV, D] — the head stride is SKV, not Sq
      const __hip_bfloat16 ov = __float2bfloat16(dv_acc[d][qi]);
      dV[((long)bh * SKV + kv_g) * D + d * 16 + (lane & 15)] =
          *reinterpret_cast<const short*>(&ov);
      const __hip_bfloat16 ok = __float2bfloat16(dk_acc[d][qi] * scale);
      dK[((long)bh * SKV + kv_g) * D + d * 16 + (lane & 15)] =
          *reinterpret_cast<const short*>(&ok);
    }
  }
}

// ---------------- GEMM-recompute backward helpers ----------------
// The production backward is 5 hipBLASLt batched GEMMs (torch.matmul) plus
// these three single-pass kernels — one HBM pass each instead of the eager
// fp32 mul/exp/mask/cast chains that cost ~6 extra S x S passes.

// delta[row] = sum_d dout[row,d] * out[row,d]; one wave per row. dO may be
// a strided BSHD view (ob/oh/os strides); O is packed.
template <int D>
__global__ void __launch_bounds__(256) fmha_delta_kernel(
    const short* __restrict__ dO, const short* __restrict__ O, float* __restrict__ delta,
    long rows, int H, int S, long ob, long oh, long os) {
  const int lane = threadIdx.x & 63;
  const long row = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  if (row >= rows) return;
  constexpr int PER = D / 64;  // elements per lane (1 for D=64, 2 for D=128)
  const long bb = row / ((long)H * S);
  const long rem = row % ((long)H * S);
  const long dob = bb * ob + (rem / S) * oh + (rem % S) * os;
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < PER; ++j) {
    const __hip_bfloat16 a = *reinterpret_cast<const __hip_bfloat16*>(dO + dob + j * 64 + lane);
    const __hip_bfloat16 b =
        *reinterpret_cast<const __hip_bfloat16*>(O + row * D + j * 64 + lane);
    acc = fmaf(__bfloat162float(a), __bfloat162float(b), acc);
  }
#pragma unroll
  for (int m = 1; m < 64; m <<= 1) acc += __shfl_xor(acc, m);
  if (lane == 0) delta[row] = acc;
}

// p[row, c] = exp(s[row, c] * scale - lse[row]) (0 where causal-masked);
// rows are (bh, q) pairs, c the kv column. 8-wide bf16 vectors (S % 32 == 0).
template <bool CAUSAL>
__global__ void __launch_bounds__(256) fmha_p_kernel(
    const short* __restrict__ Sm, const float* __restrict__ lse, short* __restrict__ P,
    long rows, long S, float scale) {
  const long total_vec = rows * (S / 8);
  for (long vi = (long)blockIdx.x * blockDim.x + threadIdx.x; vi < total_vec;
       vi += (long)gridDim.x * blockDim.x) {
    const long row = vi / (S / 8);
    const long c0 = (vi % (S / 8)) * 8;
    const float l = lse[row];
    const long qrow = row % S;  // q index within the head (rows = BH*S)
    bf16x8 sv = *reinterpret_cast<const bf16x8*>(Sm + row * S + c0);
    bf16x8 pv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const short sraw = sv[j];
      const __hip_bfloat16 sb = *reinterpret_cast<const __hip_bfloat16*>(&sraw);
      float p = __expf(__bfloat162float(sb) * scale - l);
      if (CAUSAL && (c0 + j) > qrow) p = 0.f;
      const __hip_bfloat16 pb = __float2bfloat16(p);
      pv[j] = *reinterpret_cast<const short*>(&pb);
    }
    *reinterpret_cast<bf16x8*>(P + row * S + c0) = pv;
  }
}

// ds[row, c] = p[row, c] * (dp[row, c] - delta[row]) * scale
__global__ void __launch_bounds__(256) fmha_ds_kernel(
    const short* __restrict__ P, const short* __restrict__ dP, const float* __restrict__ delta,
    short* __restrict__ dS, long rows, long S, float scale) {
  const long total_vec = rows * (S / 8);
  for (long vi = (long)blockIdx.x * blockDim.x + threadIdx.x; vi < total_vec;
       vi += (long)gridDim.x * blockDim.x) {
    const long row = vi / (S / 8);
    const long c0 = (vi % (S / 8)) * 8;
    const float d = delta[row];
    bf16x8 pv = *reinterpret_cast<const bf16x8*>(P + row * S + c0);
    bf16x8 dpv = *reinterpret_cast<const bf16x8*>(dP + row * S + c0);
    bf16x8 ov;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const short praw = pv[j];
      const short draw = dpv[j];
      const __hip_bfloat16 pb = *reinterpret_cast<const __hip_bfloat16*>(&praw);
      const __hip_bfloat16 db = *reinterpret_cast<const __hip_bfloat16*>(&draw);
      const float ds = __bfloat162float(pb) * (__bfloat162float(db) - d) * scale;
      const __hip_bfloat16 ob = __float2bfloat16(ds);
      ov[j] = *reinterpret_cast<const short*>(&ob);
    }
    *reinterpret_cast<bf16x8*>(dS + row * S + c0) = ov;
  }
}

}  // namespace

// delta = rowsum(dout * out) in fp32, shape [B, H, S]; dout may be a
// strided [B,H,S,D] view (D contiguous)
at::Tensor fmha_delta(at::Tensor dout, at::Tensor out) {
  auto doc = (dout.dim() == 4 && dout.stride(3) == 1) ? dout : dout.contiguous();
  auto oc = out.contiguous();
  const long D = doc.size(-1);
  const long rows = doc.numel() / D;
  const int H = doc.dim() == 4 ? (int)doc.size(1) : 1;
  const int S = doc.dim() == 4 ? (int)doc.size(2) : (int)doc.size(-2);
  const long ob = doc.dim() == 4 ? doc.stride(0) : rows ? doc.numel() : 0;
  const long oh = doc.dim() == 4 ? doc.stride(1) : 0;
  const long os = doc.dim() == 4 ? doc.stride(2) : D;
  TORCH_CHECK(D == 64 || D == 128, "fmha_delta: head_dim 64/128");
  auto delta = at::empty(doc.sizes().slice(0, doc.dim() - 1),
                         doc.options().dtype(at::kFloat));
  auto stream = current_stream();
  const long grid = std::min<long>((rows * 64 + 255) / 256, 65535);
  if (D == 64)
    hipLaunchKernelGGL((fmha_delta_kernel<64>), dim3((uint32_t)grid), dim3(256), 0, stream,
                       (const short*)doc.data_ptr(), (const short*)oc.data_ptr(),
                       delta.data_ptr<float>(), rows, H, S, ob, oh, os);
  else
    hipLaunchKernelGGL((fmha_delta_kernel<128>), dim3((uint32_t)grid), dim3(256), 0, stream,
                       (const short*)doc.data_ptr(), (const short*)oc.data_ptr(),
                       delta.data_ptr<float>(), rows, H, S, ob, oh, os);
  HIP_CHECK(hipGetLastError());
  return delta;
}

// p = exp(s * scale - lse[row]) (causal-masked); s [B,H,S,S] bf16, lse [B,H,S]
at::Tensor fmha_p(at::Tensor s, at::Tensor lse, double scale, bool causal) {
  auto sc = s.contiguous();
  auto lc = lse.contiguous();
  TORCH_CHECK(sc.scalar_type() == at::ScalarType::BFloat16, "fmha_p: bf16 s");
  const long S = sc.size(-1);
  TORCH_CHECK(S % 8 == 0, "fmha_p: S % 8");
  const long rows = sc.numel() / S;
  auto p = at::empty_like(sc);
  auto stream = current_stream();
  const long grid = std::min<long>((rows * (S / 8) + 255) / 256, 65535);
  if (causal)
    hipLaunchKernelGGL((fmha_p_kernel<true>), dim3((uint32_t)grid), dim3(256), 0, stream,
                       (const short*)sc.data_ptr(), lc.data_ptr<float>(),
                       (short*)p.data_ptr(), rows, S, (float)scale);
  else
    hipLaunchKernelGGL((fmha_p_kernel<false>), dim3((uint32_t)grid), dim3(256), 0, stream,
                       (const short*)sc.data_ptr(), lc.data_ptr<float>(),
                       (short*)p.data_ptr(), rows, S, (float)scale);
  HIP_CHECK(hipGetLastError());
  return p;
}

// ds = p * (dp - delta[row]) * scale
at::Tensor fmha_ds(at::Tensor p, at::Tensor dp, at::Tensor delta, double scale) {
  auto pc = p.contiguous(), dpc = dp.contiguous(), dc = delta.contiguous();
  const long S = pc.size(-1);
  const long rows = pc.numel() / S;
  auto ds = at::empty_like(pc);
  auto stream = current_stream();
  const long grid = std::min<long>((rows * (S / 8) + 255) / 256, 65535);
  hipLaunchKernelGGL(fmha_ds_kernel, dim3((uint32_t)grid), dim3(256), 0, stream,
                     (const short*)pc.data_ptr(), (const short*)dpc.data_ptr(),
                     dc.data_ptr<float>(), (short*)ds.data_ptr(), rows, S, (float)scale);
  HIP_CHECK(hipGetLastError());
  return ds;
}

namespace {
// strided [B,H,S,D] access: last dim must be contiguous and 16B-alignable;
// the kernels read rows through (b_stride, h_stride, s_stride), so BSHD
// views (e.g. q/k/v slices of a packed QKV projection) pass WITHOUT the
// .contiguous() copies that cost the round-1 BERT step ~5% (432
// direct_copy calls per trace).
inline bool fm_strided_ok(const at::Tensor& t) {
  return t.stride(3) == 1 && (t.stride(2) % 8) == 0 && (t.stride(1) % 8) == 0 &&
         (t.stride(0) % 8) == 0;
}
}  // namespace

std::vector<at::Tensor> fmha_fwd(at::Tensor q, at::Tensor k, at::Tensor v, bool causal,
                                 double scale, double dropout_p, long seed) {
  TORCH_CHECK(q.scalar_type() == at::ScalarType::BFloat16, "fmha_fwd: bf16 only");
  TORCH_CHECK(q.dim() == 4, "fmha_fwd: [B, H, S, D]");
  auto qc = fm_strided_ok(q) ? q : q.contiguous();
  auto kc = fm_strided_ok(k) ? k : k.contiguous();
  auto vc = fm_strided_ok(v) ? v : v.contiguous();
  const int B = qc.size(0), H = qc.size(1), S = qc.size(2), D = qc.size(3);
  const int SKV = kc.size(2);
  TORCH_CHECK(kc.size(0) == B && kc.size(1) == H && kc.size(3) == D &&
                  vc.sizes() == kc.sizes(),
              "fmha_fwd: k/v must be [B,H,Skv,D] matching q's B/H/D (no MQA yet)");
  TORCH_CHECK(D == 64 || D == 128, "fmha_fwd: head_dim must be 64 or 128");
  TORCH_CHECK(S % 32 == 0 && SKV % 32 == 0, "fmha_fwd: seq lens must be multiples of 32");
  TORCH_CHECK(!causal || S == SKV, "fmha_fwd: causal needs Sq == Skv");
  auto out = at::empty({B, H, S, D}, qc.options());
  auto lse = at::empty({B, H, S}, qc.options().dtype(at::kFloat));
  auto stream = current_stream();
  dim3 grid((S + FM_WAVES * FM_ROWS - 1) / (FM_WAVES * FM_ROWS), B * H);
  dim3 block(FM_WAVES * 64);
  const float sc = (float)scale;
  const float dp = (float)dropout_p;
  const float dr = dp > 0.f ? 1.f / (1.f - dp) : 1.f;
  const unsigned long long sd = (unsigned long long)seed;

#define FMHA_LAUNCH(CAUSAL, DD, DROP)                                                      \
  hipLaunchKernelGGL((fmha_fwd_kernel<CAUSAL, DD, DROP>), grid, block, 0, stream,          \
                     (const short*)qc.data_ptr(), (const short*)kc.data_ptr(),             \
                     (const short*)vc.data_ptr(), (short*)out.data_ptr(),                  \
                     lse.data_ptr<float>(), S, SKV, sc, H,                                 \
                     qc.stride(0), qc.stride(1), qc.stride(2),                             \
                     kc.stride(0), kc.stride(1), kc.stride(2),                             \
                     vc.stride(0), vc.stride(1), vc.stride(2), dp, dr, sd)
  if (dp > 0.f) {
    if (causal) {
      if (D == 64) FMHA_LAUNCH(true, 64, true);
      else FMHA_LAUNCH(true, 128, true);
    } else {
      if (D == 64) FMHA_LAUNCH(false, 64, true);
      else FMHA_LAUNCH(false, 128, true);
    }
  } else if (causal) {
    if (D == 64) FMHA_LAUNCH(true, 64, false);
    else FMHA_LAUNCH(true, 128, false);
  } else {
    if (D == 64) FMHA_LAUNCH(false, 64, false);
    else FMHA_LAUNCH(false, 128, false);
  }
#undef FMHA_LAUNCH
  HIP_CHECK(hipGetLastError());
  return {out, lse};
}

std::vector<at::Tensor> fmha_bwd(at::Tensor dout, at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor out, at::Tensor lse, bool causal, double scale,
                                 double dropout_p, long seed) {
  TORCH_CHECK(q.scalar_type() == at::ScalarType::BFloat16, "fmha_bwd: bf16 only");
  auto qc = fm_strided_ok(q) ? q : q.contiguous();
  auto kc = fm_strided_ok(k) ? k : k.contiguous();
  auto vc = fm_strided_ok(v) ? v : v.contiguous();
  auto doc = fm_strided_ok(dout) ? dout : dout.contiguous();
  auto oc = out.contiguous(), lsec = lse.contiguous();
  const int B = qc.size(0), H = qc.size(1), S = qc.size(2), D = qc.size(3);
  const int SKV = kc.size(2);
  TORCH_CHECK(D == 64 || D == 128, "fmha_bwd: head_dim must be 64 or 128");
  TORCH_CHECK(S % 32 == 0 && SKV % 32 == 0, "fmha_bwd: seq lens must be multiples of 32");
  TORCH_CHECK(!causal || S == SKV, "fmha_bwd: causal needs Sq == Skv");
  // grad outputs are packed (autograd restrides them back through the view)
  auto dq = at::empty({B, H, S, D}, qc.options());
  auto dk = at::empty({B, H, SKV, D}, kc.options());
  auto dv = at::empty({B, H, SKV, D}, vc.options());
  // delta = rowsum(dO * O) in fp32 — one fused pass (stride-aware)
  auto delta = fmha_delta(doc, oc);
  auto stream = current_stream();
  dim3 grid((S + FM_WAVES * FM_ROWS - 1) / (FM_WAVES * FM_ROWS), B * H);
  dim3 grid_kv((SKV + FM_WAVES * FM_ROWS - 1) / (FM_WAVES * FM_ROWS), B * H);
  dim3 block(FM_WAVES * 64);
  const float sc = (float)scale;
  const float dp = (float)dropout_p;
  const float dr = dp > 0.f ? 1.f / (1.f - dp) : 1.f;
  const unsigned long long sd = (unsigned long long)seed;

#define FMHA_BWD_LAUNCH(CAUSAL, DD, DROP)                                                  \
  do {                                                                                     \
    hipLaunchKernelGGL((fmha_bwd_dq_kernel<CAUSAL, DD, DROP>), grid, block, 0, stream,     \
                       (const short*)qc.data_ptr(), (const short*)kc.data_ptr(),           \
                       (const short*)vc.data_ptr(), (const short*)doc.data_ptr(),          \
                       lsec.data_ptr<float>(), delta.data_ptr<float>(),                    \
                       (short*)dq.data_ptr(), S, SKV, sc, H,                               \
                       qc.stride(0), qc.stride(1), qc.stride(2),                           \
                       kc.stride(0), kc.stride(1), kc.stride(2),                           \
                       vc.stride(0), vc.stride(1), vc.stride(2),                           \
                       doc.stride(0), doc.stride(1), doc.stride(2), dp, dr, sd);           \
    hipLaunchKernelGGL((fmha_bwd_dkv_kernel<CAUSAL, DD, DROP>), grid_kv, block, 0, stream, \
                       (const short*)qc.data_ptr(), (const short*)kc.data_ptr(),           \
                       (const short*)vc.data_ptr(), (const short*)doc.data_ptr(),          \
                       lsec.data_ptr<float>(), delta.data_ptr<float>(),                    \
                       (short*)dk.data_ptr(), (short*)dv.data_ptr(), S, SKV, sc, H,        \
                       qc.stride(0), qc.stride(1), qc.stride(2),                           \
                       kc.stride(0), kc.stride(1), kc.stride(2),                           \
                       vc.stride(0), vc.stride(1), vc.stride(2),                           \
                       doc.stride(0), doc.stride(1), doc.stride(2), dp, dr, sd);           \
  } while (0)
  if (dp > 0.f) {
    if (causal) {
      if (D == 64) FMHA_BWD_LAUNCH(true, 64, true);
      else FMHA_BWD_LAUNCH(true, 128, true);
    } else {
      if (D == 64) FMHA_BWD_LAUNCH(false, 64, true);
      else FMHA_BWD_LAUNCH(false, 128, true);
    }
  } else if (causal) {
    if (D == 64) FMHA_BWD_LAUNCH(true, 64, false);
    else FMHA_BWD_LAUNCH(true, 128, false);
  } else {
    if (D == 64) FMHA_BWD_LAUNCH(false, 64, false);
    else FMHA_BWD_LAUNCH(false, 128, false);
  }
#undef FMHA_BWD_LAUNCH
  HIP_CHECK(hipGetLastError());
  return {dq, dk, dv};
}
