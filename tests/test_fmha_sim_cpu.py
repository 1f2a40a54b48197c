"""Lane-level simulation of csrc/fmha.hip's index math (CPU).

Re-executes the kernel's exact fragment mappings — Q/K/V load expressions,
MFMA fragment layouts (as verified on-device by mfma_tile_probe), the
16-lane shfl reductions, the LDS P-transpose bounce, and the epilogue store
addresses — in pure Python, and checks the result against eager attention.
An index-math bug in the kernel shows up here without a GPU; the remaining
on-hardware risk is only intrinsic semantics, which mfma_tile_probe covers.
"""

import math

import numpy as np
import pytest
import torch

FM_ROWS, FM_BN = 16, 32


def mfma_16x16x32(a_frags, b_frags):
    """a_frags/b_frags: per-lane length-8 arrays following the verified maps
    A[r][k]: lane l -> r=l%16, k=(l/16)*8+j ; B[k][c]: lane l -> c=l%16,
    k=(l/16)*8+j. Returns per-lane length-4 D values:
    D[r][c]: lane l reg q -> r=(l>>4)*4+q, c=l&15."""
    A = np.zeros((16, 32))
    B = np.zeros((32, 16))
    for l in range(64):
        for j in range(8):
            A[l % 16][(l // 16) * 8 + j] = a_frags[l][j]
            B[(l // 16) * 8 + j][l % 16] = b_frags[l][j]
    D = A @ B
    out = np.zeros((64, 4))
    for l in range(64):
        for q in range(4):
            out[l][q] = D[(l >> 4) * 4 + q][l & 15]
    return out


def simulate_wave(Q, K, V, q0, S, D, causal, scale):
    """One wave64's pass over its 16 query rows, per the kernel."""
    NK, ND = D // 32, D // 16

    # Q A-fragments (resident)
    aq = [[None] * 64 for _ in range(NK)]
    for l in range(64):
        a_row = q0 + (l & 15)
        for c in range(NK):
            aq[c][l] = [Q[a_row][c * 32 + (l >> 4) * 8 + j] for j in range(8)]

    acc = np.zeros((ND, 64, 4))
    m_run = np.full((64, 4), -1e30)
    l_run = np.zeros((64, 4))
    kv_end = min(S, ((q0 + FM_ROWS - 1) // FM_BN + 1) * FM_BN) if causal else S

    for kv0 in range(0, kv_end, FM_BN):
        # S = scale * Q K^T (two 16x16 halves)
        s_half = []
        for j in range(2):
            s = np.zeros((64, 4))
            for c in range(NK):
                bkc = [[K[kv0 + j * 16 + (l & 15)][c * 32 + (l >> 4) * 8 + jj]
                        for jj in range(8)] for l in range(64)]
                s += mfma_16x16x32(aq[c], bkc)
            s_half.append(s)

        p_val = np.zeros((2, 64, 4))
        alpha = np.zeros((64, 4))
        for l in range(64):
            for q in range(4):
                row_g = q0 + (l >> 4) * 4 + q
                for j in range(2):
                    v = s_half[j][l][q] * scale
                    if causal and kv0 + j * 16 + (l & 15) > row_g:
                        v = -1e30
                    s_half[j][l][q] = v
        # 16-lane row reductions (lanes sharing l>>4)
        for grp in range(4):
            lanes = [l for l in range(64) if (l >> 4) == grp]
            for q in range(4):
                mx = max(s_half[j][l][q] for j in range(2) for l in lanes)
                for l in lanes:
                    m_new = max(m_run[l][q], mx)
                    alpha[l][q] = math.exp(m_run[l][q] - m_new)
                    m_run[l][q] = m_new
                sm = sum(math.exp(s_half[j][l][q] - m_run[l][q])
                         for j in range(2) for l in lanes)
                for l in lanes:
                    for j in range(2):
                        p_val[j][l][q] = math.exp(s_half[j][l][q] - m_run[l][q])
                    l_run[l][q] = l_run[l][q] * alpha[l][q] + sm

        for d in range(ND):
            acc[d] *= alpha

        # LDS bounce: D-frag layout stores -> A-frag layout reads
        pbuf = np.zeros(FM_ROWS * FM_BN)
        for l in range(64):
            for j in range(2):
                for q in range(4):
                    pbuf[((l >> 4) * 4 + q) * FM_BN + j * 16 + (l & 15)] = p_val[j][l][q]
        ap = [[pbuf[(l & 15) * FM_BN + (l >> 4) * 8 + j] for j in range(8)]
              for l in range(64)]

        for d in range(ND):
            bv = [[V[kv0 + (l >> 4) * 8 + jj][d * 16 + (l & 15)] for jj in range(8)]
                  for l in range(64)]
            acc[d] += mfma_16x16x32(ap, bv)

    # epilogue
    out = np.zeros((FM_ROWS, D))
    lse = np.zeros(FM_ROWS)
    for l in range(64):
        for q in range(4):
            r = (l >> 4) * 4 + q
            for d in range(ND):
                out[r][d * 16 + (l & 15)] = acc[d][l][q] / l_run[l][q]
            if (l & 15) == 0:
                lse[r] = m_run[l][q] + math.log(l_run[l][q])
    return out, lse


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("D", [64, 128])
def test_fmha_kernel_index_math(causal, D):
    S = 64
    rng = np.random.default_rng(0)
    Q = rng.standard_normal((S, D))
    K = rng.standard_normal((S, D))
    V = rng.standard_normal((S, D))
    scale = 1.0 / math.sqrt(D)

    # eager reference
    s = Q @ K.T * scale
    if causal:
        s = np.where(np.triu(np.ones((S, S), bool), 1), -np.inf, s)
    mx = s.max(1, keepdims=True)
    p = np.exp(s - mx)
    ref = (p / p.sum(1, keepdims=True)) @ V
    ref_lse = (mx[:, 0] + np.log(p.sum(1)))

    for q0 in (0, 16, 48):  # different waves, incl. the causal diagonal
        out, lse = simulate_wave(Q, K, V, q0, S, D, causal, scale)
        np.testing.assert_allclose(out, ref[q0:q0 + 16], rtol=1e-6, atol=1e-9)
        np.testing.assert_allclose(lse, ref_lse[q0:q0 + 16], rtol=1e-6, atol=1e-9)
