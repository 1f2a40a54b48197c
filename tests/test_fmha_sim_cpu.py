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


# ---------------- backward kernel designs (simulator-first) ----------------
# Kernel A (dQ): one wave per 16 q rows, loops kv tiles — same orientation as
# the forward. Kernel B (dK/dV): one wave per 16 kv rows, loops q tiles with
# the transposed fragment roles. Both recompute P from (q, k, lse) and use
# delta = rowsum(dO * O).


def simulate_dq_wave(Q, K, V, dO, lse, delta, q0, S, D, causal, scale):
    NK = D // 32
    ND = D // 16
    aq = [[[Q[q0 + (l & 15)][c * 32 + (l >> 4) * 8 + j] for j in range(8)]
           for l in range(64)] for c in range(NK)]
    ado = [[[dO[q0 + (l & 15)][c * 32 + (l >> 4) * 8 + j] for j in range(8)]
            for l in range(64)] for c in range(NK)]
    dq_acc = np.zeros((ND, 64, 4))
    kv_end = min(S, ((q0 + FM_ROWS - 1) // FM_BN + 1) * FM_BN) if causal else S
    for kv0 in range(0, kv_end, FM_BN):
        # recompute P = exp(scale*QK^T - lse) for the two 16-col halves
        p_val = np.zeros((2, 64, 4))
        dp_val = np.zeros((2, 64, 4))
        for j in range(2):
            s = np.zeros((64, 4))
            for c in range(NK):
                bk = [[K[kv0 + j * 16 + (l & 15)][c * 32 + (l >> 4) * 8 + jj]
                       for jj in range(8)] for l in range(64)]
                s += mfma_16x16x32(aq[c], bk)
            # dP = dO V^T for the same 16x16 half: B = V^T (vector rows of V)
            dp = np.zeros((64, 4))
            for c in range(NK):
                bvt = [[V[kv0 + j * 16 + (l & 15)][c * 32 + (l >> 4) * 8 + jj]
                        for jj in range(8)] for l in range(64)]
                dp += mfma_16x16x32(ado[c], bvt)
            for l in range(64):
                for q in range(4):
                    row_g = q0 + (l >> 4) * 4 + q
                    col_g = kv0 + j * 16 + (l & 15)
                    v = s[l][q] * scale
                    if causal and col_g > row_g:
                        p_val[j][l][q] = 0.0
                    else:
                        p_val[j][l][q] = math.exp(v - lse[row_g])
                    dp_val[j][l][q] = dp[l][q]
        # dS = P * (dP - delta_row) -> LDS bounce -> A-frag; dQ += dS K * scale
        pbuf = np.zeros(FM_ROWS * FM_BN)
        for l in range(64):
            for j in range(2):
                for q in range(4):
                    row_g = q0 + (l >> 4) * 4 + q
                    ds = p_val[j][l][q] * (dp_val[j][l][q] - delta[row_g])
                    pbuf[((l >> 4) * 4 + q) * FM_BN + j * 16 + (l & 15)] = ds
        a_ds = [[pbuf[(l & 15) * FM_BN + (l >> 4) * 8 + j] for j in range(8)]
                for l in range(64)]
        for d in range(ND):
            bK = [[K[kv0 + (l >> 4) * 8 + jj][d * 16 + (l & 15)] for jj in range(8)]
                  for l in range(64)]
            dq_acc[d] += mfma_16x16x32(a_ds, bK)
    dq = np.zeros((FM_ROWS, D))
    for l in range(64):
        for q in range(4):
            for d in range(ND):
                dq[(l >> 4) * 4 + q][d * 16 + (l & 15)] = dq_acc[d][l][q] * scale
    return dq


def simulate_dkv_wave(Q, K, V, dO, lse, delta, kv0, S, D, causal, scale):
    """One wave per 16 kv rows; transposed fragment roles (S^T = K Q^T)."""
    NK = D // 32
    ND = D // 16
    ak = [[[K[kv0 + (l & 15)][c * 32 + (l >> 4) * 8 + j] for j in range(8)]
           for l in range(64)] for c in range(NK)]
    av = [[[V[kv0 + (l & 15)][c * 32 + (l >> 4) * 8 + j] for j in range(8)]
           for l in range(64)] for c in range(NK)]
    dv_acc = np.zeros((ND, 64, 4))
    dk_acc = np.zeros((ND, 64, 4))
    # causal: q row i attends kv col j iff j <= i, so kv rows kv0..kv0+15 are
    # seen only by q rows >= kv0 — start the q loop at the tile containing kv0
    q_start = (kv0 // FM_BN) * FM_BN if causal else 0
    for q0 in range(q_start, S, FM_BN):
        pT = np.zeros((2, 64, 4))   # P^T halves: rows kv, cols q
        dpT = np.zeros((2, 64, 4))
        for j in range(2):
            sT = np.zeros((64, 4))
            dpt = np.zeros((64, 4))
            for c in range(NK):
                bq = [[Q[q0 + j * 16 + (l & 15)][c * 32 + (l >> 4) * 8 + jj]
                       for jj in range(8)] for l in range(64)]
                sT += mfma_16x16x32(ak[c], bq)
                bdo = [[dO[q0 + j * 16 + (l & 15)][c * 32 + (l >> 4) * 8 + jj]
                        for jj in range(8)] for l in range(64)]
                dpt += mfma_16x16x32(av[c], bdo)
            for l in range(64):
                for q in range(4):
                    kv_g = kv0 + (l >> 4) * 4 + q
                    q_g = q0 + j * 16 + (l & 15)
                    v = sT[l][q] * scale
                    if causal and kv_g > q_g:
                        pT[j][l][q] = 0.0
                    else:
                        pT[j][l][q] = math.exp(v - lse[q_g])
                    dpT[j][l][q] = dpt[l][q]
        # dV += P^T dO ; dK += dS^T Q (both via the LDS bounce to A-frags)
        pbuf = np.zeros(FM_ROWS * FM_BN)
        dsbuf = np.zeros(FM_ROWS * FM_BN)
        for l in range(64):
            for j in range(2):
                for q in range(4):
                    q_g = q0 + j * 16 + (l & 15)
                    idx = ((l >> 4) * 4 + q) * FM_BN + j * 16 + (l & 15)
                    pbuf[idx] = pT[j][l][q]
                    dsbuf[idx] = pT[j][l][q] * (dpT[j][l][q] - delta[q_g])
        a_p = [[pbuf[(l & 15) * FM_BN + (l >> 4) * 8 + j] for j in range(8)]
               for l in range(64)]
        a_ds = [[dsbuf[(l & 15) * FM_BN + (l >> 4) * 8 + j] for j in range(8)]
                for l in range(64)]
        for d in range(ND):
            b_do = [[dO[q0 + (l >> 4) * 8 + jj][d * 16 + (l & 15)] for jj in range(8)]
                    for l in range(64)]
            dv_acc[d] += mfma_16x16x32(a_p, b_do)
            b_q = [[Q[q0 + (l >> 4) * 8 + jj][d * 16 + (l & 15)] for jj in range(8)]
                   for l in range(64)]
            dk_acc[d] += mfma_16x16x32(a_ds, b_q)
    dv = np.zeros((FM_ROWS, D))
    dk = np.zeros((FM_ROWS, D))
    for l in range(64):
        for q in range(4):
            r = (l >> 4) * 4 + q
            for d in range(ND):
                dv[r][d * 16 + (l & 15)] = dv_acc[d][l][q]
                dk[r][d * 16 + (l & 15)] = dk_acc[d][l][q] * scale
    return dk, dv


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("D", [64, 128])
def test_fmha_backward_kernel_index_math(causal, D):
    S = 64
    rng = np.random.default_rng(1)
    Q = rng.standard_normal((S, D))
    K = rng.standard_normal((S, D))
    V = rng.standard_normal((S, D))
    dO = rng.standard_normal((S, D))
    scale = 1.0 / math.sqrt(D)

    # torch autograd reference
    qt = torch.tensor(Q, requires_grad=True)
    kt = torch.tensor(K, requires_grad=True)
    vt = torch.tensor(V, requires_grad=True)
    s = (qt @ kt.T) * scale
    if causal:
        s = s.masked_fill(torch.triu(torch.ones(S, S, dtype=torch.bool), 1),
                          float("-inf"))
    lse_t = torch.logsumexp(s, -1)
    out_t = torch.softmax(s, -1) @ vt
    out_t.backward(torch.tensor(dO))
    lse = lse_t.detach().numpy()
    delta = (dO * out_t.detach().numpy()).sum(-1)

    for q0 in (0, 48):
        dq = simulate_dq_wave(Q, K, V, dO, lse, delta, q0, S, D, causal, scale)
        np.testing.assert_allclose(dq, qt.grad.numpy()[q0:q0 + 16],
                                   rtol=1e-6, atol=1e-8)
    for kv0 in (0, 32):
        dk, dv = simulate_dkv_wave(Q, K, V, dO, lse, delta, kv0, S, D, causal, scale)
        np.testing.assert_allclose(dv, vt.grad.numpy()[kv0:kv0 + 16],
                                   rtol=1e-6, atol=1e-8)
        np.testing.assert_allclose(dk, kt.grad.numpy()[kv0:kv0 + 16],
                                   rtol=1e-6, atol=1e-8)
