"""Lane-level CPU simulation of the cooperative small-C SyncBN kernels
(csrc/syncbn.hip: welford_nhwc_kernel / reduce_bn_partials_kernel NHWC
small-C forms). Verifies the R-row-lanes-per-channel partitioning: every
(row, channel) element is visited exactly once across lanes and grid-y
splits, and the merged stats match numpy. Guards future index-math edits
the GPU numerics tests would only catch after a hardware run."""

import numpy as np
import pytest

BN_BLOCK = 256


def welford_add(x, mean, m2, count):
    count += 1.0
    delta = x - mean
    mean += delta / count
    m2 += delta * (x - mean)
    return mean, m2, count


def welford_combine(mean, m2, count, mean_b, m2_b, count_b):
    if count_b == 0.0:
        return mean, m2, count
    if count == 0.0:
        return mean_b, m2_b, count_b
    tot = count + count_b
    delta = mean_b - mean
    mean = mean + delta * (count_b / tot)
    m2 = m2 + m2_b + delta * delta * (count * count_b / tot)
    return mean, m2, tot


def simulate_welford_nhwc_small_c(x2d, S):
    """x2d: [rows, C] fp32. Returns (mean[C], var[C], visit-count map)."""
    rows, C = x2d.shape
    assert C < BN_BLOCK and BN_BLOCK % C == 0
    R = BN_BLOCK // C
    per = (rows + S - 1) // S
    part = {}  # (c, s) -> (mean, m2, count)
    visits = np.zeros((rows, C), dtype=np.int64)
    for s in range(S):
        lo, hi = s * per, min(s * per + per, rows)
        lanes = {}
        for tid in range(BN_BLOCK):
            c, k = tid % C, tid // C
            mean = m2 = count = 0.0
            r = lo + k
            while r < hi:
                mean, m2, count = welford_add(float(x2d[r, c]), mean, m2, count)
                visits[r, c] += 1
                r += R
            lanes[tid] = (mean, m2, count)
        for tid in range(C):  # k == 0 lanes merge their channel
            c = tid
            mean, m2, count = lanes[tid]
            for kk in range(1, R):
                wm, wm2, wc = lanes[kk * C + c]
                mean, m2, count = welford_combine(mean, m2, count, wm, wm2, wc)
            part[(c, s)] = (mean, m2, count)
    # welford_merge_kernel
    mean_out = np.zeros(C)
    var_out = np.zeros(C)
    for c in range(C):
        mean = m2 = count = 0.0
        for s in range(S):
            wm, wm2, wc = part[(c, s)]
            mean, m2, count = welford_combine(mean, m2, count, wm, wm2, wc)
        mean_out[c] = mean
        var_out[c] = m2 / count if count > 0 else 0.0
    return mean_out, var_out, visits


def simulate_reduce_bn_small_c(dy2d, x2d, mean, S):
    """reduce_bn_partials_kernel NHWC small-C: returns (sum_dy, sum_dy_xmu)."""
    rows, C = x2d.shape
    R = BN_BLOCK // C
    per = (rows + S - 1) // S
    s1_out = np.zeros(C)
    s2_out = np.zeros(C)
    visits = np.zeros((rows, C), dtype=np.int64)
    for s in range(S):
        lo, hi = s * per, min(s * per + per, rows)
        lanes = np.zeros((BN_BLOCK, 2))
        for tid in range(BN_BLOCK):
            c, k = tid % C, tid // C
            s1 = s2 = 0.0
            r = lo + k
            while r < hi:
                d = float(dy2d[r, c])
                s1 += d
                s2 += d * (float(x2d[r, c]) - mean[c])
                visits[r, c] += 1
                r += R
            lanes[tid] = (s1, s2)
        for c in range(C):
            s1, s2 = lanes[c]
            for kk in range(1, R):
                s1 += lanes[kk * C + c, 0]
                s2 += lanes[kk * C + c, 1]
            s1_out[c] += s1
            s2_out[c] += s2
    return s1_out, s2_out, visits


@pytest.mark.parametrize("rows,C,S", [
    (37, 4, 1),     # rows not divisible by R, single split
    (1000, 64, 3),  # ResNet-ish C=64, odd split count
    (256, 8, 5),    # per-split remainder hits the last split
    (5, 128, 2),    # fewer rows than row-lanes (idle lanes)
    (64, 16, 64),   # more splits than needed (empty splits)
])
def test_cooperative_welford_partition_and_numerics(rows, C, S):
    rng = np.random.default_rng(rows * 1000 + C + S)
    x = rng.normal(2.0, 3.0, size=(rows, C))
    mean, var, visits = simulate_welford_nhwc_small_c(x, S)
    assert (visits == 1).all(), "each (row, channel) must be read exactly once"
    np.testing.assert_allclose(mean, x.mean(0), rtol=1e-12, atol=1e-12)
    np.testing.assert_allclose(var, x.var(0), rtol=1e-10, atol=1e-12)


@pytest.mark.parametrize("rows,C,S", [(37, 4, 1), (1000, 64, 3), (5, 128, 2)])
def test_cooperative_reduce_partition_and_numerics(rows, C, S):
    rng = np.random.default_rng(rows + C * 7 + S)
    x = rng.normal(size=(rows, C))
    dy = rng.normal(size=(rows, C))
    mu = x.mean(0)
    s1, s2, visits = simulate_reduce_bn_small_c(dy, x, mu, S)
    assert (visits == 1).all()
    np.testing.assert_allclose(s1, dy.sum(0), rtol=1e-10, atol=1e-10)
    np.testing.assert_allclose(s2, (dy * (x - mu)).sum(0), rtol=1e-9, atol=1e-9)
