"""Lane-level CPU simulation of the one-pass GroupNorm forward's VW-vector
partitioning (csrc/group_norm.hip gn_fwd_onepass_kernel:261-308): grid-
strided vector index vi -> (row r = vi/vpr, channel cc = (vi%vpr)*VW, j<VW)
must visit every (row, in-group channel) element exactly once, and the
simulated stats must match numpy for NHWC data."""

import numpy as np
import pytest

GN_BLOCK = 256


def simulate_onepass_stats(x_group, VW, block=GN_BLOCK):
    """x_group: [HW, cpg] one (n, g) slab. Mirrors pass A."""
    HW, cpg = x_group.shape
    assert cpg % VW == 0
    vpr = cpg // VW
    nvec = HW * vpr
    visits = np.zeros((HW, cpg), dtype=np.int64)
    total_sum = 0.0
    total_sq = 0.0
    for tid in range(block):
        s = sq = 0.0
        vi = tid
        while vi < nvec:
            r = vi // vpr
            cc = (vi % vpr) * VW
            for j in range(VW):
                f = float(x_group[r, cc + j])
                s += f
                sq += f * f
                visits[r, cc + j] += 1
            vi += block
        total_sum += s
        total_sq += sq
    count = HW * cpg
    mu = total_sum / count
    var = max(total_sq / count - mu * mu, 0.0)
    return mu, var, visits


@pytest.mark.parametrize("HW,cpg,VW", [
    (49, 8, 8),     # odd spatial, one vector per row at VW=8
    (56 * 56, 16, 8),
    (17, 32, 4),
    (100, 10, 2),   # cpg not a power of two
    (3, 512, 8),    # fewer rows than block (idle lanes)
    (640, 4, 1),    # scalar fallback VW=1
])
def test_onepass_vector_partition_and_stats(HW, cpg, VW):
    rng = np.random.default_rng(HW + cpg * 13 + VW)
    x = rng.normal(1.5, 2.0, size=(HW, cpg))
    mu, var, visits = simulate_onepass_stats(x, VW)
    assert (visits == 1).all(), "every (row, channel) read exactly once"
    np.testing.assert_allclose(mu, x.mean(), rtol=1e-12)
    np.testing.assert_allclose(var, x.var(), rtol=1e-9, atol=1e-12)
