"""Lane-level CPU simulation of the fused_norm wave-per-row reduction
(csrc/fused_norm.hip ln_fwd_wave_kernel:79-112): lane-local streaming
Welford over the NPACK*W strided elements, then the 6-step shfl_xor Chan
merge. Verifies element coverage, the merge tree's algebra (all 64 lanes
converge to the full-row stats), and the RMS identity
sum(x^2) = m2 + n*mean^2."""

import numpy as np
import pytest

WAVE = 64


class Welford:
    def __init__(self):
        self.mean = 0.0
        self.m2 = 0.0
        self.count = 0.0

    def add(self, x):
        self.count += 1.0
        d = x - self.mean
        self.mean += d / self.count
        self.m2 += d * (x - self.mean)

    def combine(self, mb, m2b, nb):
        if nb == 0.0:
            return
        if self.count == 0.0:
            self.mean, self.m2, self.count = mb, m2b, nb
            return
        tot = self.count + nb
        d = mb - self.mean
        self.mean += d * (nb / tot)
        self.m2 += m2b + d * d * (self.count * nb / tot)
        self.count = tot


def simulate_wave_row(row, W):
    """row: [n2] fp32; W: vector width. Returns per-lane merged stats."""
    n2 = row.shape[0]
    npack = (n2 + WAVE * W - 1) // (WAVE * W)
    lanes = [Welford() for _ in range(WAVE)]
    visits = np.zeros(n2, dtype=np.int64)
    for lane in range(WAVE):
        for k in range(npack):
            i = (k * WAVE + lane) * W
            if i < n2:
                for j in range(W):
                    lanes[lane].add(float(row[i + j]))
                    visits[i + j] += 1
    # shfl_xor butterfly: every lane ends with the full reduction
    for off in [32, 16, 8, 4, 2, 1]:
        snapshot = [(w.mean, w.m2, w.count) for w in lanes]
        for lane in range(WAVE):
            mb, m2b, nb = snapshot[lane ^ off]
            lanes[lane].combine(mb, m2b, nb)
    return lanes, visits


@pytest.mark.parametrize("n2,W", [
    (768, 8),      # bert hidden, bf16 pack (one pack exactly: 64*8=512 < 768 -> 2 packs)
    (1024, 8),
    (256, 4),      # fp32 pack width
    (2048, 4),
    (64, 1),       # tiny row, most lanes idle in later packs
])
def test_wave_welford_partition_and_merge(n2, W):
    assert n2 % W == 0, "kernel requires vector-aligned rows"
    rng = np.random.default_rng(n2 + W)
    row = rng.normal(3.0, 2.0, size=n2)
    lanes, visits = simulate_wave_row(row, W)
    assert (visits == 1).all(), "every element read exactly once"
    for w in lanes:  # butterfly leaves the SAME totals in every lane
        assert w.count == n2
        np.testing.assert_allclose(w.mean, row.mean(), rtol=1e-12)
        np.testing.assert_allclose(w.m2, ((row - row.mean()) ** 2).sum(), rtol=1e-9)
    # RMS identity used by the kernel: sum(x^2) = m2 + n*mean^2
    w = lanes[0]
    np.testing.assert_allclose(w.m2 + w.count * w.mean ** 2, (row ** 2).sum(),
                               rtol=1e-9)
