"""L1 cross-product sweep harness smoke (2 combos, tiny shapes — the full
sweep runs via scripts/run_l1_sweep.py on a GPU box)."""

import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "scripts"))


@pytest.mark.parametrize("combo", [("O1", "dynamic", None), ("O2", "128.0", "True")])
def test_l1_sweep_combo(combo):
    from run_l1_sweep import run_combo

    opt, ls, kbn = combo
    res = run_combo(opt, ls, kbn, iters=5, batch=2, image_size=32,
                    timeout=300, lr=0.02)
    assert res["ok"], res
