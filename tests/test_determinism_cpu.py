"""Bitwise run-to-run determinism of the amp training harness (the
reference L1 contract: fixed seed + --deterministic ⇒ identical loss
records). CPU tier; the gpu tier repeats this on hardware where it actually
exercises the deterministic two-stage kernel reductions."""

import json
import os
import subprocess
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HARNESS = os.path.join(REPO, "examples", "imagenet", "main_amp.py")


def run_harness(opt_level, iters=4, batch=2, image=32):
    with tempfile.NamedTemporaryFile(suffix=".json", delete=False) as f:
        out = f.name
    cmd = [sys.executable, HARNESS, "--opt-level", opt_level, "--iters", str(iters),
           "--batch-size", str(batch), "--image-size", str(image),
           "--lr", "0.02", "--deterministic", "--json-out", out]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-1500:]
    try:
        with open(out) as f:
            return [rec["loss"] for rec in json.load(f)["records"]]
    finally:
        os.unlink(out)


def test_harness_bitwise_deterministic_o1():
    a = run_harness("O1")
    b = run_harness("O1")
    assert a == b  # bitwise-equal floats


def test_harness_bitwise_deterministic_o2():
    a = run_harness("O2")
    b = run_harness("O2")
    assert a == b
