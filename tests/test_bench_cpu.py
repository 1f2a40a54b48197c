"""End-to-end bench.py flow validation on CPU/gloo (2 ranks via torchrun) —
de-risks the driver's multi-GPU invocation: rendezvous, DDP wrap, barriers,
MAX-over-ranks, single JSON line from rank 0."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_bench_cpu_gloo_two_ranks():
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1", "--master-port", "29611",
        "bench.py", "--device", "cpu", "--batch", "2", "--image-size", "64",
        "--steps", "2", "--warmup", "1", "--skip-adam-bench",
    ]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True, timeout=560)
    assert out.returncode == 0, f"bench failed:\n{out.stdout[-2000:]}\n{out.stderr[-2000:]}"
    json_lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, f"expected exactly one JSON line, got: {json_lines}"
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 2
    assert rec["value"] > 0
    assert rec["config"]["parallelism"] == "dp2"
    for key in ["metric", "unit", "steps", "warmup", "ms_per_step", "higher_is_better",
                "scaling", "vs_baseline", "dtype", "data", "config"]:
        assert key in rec


@pytest.mark.timeout(600)
def test_bench_cpu_gloo_two_ranks_bert():
    # transformer path at world>1: amp O2 bf16 + DDP bucketed all-reduce +
    # FusedAdam (eager — capture is disabled when distributed)
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1", "--master-port", "29612",
        "bench.py", "--device", "cpu", "--model", "bert", "--seq-len", "64",
        "--batch", "2", "--steps", "2", "--warmup", "1", "--skip-adam-bench",
    ]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True, timeout=560)
    assert out.returncode == 0, f"bench failed:\n{out.stdout[-2000:]}\n{out.stderr[-2000:]}"
    json_lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 2
    assert rec["value"] > 0
    assert rec["config"]["model"] == "bert-base"
    assert rec["config"]["hipgraph_step"] is False


@pytest.mark.timeout(300)
def test_dcgan_example_cpu_smoke():
    # the dcgan example (reference examples/dcgan/main_amp.py twin: two
    # optimizers sharing one amp.initialize, multi-loss scale_loss) must run
    # end to end on CPU
    cmd = [sys.executable, "examples/dcgan/main_amp.py",
           "--iters", "2", "--batch-size", "4", "--opt-level", "O1"]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, f"dcgan failed:\n{out.stdout[-1500:]}\n{out.stderr[-1500:]}"


@pytest.mark.timeout(300)
@pytest.mark.parametrize("script", [
    "examples/simple/distributed/distributed_data_parallel.py",
    "examples/simple/distributed/zero2_distributed_fused_adam.py",
])
def test_simple_distributed_examples_two_ranks(script):
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "2",
           "--master-addr", "127.0.0.1", "--master-port", "29613",
           script]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, f"{script} failed:\n{out.stdout[-1500:]}\n{out.stderr[-1500:]}"
