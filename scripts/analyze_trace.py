"""Summarize a rocprofv3 runtime trace: GPU busy/idle in a steady-state
window and which HIP API calls the host was inside during the idle gaps."""

import csv
import sys
from collections import defaultdict


def main(d, prefix, marker="AdamFunctor"):
    kt = list(csv.DictReader(open(f"{d}/{prefix}_kernel_trace.csv")))
    evs = sorted((int(r["Start_Timestamp"]), int(r["End_Timestamp"]), r["Kernel_Name"]) for r in kt)
    adam = [e for s, e, n in evs if marker in n]
    # optimizer launches cluster at step ends; keep the last of each cluster
    step_ends = []
    for t in adam:
        if not step_ends or t - step_ends[-1] > 5_000_000:  # >5 ms apart
            step_ends.append(t)
        else:
            step_ends[-1] = t
    if len(step_ends) >= 2:
        w0, w1 = step_ends[-2], step_ends[-1]
    else:
        w0, w1 = evs[0][0], evs[-1][1]
    window = [(s, e, n) for s, e, n in evs if s >= w0 and e <= w1]
    span = w1 - w0
    busy = 0
    last = 0
    gaps = []
    for s, e, n in window:
        if s > last:
            if last:
                gaps.append((last, s, n))
            busy += e - s
            last = e
        elif e > last:
            busy += e - last
            last = e
    idle = sum(g1 - g0 for g0, g1, _ in gaps)
    print(f"1-step window: span {span/1e6:.1f} ms busy {busy/1e6:.1f} idle {idle/1e6:.1f} "
          f"dispatches {len(window)}")

    # attribute idle gaps to host API activity
    api = []
    with open(f"{d}/{prefix}_hip_api_trace.csv") as f:
        for r in csv.DictReader(f):
            api.append((int(r["Start_Timestamp"]), int(r["End_Timestamp"]), r["Function"]))
    api.sort()

    gaps.sort(key=lambda g: g[0] - g[1])  # longest first
    for g0, g1, nk in gaps[:6]:
        print(f"\n-- gap {(g1-g0)/1e6:.2f} ms before {nk[:70]}")
        during = defaultdict(float)
        cnt = defaultdict(int)
        for s, e, fn in api:
            if e < g0 or s > g1:
                continue
            during[fn] += (min(e, g1) - max(s, g0)) / 1e6
            cnt[fn] += 1
        for fn, t in sorted(during.items(), key=lambda kv: -kv[1])[:6]:
            print(f"   {t:8.2f} ms x{cnt[fn]:5} {fn}")

    # total API time in window
    tot = defaultdict(float)
    cc = defaultdict(int)
    for s, e, fn in api:
        if e < w0 or s > w1:
            continue
        tot[fn] += (min(e, w1) - max(s, w0)) / 1e6
        cc[fn] += 1
    print("\ntop HIP APIs in window:")
    for fn, t in sorted(tot.items(), key=lambda kv: -kv[1])[:12]:
        print(f"  {t:8.2f} ms x{cc[fn]:6} {fn}")

    # per-kernel-family GPU time inside the steady-state window
    fam = defaultdict(float)
    fc = defaultdict(int)
    for s, e, n in window:
        key = ("naive_conv" if "naive_conv" in n else
               "ck_conv" if "ck::" in n else
               "Tensile" if "Cijk" in n else
               "miopen" if ("miopen" in n.lower() or "Winograd" in n or "Conv" in n) else
               "welford/bn" if ("welford" in n or "bn_" in n or "reduce_bn" in n
                                or "batchnorm" in n.lower()) else
               "sgd/adam/mta" if ("multi_tensor" in n or "Functor" in n) else
               "pool" if "pool" in n.lower() else
               "elementwise/copy" if ("elementwise" in n or "copy" in n or "Fill" in n) else
               "fmha" if "fmha" in n else
               "softmax/ln" if ("softmax" in n or "ln_" in n or "xentropy" in n) else
               "other")
        fam[key] += (e - s) / 1e6
        fc[key] += 1
    print("\nkernel families in window (GPU ms):")
    for k, t in sorted(fam.items(), key=lambda kv: -kv[1]):
        print(f"  {t:8.2f} ms x{fc[k]:6} {k}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2], sys.argv[3] if len(sys.argv) > 3 else "AdamFunctor")
