"""Probe: every unique ResNet-50 conv shape, NHWC bf16, fwd/dgrad/wgrad
separately — hunts the shapes whose wgrad falls back to MIOpen's naive
kernel (the 6x channels_last bench regression). Set MIOPEN_FIND_MODE via
env to compare find modes."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=5, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0


# (Cin, Cout, k, stride, H) — unique ResNet-50 convs at 224x224, N=128
CONVS = [
    (3, 64, 7, 2, 224),
    (64, 64, 1, 1, 56), (64, 64, 3, 1, 56), (64, 256, 1, 1, 56),
    (256, 64, 1, 1, 56), (256, 128, 1, 1, 56), (128, 128, 3, 2, 56),
    (128, 512, 1, 1, 28), (256, 512, 1, 2, 56), (512, 128, 1, 1, 28),
    (128, 128, 3, 1, 28), (512, 256, 1, 1, 28), (256, 256, 3, 2, 28),
    (256, 1024, 1, 1, 14), (512, 1024, 1, 2, 28), (1024, 256, 1, 1, 14),
    (256, 256, 3, 1, 14), (1024, 512, 1, 1, 14), (512, 512, 3, 2, 14),
    (512, 2048, 1, 1, 7), (1024, 2048, 1, 2, 14), (2048, 512, 1, 1, 7),
    (512, 512, 3, 1, 7),
]


def main():
    N = 128
    torch.backends.cudnn.benchmark = True
    print(f"MIOPEN_FIND_MODE={os.environ.get('MIOPEN_FIND_MODE', '<default>')}")
    total = {"fwd": 0.0, "dgrad": 0.0, "wgrad": 0.0}
    for (ci, co, k, s, h) in CONVS:
        pad = k // 2 if k > 1 else 0
        x = torch.randn(N, ci, h, h, device="cuda", dtype=torch.bfloat16
                        ).to(memory_format=torch.channels_last)
        w = torch.randn(co, ci, k, k, device="cuda", dtype=torch.bfloat16
                        ).to(memory_format=torch.channels_last)
        y = torch.nn.functional.conv2d(x, w, stride=s, padding=pad)
        dy = torch.randn_like(y)

        t_f = timeit(lambda: torch.nn.functional.conv2d(x, w, stride=s, padding=pad))
        t_d = timeit(lambda: torch.ops.aten.convolution_backward(
            dy, x, w, None, [s, s], [pad, pad], [1, 1], False, [0, 0], 1,
            [True, False, False])) if ci > 3 else 0.0
        t_w = timeit(lambda: torch.ops.aten.convolution_backward(
            dy, x, w, None, [s, s], [pad, pad], [1, 1], False, [0, 0], 1,
            [False, True, False]))
        total["fwd"] += t_f
        total["dgrad"] += t_d
        total["wgrad"] += t_w
        flag = "  <<< SLOW" if max(t_f, t_d, t_w) > 2.0 else ""
        print(f"C{ci:4d}->{co:4d} k{k} s{s} {h:3d}x{h:<3d}: "
              f"fwd {t_f:7.2f}  dgrad {t_d:7.2f}  wgrad {t_w:7.2f} ms{flag}",
              flush=True)
    print(f"TOTALS: fwd {total['fwd']:.2f}  dgrad {total['dgrad']:.2f}  "
          f"wgrad {total['wgrad']:.2f} ms")


if __name__ == "__main__":
    main()
