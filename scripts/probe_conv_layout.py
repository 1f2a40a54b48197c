"""Probe: MIOpen conv speed NCHW vs channels_last (bf16 + fp32) at ResNet-50
shapes, fwd and fwd+bwd — diagnoses the 6x NHWC bench regression (597 vs
3568 img/s) seen in round 2."""

import os
import sys
import time

os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0


SHAPES = [
    # (N, Cin, H, W, Cout, k, stride, pad) — ResNet-50 stages
    (128, 3, 224, 224, 64, 7, 2, 3),
    (128, 64, 56, 56, 64, 1, 1, 0),
    (128, 64, 56, 56, 64, 3, 1, 1),
    (128, 256, 56, 56, 128, 1, 1, 0),
    (128, 128, 28, 28, 128, 3, 2, 1),
    (128, 512, 28, 28, 256, 1, 1, 0),
    (128, 1024, 14, 14, 512, 1, 1, 0),
    (128, 512, 7, 7, 512, 3, 1, 1),
]


def main():
    torch.backends.cudnn.benchmark = True
    for dtype in (torch.bfloat16,):
        print(f"== dtype {dtype} ==", flush=True)
        for (N, Ci, H, W, Co, k, s, p) in SHAPES:
            res = {}
            for layout in ("nchw", "nhwc"):
                x = torch.randn(N, Ci, H, W, device="cuda", dtype=dtype,
                                requires_grad=True)
                w = torch.randn(Co, Ci, k, k, device="cuda", dtype=dtype,
                                requires_grad=True)
                if layout == "nhwc":
                    x = x.detach().to(memory_format=torch.channels_last).requires_grad_(True)
                    w = w.detach().to(memory_format=torch.channels_last).requires_grad_(True)

                def fwd():
                    return torch.nn.functional.conv2d(x, w, stride=s, padding=p)

                y = fwd()
                g = torch.randn_like(y)

                def fb():
                    out = torch.nn.functional.conv2d(x, w, stride=s, padding=p)
                    gx, gw = torch.autograd.grad(out, (x, w), g)
                    return gx

                res[layout] = (timeit(fwd), timeit(fb))
            (f_nchw, b_nchw), (f_nhwc, b_nhwc) = res["nchw"], res["nhwc"]
            print(f"N{N} C{Ci}->{Co} {H}x{W} k{k} s{s}: "
                  f"fwd nchw {f_nchw:7.2f} nhwc {f_nhwc:7.2f} ms ({f_nchw/f_nhwc:4.2f}x)  "
                  f"f+b nchw {b_nchw:7.2f} nhwc {b_nhwc:7.2f} ms ({b_nchw/b_nhwc:4.2f}x)",
                  flush=True)


if __name__ == "__main__":
    main()
