"""A/B the fused residual-add+LayerNorm kernel against the eager
composition at transformer shapes (fwd and fwd+bwd)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from apex_amd.normalization import FusedLayerNorm, fused_add_layer_norm_affine


def t(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    for rows, hidden in ((16384, 768), (65536, 768), (8192, 1024)):
        for dtype in (torch.bfloat16,):
            ln = FusedLayerNorm(hidden).cuda().to(dtype)
            x = torch.randn(rows, hidden, device="cuda", dtype=dtype)
            d = torch.randn_like(x)

            def eager():
                z = x + d
                return ln(z), z

            def fused():
                return fused_add_layer_norm_affine(x, d, ln.weight, ln.bias,
                                                   (hidden,), ln.eps)

            te, tf = t(eager), t(fused)
            y0, z0 = eager()
            y1, z1 = fused()
            err = (y0.float() - y1.float()).abs().max().item()
            gb = rows * hidden * dtype.itemsize
            # eager: add reads 2x+writes z, LN reads z+writes y = 5 passes;
            # fused: reads x,d writes z,y = 4 passes
            print(f"[{rows}x{hidden} {dtype}] eager {te:.3f} ms "
                  f"({5*gb/te/1e6:.0f} GB/s)  fused {tf:.3f} ms "
                  f"({4*gb/tf/1e6:.0f} GB/s)  speedup {te/tf:.2f}x  maxerr {err:.2e}")

            # training step shape: fwd+bwd with a downstream grad on z
            xg = x.clone().requires_grad_(True)
            dg = d.clone().requires_grad_(True)
            go = torch.randn_like(x)
            gz = torch.randn_like(x)

            def eager_fb():
                z = xg + dg
                y = ln(z)
                (y * go + z * gz).sum().backward()
                xg.grad = dg.grad = None
                ln.weight.grad = ln.bias.grad = None

            def fused_fb():
                y, z = fused_add_layer_norm_affine(xg, dg, ln.weight, ln.bias,
                                                   (hidden,), ln.eps)
                (y * go + z * gz).sum().backward()
                xg.grad = dg.grad = None
                ln.weight.grad = ln.bias.grad = None

            tef, tff = t(eager_fb, 20), t(fused_fb, 20)
            print(f"    fwd+bwd: eager {tef:.3f} ms  fused {tff:.3f} ms  "
                  f"speedup {tef/tff:.2f}x")


if __name__ == "__main__":
    main()
