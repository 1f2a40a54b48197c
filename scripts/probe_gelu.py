"""Probe: which GELU does hipBLASLt's epilogue implement (erf vs tanh)?
And how do the fused MLP grads compare against same-dtype torch references?"""

import torch

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import apex_amd._fused_dense as fd


def main():
    torch.manual_seed(0)
    n = 256
    x = torch.linspace(-4, 4, n, device="cuda", dtype=torch.float32).reshape(1, n).t().contiguous()  # [n,1]
    w = torch.ones(1, 1, device="cuda")
    b = torch.zeros(1, device="cuda")
    w2 = torch.ones(1, 1, device="cuda")
    b2 = torch.zeros(1, device="cuda")
    o1, o2, gelu_in = fd.linear_gelu_linear_forward(x, w, b, w2, b2)
    y = o1.flatten()
    xin = x.flatten()
    erf = torch.nn.functional.gelu(xin)
    tanh = torch.nn.functional.gelu(xin, approximate="tanh")
    print("max |y-erf| :", (y - erf).abs().max().item())
    print("max |y-tanh|:", (y - tanh).abs().max().item())

    # dgelu probe via linear_gelu_linear_backward
    gi = xin.reshape(-1, 1)
    dy = torch.ones_like(gi)
    out = fd.linear_gelu_linear_backward(gi, gi, o1, w, w2, dy)
    dx = out[0].flatten()
    xg = xin.clone().requires_grad_(True)
    torch.nn.functional.gelu(xg).sum().backward()
    derf = xg.grad.clone()
    xg2 = xin.clone().requires_grad_(True)
    torch.nn.functional.gelu(xg2, approximate="tanh").sum().backward()
    dtanh = xg2.grad.clone()
    print("max |dx-derf| :", (dx - derf).abs().max().item())
    print("max |dx-dtanh|:", (dx - dtanh).abs().max().item())


if __name__ == "__main__":
    main()
