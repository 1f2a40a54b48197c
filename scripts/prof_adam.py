"""rocprofv3 target: FusedAdam 350M steps + LayerNorm fwd/bwd (kernel-level
profiling payload, small and fast)."""

import torch

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


from apex_amd.optimizers import FusedAdam
from apex_amd.normalization import FusedLayerNorm


def main():
    device = "cuda"
    torch.manual_seed(0)
    n_tensors = 192
    numel = 350_000_000 // n_tensors
    params = [torch.empty(numel, device=device).normal_(0, 0.02).requires_grad_(True)
              for _ in range(n_tensors)]
    for p in params:
        p.grad = torch.empty_like(p).normal_(0, 0.01)
    opt = FusedAdam(params, lr=1e-3, weight_decay=0.01)
    for _ in range(3):
        opt.step()
    torch.cuda.synchronize()
    for _ in range(10):
        opt.step()
    torch.cuda.synchronize()
    del params, opt
    torch.cuda.empty_cache()

    # LayerNorm: BERT-ish shape
    ln = FusedLayerNorm(1024).cuda().to(torch.bfloat16)
    x = torch.randn(8192, 1024, device=device, dtype=torch.bfloat16, requires_grad=True)
    for _ in range(10):
        y = ln(x)
        y.backward(torch.ones_like(y))
    torch.cuda.synchronize()
    print("prof payload done")


if __name__ == "__main__":
    main()
