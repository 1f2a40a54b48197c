"""FastLayerNorm — the contrib persistent-LayerNorm surface.

API parity with the reference ``apex.contrib.layer_norm.FastLayerNorm``
(apex/contrib/layer_norm/layer_norm.py:8-45). The reference maintains a
registry of hand-tuned persistent kernels for ~40 fixed hidden sizes
(768..65536); the gfx950 wave64 Welford kernels in csrc/fused_norm.hip are
shape-generic and already row-persistent (one workgroup per row), so
FastLayerNorm shares that implementation and supports ANY hidden size
— a strict superset of the reference's coverage.
"""

import torch

from ...normalization.fused_layer_norm import FusedLayerNormAffineFunction


class FastLayerNormFN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, epsilon, memory_efficient=False):
        return FusedLayerNormAffineFunction.forward(
            ctx, x, gamma, beta, (x.shape[-1],), epsilon, memory_efficient
        )

    @staticmethod
    def backward(ctx, dz):
        dx, dgamma, dbeta, *_ = FusedLayerNormAffineFunction.backward(ctx, dz)
        return dx, dgamma, dbeta, None, None


def _fast_layer_norm(x, weight, bias, epsilon):
    args = _cast_if_autocast_enabled(x, weight, bias, epsilon)
    with torch.amp.autocast("cuda", enabled=False):
        return FastLayerNormFN.apply(*args)


def _cast_if_autocast_enabled(*args):
    if not torch.is_autocast_enabled("cuda"):
        return args
    return torch.amp.autocast_mode._cast(args, "cuda", torch.get_autocast_dtype("cuda"))


class FastLayerNorm(torch.nn.Module):
    def __init__(self, hidden_size, eps=1e-5, memory_efficient=False):
        super().__init__()
        self.epsilon = eps
        self.memory_efficient = memory_efficient
        self.weight = torch.nn.Parameter(torch.empty(hidden_size))
        self.bias = torch.nn.Parameter(torch.empty(hidden_size))
        self.reset_parameters()

    def reset_parameters(self):
        torch.nn.init.ones_(self.weight)
        torch.nn.init.zeros_(self.bias)

    def forward(self, x):
        if not x.is_cuda:
            return torch.nn.functional.layer_norm(
                x, (x.shape[-1],), self.weight, self.bias, self.epsilon
            )
        return FastLayerNormFN.apply(x, self.weight, self.bias, self.epsilon,
                                     self.memory_efficient)
