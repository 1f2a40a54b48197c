"""NHWC GroupNorm with fused SiLU (diffusion-UNet norm).

API parity with the reference ``apex.contrib.group_norm.GroupNorm``
(apex/contrib/group_norm/group_norm.py:211+): ``act`` in {"", "silu",
"swish"}; NHWC (channels_last) memory format. Unlike the reference — whose
one-pass kernel only supports a fixed list of channel counts and silently
falls back to PyTorch otherwise — the gfx950 kernels are shape-generic, so
every (C, G) combination takes the fused path.
"""

import torch

from ..._ext import get_ext


class _GroupNormNHWCFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, num_groups, eps, act):
        ext = get_ext("group_norm")
        silu = act in ("silu", "swish")
        y, mean, rstd = ext.fwd(x, weight, bias, num_groups, eps, silu)
        ctx.save_for_backward(x, weight, bias, mean, rstd)
        ctx.num_groups = num_groups
        ctx.silu = silu
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext("group_norm")
        x, weight, bias, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext.bwd(dy, x, mean, rstd, weight, bias, ctx.num_groups, ctx.silu)
        if weight is not None:
            dw = dw.to(weight.dtype)
            db = db.to(weight.dtype)
        else:
            dw = db = None
        return dx, dw, db, None, None, None


def cuda_group_norm_nhwc_forward(x, weight, bias, num_groups, eps=1e-5, act=""):
    return _GroupNormNHWCFunction.apply(x, weight, bias, num_groups, eps, act)


class GroupNorm(torch.nn.GroupNorm):
    """Drop-in torch.nn.GroupNorm with NHWC fused kernels and optional fused
    SiLU. Input layout: channels_last 4D [N, C, H, W] (torch semantics) or an
    explicit NHWC tensor via ``forward_nhwc``."""

    def __init__(self, num_groups, num_channels, eps=1e-5, affine=True, device=None, dtype=None,
                 act=""):
        super().__init__(num_groups, num_channels, eps=eps, affine=affine, device=device,
                         dtype=dtype)
        act = act.lower()
        if act not in ("", "silu", "swish"):
            raise ValueError(f"GroupNorm: unsupported activation {act}")
        self.act = act

    def forward(self, input):
        if input.is_cuda and input.dim() == 4:
            # run in NHWC: channels_last tensors pass through without copies
            x_nhwc = input.permute(0, 2, 3, 1)
            y = cuda_group_norm_nhwc_forward(
                x_nhwc, self.weight, self.bias, self.num_groups, self.eps, self.act
            )
            return y.permute(0, 3, 1, 2)
        out = torch.nn.functional.group_norm(input, self.num_groups, self.weight, self.bias, self.eps)
        if self.act:
            out = torch.nn.functional.silu(out)
        return out

    def forward_nhwc(self, input_nhwc):
        if input_nhwc.is_cuda:
            return cuda_group_norm_nhwc_forward(
                input_nhwc, self.weight, self.bias, self.num_groups, self.eps, self.act
            )
        x = input_nhwc.permute(0, 3, 1, 2)
        out = torch.nn.functional.group_norm(x, self.num_groups, self.weight, self.bias, self.eps)
        if self.act:
            out = torch.nn.functional.silu(out)
        return out.permute(0, 2, 3, 1)
