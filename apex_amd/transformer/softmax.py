"""Megatron-style fused scaled softmax family for MI355X.

API parity with the reference csrc/megatron softmax extensions
(scaled_softmax_cuda, scaled_masked_softmax_cuda,
scaled_upper_triang_masked_softmax_cuda, generic_scaled_masked_softmax_cuda —
see SURVEY.md §2.2.6). The device kernels (csrc/softmax.hip) are wave64
row-softmax: one wavefront handles one or more rows with 16-byte vector
loads and 64-lane shuffle reductions (the reference's 32-wide warp tiling is
exactly what we do NOT port).

Shapes follow the reference contracts:
- scaled_softmax / scaled_masked_softmax: input [b, np, sq, sk]; mask
  [b, 1, sq, sk] broadcast over heads, additive-bool (masked = -10000).
  The reference caps sk at 16384 (scaled_masked_softmax_cuda.cu:43); the
  wave64 online-softmax kernels here have NO row ceiling, so "generic"
  is the same code path rather than a separate extension.
- scaled_upper_triang_masked_softmax: input [attn_batches, sq, sq] (causal).
- generic: arbitrary sk via block-level reduction.
"""

import torch

from .._ext import get_ext


def _ref_scaled_masked_softmax(inputs, mask, scale):
    x = inputs.float() * scale
    if mask is not None:
        x = x.masked_fill(mask.to(torch.bool), -10000.0)
    return torch.softmax(x, dim=-1).to(inputs.dtype)


class ScaledSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inputs, scale):
        if inputs.is_cuda:
            ext = get_ext("softmax")
            results = ext.scaled_softmax_forward(inputs, scale)
        else:
            results = _ref_scaled_masked_softmax(inputs, None, scale)
        ctx.save_for_backward(results)
        ctx.scale = scale
        return results

    @staticmethod
    def backward(ctx, grad_output):
        (softmax_results,) = ctx.saved_tensors
        if softmax_results.is_cuda:
            ext = get_ext("softmax")
            grad = ext.scaled_softmax_backward(grad_output.contiguous(), softmax_results, ctx.scale)
        else:
            y = softmax_results.float()
            g = grad_output.float()
            grad = (y * (g - (g * y).sum(dim=-1, keepdim=True)) * ctx.scale).to(softmax_results.dtype)
        return grad, None


class ScaledMaskedSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inputs, mask, scale):
        if inputs.is_cuda:
            ext = get_ext("softmax")
            results = ext.scaled_masked_softmax_forward(inputs, mask, scale)
        else:
            results = _ref_scaled_masked_softmax(inputs, mask, scale)
        ctx.save_for_backward(results)
        ctx.scale = scale
        return results

    @staticmethod
    def backward(ctx, grad_output):
        (softmax_results,) = ctx.saved_tensors
        if softmax_results.is_cuda:
            ext = get_ext("softmax")
            grad = ext.scaled_masked_softmax_backward(grad_output.contiguous(), softmax_results, ctx.scale)
        else:
            y = softmax_results.float()
            g = grad_output.float()
            grad = (y * (g - (g * y).sum(dim=-1, keepdim=True)) * ctx.scale).to(softmax_results.dtype)
        return grad, None, None


class ScaledUpperTriangMaskedSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inputs, scale):
        if inputs.is_cuda:
            ext = get_ext("softmax")
            results = ext.scaled_upper_triang_masked_softmax_forward(inputs, scale)
        else:
            sq = inputs.shape[-1]
            mask = torch.triu(torch.ones(sq, sq, dtype=torch.bool, device=inputs.device), diagonal=1)
            results = _ref_scaled_masked_softmax(inputs, mask.expand(inputs.shape[0], sq, sq), scale)
        ctx.save_for_backward(results)
        ctx.scale = scale
        return results

    @staticmethod
    def backward(ctx, grad_output):
        (softmax_results,) = ctx.saved_tensors
        if softmax_results.is_cuda:
            ext = get_ext("softmax")
            grad = ext.scaled_upper_triang_masked_softmax_backward(
                grad_output.contiguous(), softmax_results, ctx.scale
            )
        else:
            y = softmax_results.float()
            g = grad_output.float()
            grad = (y * (g - (g * y).sum(dim=-1, keepdim=True)) * ctx.scale).to(softmax_results.dtype)
        return grad, None


class GenericScaledMaskedSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inputs, mask, scale):
        if inputs.is_cuda:
            ext = get_ext("softmax")
            results = ext.generic_scaled_masked_softmax_forward(inputs, mask, scale)
        else:
            results = _ref_scaled_masked_softmax(inputs, mask, scale)
        ctx.save_for_backward(results)
        ctx.scale = scale
        return results

    @staticmethod
    def backward(ctx, grad_output):
        (softmax_results,) = ctx.saved_tensors
        if softmax_results.is_cuda:
            ext = get_ext("softmax")
            grad = ext.generic_scaled_masked_softmax_backward(
                grad_output.contiguous(), softmax_results, ctx.scale
            )
        else:
            y = softmax_results.float()
            g = grad_output.float()
            grad = (y * (g - (g * y).sum(dim=-1, keepdim=True)) * ctx.scale).to(softmax_results.dtype)
        return grad, None, None


def scaled_softmax(inputs, scale):
    return ScaledSoftmax.apply(inputs, scale)


def scaled_masked_softmax(inputs, mask, scale):
    return ScaledMaskedSoftmax.apply(inputs, mask, scale)


def scaled_upper_triang_masked_softmax(inputs, scale):
    return ScaledUpperTriangMaskedSoftmax.apply(inputs, scale)


def generic_scaled_masked_softmax(inputs, mask, scale):
    return GenericScaledMaskedSoftmax.apply(inputs, mask, scale)
