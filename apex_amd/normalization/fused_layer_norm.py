"""FusedLayerNorm / FusedRMSNorm for MI355X.

API parity with the reference ``apex.normalization.fused_layer_norm``
(apex/normalization/fused_layer_norm.py): affine / non-affine / mixed-dtype
autograd functions, ``memory_efficient`` backward (recompute from output,
save only invvar), functional wrappers, and the four modules.

Device kernels (csrc/fused_norm.hip) do the Welford statistics with wave64
shuffle reductions (one row per wavefront for typical hidden sizes, LDS
inter-wave tree for wide rows) — see the kernel file for the gfx950 launch
geometry. CPU falls back to ``torch.nn.functional.layer_norm`` / manual RMS.
"""

import numbers

import torch
from torch.nn import functional as F
from torch.nn import init
from torch.nn.parameter import Parameter

from .._ext import get_ext


def manual_rms_norm(input, normalized_shape, weight, eps):
    # reference impl: apex/normalization/fused_layer_norm.py:22-35
    dims = tuple(i for i in range(-1, -len(normalized_shape) - 1, -1))
    variance = input.to(torch.float32).pow(2).mean(dims, keepdim=True)
    input = input * torch.rsqrt(variance + eps)
    if weight is None:
        return input
    # handle weight in higher precision than input
    if weight.dtype in [torch.float16, torch.bfloat16]:
        input = input.to(weight.dtype)
    elif input.dtype in [torch.float16, torch.bfloat16]:
        input = input.to(weight.dtype) if weight.dtype == torch.float32 else input
    return weight * input


class FusedLayerNormAffineFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, weight, bias, normalized_shape, eps, memory_efficient=False):
        ext = get_ext("fused_norm")
        ctx.normalized_shape = normalized_shape
        ctx.eps = eps
        ctx.memory_efficient = memory_efficient
        input_ = input.contiguous()
        weight_ = weight.contiguous()
        bias_ = bias.contiguous()
        output, mean, invvar = ext.forward_affine(input_, ctx.normalized_shape, weight_, bias_, ctx.eps)
        if ctx.memory_efficient:
            ctx.save_for_backward(output, weight_, bias_, None, invvar)
        else:
            ctx.save_for_backward(input_, weight_, bias_, mean, invvar)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        ext = get_ext("fused_norm")
        input_or_output, weight_, bias_, mean, invvar = ctx.saved_tensors
        grad_input, grad_weight, grad_bias = ext.backward_affine(
            grad_output.contiguous(), mean, invvar, input_or_output,
            ctx.normalized_shape, weight_, bias_, ctx.eps, ctx.memory_efficient,
        )
        return grad_input, grad_weight, grad_bias, None, None, None


class FusedRMSNormAffineFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, weight, normalized_shape, eps, memory_efficient=False):
        ext = get_ext("fused_norm")
        ctx.normalized_shape = normalized_shape
        ctx.eps = eps
        ctx.memory_efficient = memory_efficient
        input_ = input.contiguous()
        weight_ = weight.contiguous()
        output, invvar = ext.rms_forward_affine(input_, ctx.normalized_shape, weight_, ctx.eps)
        if ctx.memory_efficient:
            ctx.save_for_backward(output, weight_, invvar)
        else:
            ctx.save_for_backward(input_, weight_, invvar)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        ext = get_ext("fused_norm")
        input_or_output, weight_, invvar = ctx.saved_tensors
        grad_input, grad_weight = ext.rms_backward_affine(
            grad_output.contiguous(), invvar, input_or_output,
            ctx.normalized_shape, weight_, ctx.eps, ctx.memory_efficient,
        )
        return grad_input, grad_weight, None, None, None


class FusedLayerNormAffineMixedDtypesFunction(FusedLayerNormAffineFunction):
    @staticmethod
    def forward(ctx, input, weight, bias, normalized_shape, eps, memory_efficient=False):
        ext = get_ext("fused_norm")
        ctx.normalized_shape = normalized_shape
        ctx.eps = eps
        ctx.memory_efficient = memory_efficient
        input_ = input.contiguous()
        weight_ = weight.contiguous()
        bias_ = bias.contiguous()
        output, mean, invvar = ext.forward_affine_mixed_dtypes(
            input_, ctx.normalized_shape, weight_, bias_, ctx.eps
        )
        if ctx.memory_efficient:
            ctx.save_for_backward(output, weight_, bias_, None, invvar)
        else:
            ctx.save_for_backward(input_, weight_, bias_, mean, invvar)
        return output


class FusedRMSNormAffineMixedDtypesFunction(FusedRMSNormAffineFunction):
    @staticmethod
    def forward(ctx, input, weight, normalized_shape, eps, memory_efficient=False):
        ext = get_ext("fused_norm")
        ctx.normalized_shape = normalized_shape
        ctx.eps = eps
        ctx.memory_efficient = memory_efficient
        input_ = input.contiguous()
        weight_ = weight.contiguous()
        output, invvar = ext.rms_forward_affine_mixed_dtypes(
            input_, ctx.normalized_shape, weight_, ctx.eps
        )
        if ctx.memory_efficient:
            ctx.save_for_backward(output, weight_, invvar)
        else:
            ctx.save_for_backward(input_, weight_, invvar)
        return output


class FusedLayerNormFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, normalized_shape, eps, memory_efficient=False):
        ext = get_ext("fused_norm")
        ctx.normalized_shape = normalized_shape
        ctx.eps = eps
        ctx.memory_efficient = memory_efficient
        input_ = input.contiguous()
        output, mean, invvar = ext.forward(input_, ctx.normalized_shape, ctx.eps)
        if ctx.memory_efficient:
            ctx.save_for_backward(output, None, invvar)
        else:
            ctx.save_for_backward(input_, mean, invvar)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        ext = get_ext("fused_norm")
        input_or_output, mean, invvar = ctx.saved_tensors
        grad_input = ext.backward(
            grad_output.contiguous(), mean, invvar, input_or_output,
            ctx.normalized_shape, ctx.eps, ctx.memory_efficient,
        )
        return grad_input, None, None, None


class FusedRMSNormFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, normalized_shape, eps, memory_efficient=False):
        ext = get_ext("fused_norm")
        ctx.normalized_shape = normalized_shape
        ctx.eps = eps
        ctx.memory_efficient = memory_efficient
        input_ = input.contiguous()
        output, invvar = ext.rms_forward(input_, ctx.normalized_shape, ctx.eps)
        if ctx.memory_efficient:
            ctx.save_for_backward(output, invvar)
        else:
            ctx.save_for_backward(input_, invvar)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        ext = get_ext("fused_norm")
        input_or_output, invvar = ctx.saved_tensors
        grad_input = ext.rms_backward(
            grad_output.contiguous(), invvar, input_or_output,
            ctx.normalized_shape, ctx.eps, ctx.memory_efficient,
        )
        return grad_input, None, None, None


# ----- functional wrappers (reference :670-720) -----

def fused_layer_norm_affine(input, weight, bias, normalized_shape, eps=1e-6, memory_efficient=False):
    return FusedLayerNormAffineFunction.apply(input, weight, bias, normalized_shape, eps, memory_efficient)


def fused_layer_norm(input, normalized_shape, eps=1e-6, memory_efficient=False):
    return FusedLayerNormFunction.apply(input, normalized_shape, eps, memory_efficient)


def mixed_dtype_fused_layer_norm_affine(input, weight, bias, normalized_shape, eps=1e-6, memory_efficient=False):
    return FusedLayerNormAffineMixedDtypesFunction.apply(input, weight, bias, normalized_shape, eps, memory_efficient)


def fused_rms_norm_affine(input, weight, normalized_shape, eps=1e-6, memory_efficient=False):
    return FusedRMSNormAffineFunction.apply(input, weight, normalized_shape, eps, memory_efficient)


def fused_rms_norm(input, normalized_shape, eps=1e-6, memory_efficient=False):
    return FusedRMSNormFunction.apply(input, normalized_shape, eps, memory_efficient)


def mixed_dtype_fused_rms_norm_affine(input, weight, normalized_shape, eps=1e-6, memory_efficient=False):
    return FusedRMSNormAffineMixedDtypesFunction.apply(input, weight, normalized_shape, eps, memory_efficient)


# ----- fused residual-add + norm -----
# Pre-LN transformers interleave `z = x + sublayer(…); y = norm(z)` — the
# fused kernel reads x and the sublayer output once and writes both z and y,
# saving a full re-read of z vs. eager add followed by norm. Backward needs
# no new kernel: d(add) is identity, so dX = dResidual = LN-backward(dy, z)
# plus the downstream gradient that arrives on z.

class FusedAddLayerNormAffineFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, residual, weight, bias, normalized_shape, eps):
        ext = get_ext("fused_norm")
        ctx.normalized_shape = normalized_shape
        ctx.eps = eps
        input_ = input.contiguous()
        residual_ = residual.contiguous()
        weight_ = weight.contiguous()
        bias_ = bias.contiguous()
        output, z, mean, invvar = ext.forward_add_affine(
            input_, residual_, ctx.normalized_shape, weight_, bias_, ctx.eps)
        ctx.save_for_backward(z, weight_, bias_, mean, invvar)
        return output, z

    @staticmethod
    def backward(ctx, grad_output, grad_z):
        ext = get_ext("fused_norm")
        z, weight_, bias_, mean, invvar = ctx.saved_tensors
        grad_input, grad_weight, grad_bias = ext.backward_affine(
            grad_output.contiguous(), mean, invvar, z,
            ctx.normalized_shape, weight_, bias_, ctx.eps, False,
        )
        if grad_z is not None:
            grad_input = grad_input + grad_z
        return grad_input, grad_input, grad_weight, grad_bias, None, None


class FusedAddRMSNormAffineFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, residual, weight, normalized_shape, eps):
        ext = get_ext("fused_norm")
        ctx.normalized_shape = normalized_shape
        ctx.eps = eps
        input_ = input.contiguous()
        residual_ = residual.contiguous()
        weight_ = weight.contiguous()
        output, z, invvar = ext.rms_forward_add_affine(
            input_, residual_, ctx.normalized_shape, weight_, ctx.eps)
        ctx.save_for_backward(z, weight_, invvar)
        return output, z

    @staticmethod
    def backward(ctx, grad_output, grad_z):
        ext = get_ext("fused_norm")
        z, weight_, invvar = ctx.saved_tensors
        grad_input, grad_weight = ext.rms_backward_affine(
            grad_output.contiguous(), invvar, z,
            ctx.normalized_shape, weight_, ctx.eps, False,
        )
        if grad_z is not None:
            grad_input = grad_input + grad_z
        return grad_input, grad_input, grad_weight, None, None


def fused_add_layer_norm_affine(input, residual, weight, bias, normalized_shape, eps=1e-6):
    """Returns (layer_norm(input + residual), input + residual)."""
    return FusedAddLayerNormAffineFunction.apply(input, residual, weight, bias,
                                                 normalized_shape, eps)


def fused_add_rms_norm_affine(input, residual, weight, normalized_shape, eps=1e-6):
    """Returns (rms_norm(input + residual), input + residual)."""
    return FusedAddRMSNormAffineFunction.apply(input, residual, weight, normalized_shape, eps)


def fused_add_norm(x, delta, norm):
    """z = x + delta; y = norm(z) through a FusedLayerNorm / FusedRMSNorm
    module — fused on GPU, eager composition elsewhere. Returns (y, z)."""
    if x.is_cuda and norm.elementwise_affine:
        if isinstance(norm, FusedRMSNorm):
            return fused_add_rms_norm_affine(x, delta, norm.weight,
                                             norm.normalized_shape, norm.eps)
        return fused_add_layer_norm_affine(x, delta, norm.weight, norm.bias,
                                           norm.normalized_shape, norm.eps)
    z = x + delta
    return norm(z), z


# ----- modules -----

class FusedLayerNorm(torch.nn.Module):
    """Drop-in replacement for torch.nn.LayerNorm backed by the wave64 HIP
    kernel (reference module: apex/normalization/fused_layer_norm.py:724)."""

    def __init__(self, normalized_shape, eps=1e-5, elementwise_affine=True, memory_efficient=False):
        super().__init__()
        if isinstance(normalized_shape, numbers.Integral):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = torch.Size(normalized_shape)
        self.eps = eps
        self.elementwise_affine = elementwise_affine
        self.memory_efficient = memory_efficient
        if self.elementwise_affine:
            self.weight = Parameter(torch.empty(*normalized_shape))
            self.bias = Parameter(torch.empty(*normalized_shape))
        else:
            self.register_parameter("weight", None)
            self.register_parameter("bias", None)
        self.reset_parameters()

    def reset_parameters(self):
        if self.elementwise_affine:
            init.ones_(self.weight)
            init.zeros_(self.bias)

    def forward(self, input):
        if not input.is_cuda:
            return F.layer_norm(input, self.normalized_shape, self.weight, self.bias, self.eps)
        if self.elementwise_affine:
            return fused_layer_norm_affine(
                input, self.weight, self.bias, self.normalized_shape, self.eps, self.memory_efficient
            )
        return fused_layer_norm(input, self.normalized_shape, self.eps, self.memory_efficient)

    def extra_repr(self):
        return "{normalized_shape}, eps={eps}, elementwise_affine={elementwise_affine}".format(**self.__dict__)


class FusedRMSNorm(torch.nn.Module):
    """Root-mean-square norm (reference module: fused_layer_norm.py:841)."""

    def __init__(self, normalized_shape, eps=1e-5, elementwise_affine=True, memory_efficient=False):
        super().__init__()
        if isinstance(normalized_shape, numbers.Integral):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = torch.Size(normalized_shape)
        self.eps = eps
        self.elementwise_affine = elementwise_affine
        self.memory_efficient = memory_efficient
        if self.elementwise_affine:
            self.weight = Parameter(torch.empty(*normalized_shape))
        else:
            self.register_parameter("weight", None)
        self.reset_parameters()

    def reset_parameters(self):
        if self.elementwise_affine:
            init.ones_(self.weight)

    def forward(self, input):
        if not input.is_cuda:
            return manual_rms_norm(input, self.normalized_shape, self.weight, self.eps)
        if self.elementwise_affine:
            return fused_rms_norm_affine(input, self.weight, self.normalized_shape, self.eps, self.memory_efficient)
        return fused_rms_norm(input, self.normalized_shape, self.eps, self.memory_efficient)

    def extra_repr(self):
        return "{normalized_shape}, eps={eps}, elementwise_affine={elementwise_affine}".format(**self.__dict__)


class MixedFusedLayerNorm(FusedLayerNorm):
    """LayerNorm with fp16/bf16 input + fp32 params (reference :959)."""

    def __init__(self, normalized_shape, eps=1e-5, **kwargs):
        if "elementwise_affine" in kwargs:
            if not kwargs.pop("elementwise_affine"):
                raise RuntimeError("MixedFusedLayerNorm does not support elementwise_affine=False")
        super().__init__(normalized_shape=normalized_shape, eps=eps, elementwise_affine=True, **kwargs)

    def forward(self, input):
        if not input.is_cuda:
            return F.layer_norm(
                input.float(), self.normalized_shape, self.weight, self.bias, self.eps
            ).to(input.dtype)
        return mixed_dtype_fused_layer_norm_affine(
            input, self.weight, self.bias, self.normalized_shape, self.eps, self.memory_efficient
        )


class MixedFusedRMSNorm(FusedRMSNorm):
    """RMSNorm with fp16/bf16 input + fp32 params (reference :1000)."""

    def __init__(self, normalized_shape, eps=1e-5, **kwargs):
        if "elementwise_affine" in kwargs:
            if not kwargs.pop("elementwise_affine"):
                raise RuntimeError("MixedFusedRMSNorm does not support elementwise_affine=False")
        super().__init__(normalized_shape=normalized_shape, eps=eps, elementwise_affine=True, **kwargs)

    def forward(self, input):
        if not input.is_cuda:
            return manual_rms_norm(input, self.normalized_shape, self.weight, self.eps)
        return mixed_dtype_fused_rms_norm_affine(
            input, self.weight, self.normalized_shape, self.eps, self.memory_efficient
        )
