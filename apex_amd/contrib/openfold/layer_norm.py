"""Small-shape LayerNorm for OpenFold (reference: openfold_triton's
LayerNormSmallShapeOptImpl) — served by the shape-generic wave64 kernels."""

import torch

from ...normalization.fused_layer_norm import FusedLayerNormAffineFunction


class LayerNormSmallShapeOptImpl(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inputs, normalized_shape, weight, bias, eps):
        return FusedLayerNormAffineFunction.forward(ctx, inputs, weight, bias,
                                                    tuple(normalized_shape), eps, False)

    @staticmethod
    def backward(ctx, grad_out):
        dx, dw, db, *_ = FusedLayerNormAffineFunction.backward(ctx, grad_out)
        return dx, None, dw, db, None
