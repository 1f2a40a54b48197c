"""Fused indexed elementwise multiply: out[i,:] = in1[idx[i],:] * in2[i,:].

API parity with the reference ``apex.contrib.index_mul_2d``
(apex/contrib/index_mul_2d/index_mul_2d.py): index on dim 0 of a 2D tensor,
fused backward (scatter-add for the indexed operand). Double backward falls
back to composable torch ops.
"""

import torch

from ..._ext import get_ext


class IndexMul2d_(torch.autograd.Function):
    @staticmethod
    def forward(ctx, in1, in2, idx1):
        assert in2.size(0) == idx1.size(0)
        if in1.dim() != 2 or in2.dim() != 2:
            raise RuntimeError("in1 and in2 must be 2-dimension tensor.")
        if idx1.dim() != 1:
            raise RuntimeError("idx1 must be 1-dimension tensor.")
        in1 = in1.contiguous()
        in2 = in2.contiguous()
        idx1 = idx1.contiguous()
        out = torch.empty_like(in2)
        ext = get_ext("index_mul_2d")
        ext.forward(out, in1, in2, idx1)
        ctx.save_for_backward(in1, in2, idx1)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        in1, in2, idx1 = ctx.saved_tensors
        if torch.is_grad_enabled():  # create_graph: differentiable composition
            grad_in1 = torch.zeros_like(in1).index_add(0, idx1, grad_out * in2)
            grad_in2 = in1.index_select(0, idx1) * grad_out
            return grad_in1, grad_in2, None
        ext = get_ext("index_mul_2d")
        grad_in1, grad_in2 = ext.backward(in1, in2, idx1, grad_out.contiguous())
        return grad_in1, grad_in2, None


def index_mul_2d(in1, in2, idx1):
    if not in2.is_cuda:
        out = in1.index_select(0, idx1) * in2
        return out
    return IndexMul2d_.apply(in1, in2, idx1)
