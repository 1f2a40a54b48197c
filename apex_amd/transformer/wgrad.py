"""Fused weight-gradient GEMM with in-place fp32/fp16 accumulation.

API parity with the reference ``fused_weight_gradient_mlp_cuda``
(csrc/megatron/fused_weight_gradient_dense.cpp:11-13): computes
``main_grad += grad_output^T @ input`` as a single beta=1 GEMM so the weight
gradient lands directly in the persistent main-grad buffer (no intermediate
allocation, no add pass). Device path: rocBLAS/hipBLASLt GEMM with fp32
compute and beta=1 accumulate (csrc/fused_dense.hip).
"""

import torch

from .._ext import get_ext


def _check(input, grad_output, main_grad, acc_dtype):
    assert main_grad.dtype == acc_dtype
    in2d = input.reshape(-1, input.shape[-1])
    go2d = grad_output.reshape(-1, grad_output.shape[-1])
    assert in2d.shape[0] == go2d.shape[0]
    return in2d, go2d


def wgrad_gemm_accum_fp32(input, grad_output, main_grad):
    """main_grad(fp32) += grad_output^T @ input."""
    in2d, go2d = _check(input, grad_output, main_grad, torch.float32)
    if input.is_cuda:
        ext = get_ext("fused_dense")
        ext.wgrad_gemm_accum_fp32(in2d, go2d, main_grad)
    else:
        main_grad.add_(go2d.t().float() @ in2d.float())
    return main_grad


def wgrad_gemm_accum_fp16(input, grad_output, main_grad):
    """main_grad(fp16/bf16) += grad_output^T @ input (16-bit main grads)."""
    assert main_grad.dtype in (torch.float16, torch.bfloat16)
    in2d = input.reshape(-1, input.shape[-1])
    go2d = grad_output.reshape(-1, grad_output.shape[-1])
    if input.is_cuda:
        ext = get_ext("fused_dense")
        ext.wgrad_gemm_accum_fp16(in2d, go2d, main_grad)
    else:
        main_grad.add_((go2d.t().float() @ in2d.float()).to(main_grad.dtype))
    return main_grad
