"""Fused Conv+Bias(+Mask)+ReLU surface.

API parity with the reference ``apex.contrib.conv_bias_relu``
(apex/contrib/conv_bias_relu/conv_bias_relu.py:9-110 — ConvBias,
ConvBiasReLU, ConvBiasMaskReLU, ConvFrozenScaleBiasReLU). The reference
fuses through the cuDNN-frontend runtime-fusion graph API. On MI355X:

* **1x1 convolutions** (the two pointwise convs of every ResNet bottleneck
  and its downsample) ARE GEMMs over NHWC — they run as ONE hipBLASLt
  launch with the bias+ReLU applied in-register (RELU_BIAS epilogue;
  frozen-BN scale is folded into the weight since conv is linear in W).
  Backward is three fused GEMMs (dgrad; wgrad with BGRADB bias-grad
  epilogue) on the ReLU-masked grad.
* **spatial convolutions** run through MIOpen (torch conv2d) with the
  bias/scale/mask/ReLU epilogue composed around them — MIOpen exposes no
  cuDNN-style runtime fusion, and the conv dominates there.
"""

import torch

from ..._ext import get_ext


def _is_1x1(weight, padding):
    p = (padding, padding) if isinstance(padding, int) else tuple(padding)
    return tuple(weight.shape[2:]) == (1, 1) and p == (0, 0)


def _gemm_view(x, stride):
    """[N,C,H,W] (+optional stride subsample) -> [N*H'*W', C] rows.

    Zero-copy for channels_last stride-1 tensors; otherwise one packing pass
    (the same reshape a conv algorithm would do internally)."""
    s = (stride, stride) if isinstance(stride, int) else tuple(stride)
    if s != (1, 1):
        x = x[:, :, ::s[0], ::s[1]]
    n, c, h, w = x.shape
    return x.permute(0, 2, 3, 1).reshape(n * h * w, c), (n, h, w, c)


def _from_gemm_view(out2d, n, h, w, c_out):
    """[N*H*W, Cout] -> [N,Cout,H,W]-shaped channels_last tensor (zero copy).

    The permute result is returned from inside autograd.Functions only as a
    FINAL tensor (never modified in place afterwards by this module)."""
    return out2d.view(n, h, w, c_out).permute(0, 3, 1, 2)


class _Conv1x1BiasReLU(torch.autograd.Function):
    """relu(x @ w^T + b) for 1x1 convs — one fused hipBLASLt launch fwd
    (RELU_BIAS epilogue), three fused GEMMs bwd (ReLU mask -> dgrad /
    wgrad+BGRADB). Returns the GEMM rows [N*H'*W', Cout]; the public
    wrappers view/permute OUTSIDE the Function (a custom Function must not
    return views of its own output)."""

    @staticmethod
    def forward(ctx, x, weight, bias, stride, with_relu):
        fd = get_ext("fused_dense")
        w2d = weight.reshape(weight.shape[0], weight.shape[1])
        x2d, _ = _gemm_view(x, stride)
        x2d = x2d.contiguous()
        if with_relu:
            out2d = fd.linear_bias_relu_forward(x2d, w2d, bias.reshape(-1))
        else:
            out2d = fd.linear_bias_forward(x2d, w2d, bias.reshape(-1))
        ctx.save_for_backward(x2d, w2d, out2d)
        ctx.meta = (x.shape, stride, with_relu)
        return out2d

    @staticmethod
    def backward(ctx, dy2d):
        fd = get_ext("fused_dense")
        x2d, w2d, out2d = ctx.saved_tensors
        (n, ci, H, W), stride, with_relu = ctx.meta
        co = w2d.shape[0]
        if with_relu:
            dy2d = dy2d * (out2d > 0).to(dy2d.dtype)
        dx2d, dw2d, db = fd.linear_bias_backward(x2d, w2d, dy2d.contiguous())
        s = (stride, stride) if isinstance(stride, int) else tuple(stride)
        h, w = -(-H // s[0]), -(-W // s[1])
        if s != (1, 1):
            gx = torch.empty(n, ci, H, W, dtype=dy2d.dtype, device=dy2d.device,
                             memory_format=torch.channels_last).zero_()
            gx[:, :, ::s[0], ::s[1]] = _from_gemm_view(dx2d, n, h, w, ci)
        else:
            gx = _from_gemm_view(dx2d, n, h, w, ci).contiguous(
                memory_format=torch.channels_last)
        return gx, dw2d.view(co, ci, 1, 1), db.reshape(1, co, 1, 1), None, None


def _conv1x1_out(x, weight, bias, stride, with_relu):
    """Fused 1x1 path wrapper: run the GEMM Function, then shape the rows
    back to [N,Cout,H',W'] (channels_last) with plain autograd view ops."""
    s = (stride, stride) if isinstance(stride, int) else tuple(stride)
    n, _, H, W = x.shape
    h, w = -(-H // s[0]), -(-W // s[1])
    out2d = _Conv1x1BiasReLU.apply(x, weight, bias, stride, with_relu)
    return out2d.view(n, h, w, weight.shape[0]).permute(0, 3, 1, 2)


def _use_fused_1x1(x, weight, padding):
    return (x.is_cuda and _is_1x1(weight, padding)
            and x.dtype in (torch.float32, torch.float16, torch.bfloat16))


class ConvBiasReLU_(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, padding, stride):
        out = torch.nn.functional.conv2d(x, weight, bias.reshape(-1), stride, padding)
        out = out.relu_()
        ctx.save_for_backward(x, weight, out)
        ctx.padding = padding
        ctx.stride = stride
        return out

    @staticmethod
    def backward(ctx, grad_output):
        x, weight, out = ctx.saved_tensors
        dy = grad_output * (out > 0).to(grad_output.dtype)
        gi, gw, gb = _conv_bwd(x, weight, dy, ctx.padding, ctx.stride, bias=True)
        return gi, gw, gb, None, None


class ConvBias_(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, padding, stride):
        out = torch.nn.functional.conv2d(x, weight, bias.reshape(-1), stride, padding)
        ctx.save_for_backward(x, weight)
        ctx.padding = padding
        ctx.stride = stride
        return out

    @staticmethod
    def backward(ctx, grad_output):
        x, weight = ctx.saved_tensors
        gi, gw, gb = _conv_bwd(x, weight, grad_output, ctx.padding, ctx.stride, bias=True)
        return gi, gw, gb, None, None


class ConvBiasMaskReLU_(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, mask, padding, stride):
        out = torch.nn.functional.conv2d(x, weight, bias.reshape(-1), stride, padding)
        out = out.mul_(mask).relu_()
        ctx.save_for_backward(x, weight, out, mask)
        ctx.padding = padding
        ctx.stride = stride
        return out

    @staticmethod
    def backward(ctx, grad_output):
        x, weight, out, mask = ctx.saved_tensors
        dy = grad_output * (out > 0).to(grad_output.dtype) * mask
        gi, gw, gb = _conv_bwd(x, weight, dy, ctx.padding, ctx.stride, bias=True)
        return gi, gw, gb, None, None, None


class ConvFrozenScaleBiasReLU_(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, scale, bias, padding, stride):
        conv = torch.nn.functional.conv2d(x, weight, None, stride, padding)
        out = (conv * scale + bias).relu_()
        ctx.save_for_backward(x, weight, scale, out)
        ctx.padding = padding
        ctx.stride = stride
        return out

    @staticmethod
    def backward(ctx, grad_output):
        x, weight, scale, out = ctx.saved_tensors
        dy = grad_output * (out > 0).to(grad_output.dtype) * scale
        gi, gw, _ = _conv_bwd(x, weight, dy, ctx.padding, ctx.stride, bias=False)
        return gi, gw, None, None, None, None


def _conv_bwd(x, weight, dy, padding, stride, bias):
    dy = dy.contiguous(memory_format=torch.channels_last) if dy.is_contiguous(
        memory_format=torch.channels_last) else dy.contiguous()
    gi, gw = torch.ops.aten.convolution_backward(
        dy, x, weight, None,
        [stride, stride] if isinstance(stride, int) else list(stride),
        [padding, padding] if isinstance(padding, int) else list(padding),
        [1, 1], False, [0, 0], 1, [True, True, False],
    )[:2]
    gb = dy.sum(dim=(0, 2, 3)).reshape(1, -1, 1, 1) if bias else None
    return gi, gw, gb


def ConvBias(x, weight, bias, padding=0, stride=1):
    if _use_fused_1x1(x, weight, padding):
        return _conv1x1_out(x, weight, bias, stride, with_relu=False)
    return ConvBias_.apply(x, weight, bias, padding, stride)


def ConvBiasReLU(x, weight, bias, padding=0, stride=1):
    if _use_fused_1x1(x, weight, padding):
        return _conv1x1_out(x, weight, bias, stride, with_relu=True)
    return ConvBiasReLU_.apply(x, weight, bias, padding, stride)


def ConvBiasMaskReLU(x, weight, bias, mask, padding=0, stride=1):
    return ConvBiasMaskReLU_.apply(x, weight, bias, mask, padding, stride)


def ConvFrozenScaleBiasReLU(x, weight, scale, bias, padding=0, stride=1):
    if _use_fused_1x1(x, weight, padding):
        # conv is linear in W: relu(conv(x, W)*scale + bias) ==
        # relu(conv(x, W*scale) + bias). scale/bias are frozen (no grads);
        # the tiny weight-fold is differentiable w.r.t. W automatically.
        w_eff = weight * scale.reshape(-1, 1, 1, 1)
        return _conv1x1_out(x, w_eff, bias, stride, with_relu=True)
    return ConvFrozenScaleBiasReLU_.apply(x, weight, scale, bias, padding, stride)
