"""Fused rotary positional embedding (RoPE) for MI355X.

API parity with the reference ``fused_rotary_positional_embedding``
extension (csrc/megatron/fused_rotary_positional_embedding.cpp:176-193):
four layouts — plain sbhd, cached cos/sin, packed-varlen thd (cu_seqlens),
and 2d (image grid). Device kernels in csrc/rope.hip are vectorized
rotate-half elementwise passes; cos/sin are always precomputed tables
(on-device trig would turn this memory-bound op VALU-bound on CDNA4).
CPU path: reference torch math.
"""

import torch

from .._ext import get_ext


def _rotate_half(x):
    x1, x2 = torch.chunk(x, 2, dim=-1)
    return torch.cat((-x2, x1), dim=-1)


def _ref_apply_rope(t, freqs):
    # t: [s, b, h, d], freqs: [s, 1, 1, d_rot]
    rot_dim = freqs.shape[-1]
    t_rot, t_pass = t[..., :rot_dim], t[..., rot_dim:]
    cos_ = torch.cos(freqs).to(t.dtype)
    sin_ = torch.sin(freqs).to(t.dtype)
    t_rot = (t_rot * cos_) + (_rotate_half(t_rot) * sin_)
    return torch.cat((t_rot, t_pass), dim=-1)


class FusedRoPEFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, t, freqs, transpose_output_memory=False):
        if t.is_cuda:
            ext = get_ext("rope")
            output = ext.forward(t, freqs, transpose_output_memory)
        else:
            output = _ref_apply_rope(t, freqs)
        ctx.save_for_backward(freqs)
        ctx.transpose_output_memory = transpose_output_memory
        return output

    @staticmethod
    def backward(ctx, grad_output):
        (freqs,) = ctx.saved_tensors
        if grad_output.is_cuda:
            ext = get_ext("rope")
            grad_input = ext.backward(grad_output.contiguous(), freqs, ctx.transpose_output_memory)
        else:
            # d/dt of rope: cos * g - rotate_half^T(sin * g); rotate_half is
            # orthogonal: inverse rotation = apply with negated sin
            grad_input = _ref_apply_rope(grad_output, -freqs)
        return grad_input, None, None


class FusedRoPECachedFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, t, cos_, sin_):
        if t.is_cuda:
            ext = get_ext("rope")
            output = ext.forward_cached(t, cos_, sin_)
        else:
            rot_dim = cos_.shape[-1]
            t_rot, t_pass = t[..., :rot_dim], t[..., rot_dim:]
            t_rot = (t_rot * cos_.to(t.dtype)) + (_rotate_half(t_rot) * sin_.to(t.dtype))
            output = torch.cat((t_rot, t_pass), dim=-1)
        ctx.save_for_backward(cos_, sin_)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        cos_, sin_ = ctx.saved_tensors
        if grad_output.is_cuda:
            ext = get_ext("rope")
            grad_input = ext.backward_cached(grad_output.contiguous(), cos_, sin_)
        else:
            rot_dim = cos_.shape[-1]
            g_rot, g_pass = grad_output[..., :rot_dim], grad_output[..., rot_dim:]
            g_rot = (g_rot * cos_.to(grad_output.dtype)) + (_rotate_half(g_rot) * (-sin_).to(grad_output.dtype))
            grad_input = torch.cat((g_rot, g_pass), dim=-1)
        return grad_input, None, None


class FusedRoPETHDFunc(torch.autograd.Function):
    """Packed varlen layout: t [total_tokens, h, d] with cu_seqlens."""

    @staticmethod
    def forward(ctx, t, cu_seqlens, freqs):
        if t.is_cuda:
            ext = get_ext("rope")
            output = ext.forward_thd(t, cu_seqlens, freqs)
        else:
            output = torch.empty_like(t)
            for i in range(cu_seqlens.numel() - 1):
                s0, s1 = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
                seg = t[s0:s1].unsqueeze(1)  # [s, 1, h, d]
                output[s0:s1] = _ref_apply_rope(seg, freqs[: s1 - s0]).squeeze(1)
        ctx.save_for_backward(cu_seqlens, freqs)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        cu_seqlens, freqs = ctx.saved_tensors
        if grad_output.is_cuda:
            ext = get_ext("rope")
            grad_input = ext.backward_thd(grad_output.contiguous(), cu_seqlens, freqs)
        else:
            grad_input = torch.empty_like(grad_output)
            for i in range(cu_seqlens.numel() - 1):
                s0, s1 = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
                seg = grad_output[s0:s1].unsqueeze(1)
                grad_input[s0:s1] = _ref_apply_rope(seg, -freqs[: s1 - s0]).squeeze(1)
        return grad_input, None, None


class FusedRoPE2DFunc(torch.autograd.Function):
    """2D (image-grid) rope: t [b, img_h, img_w, h, d]; separate h/w freqs."""

    @staticmethod
    def forward(ctx, t, img_h, img_w, cos_h, sin_h, cos_w, sin_w):
        b = t.shape[0]
        t = t.view(b, img_h, img_w, t.shape[2], t.shape[3]) if t.dim() == 4 else t
        if t.is_cuda:
            ext = get_ext("rope")
            output = ext.forward_2d(t, cos_h, sin_h, cos_w, sin_w)
        else:
            d = t.shape[-1]
            t_h, t_w = t[..., : d // 2], t[..., d // 2:]
            ch = cos_h[:, :img_h].unsqueeze(2).to(t.dtype)  # [1, H, 1, 1, d/2]
            sh = sin_h[:, :img_h].unsqueeze(2).to(t.dtype)
            cw = cos_w[:, :img_w].unsqueeze(1).to(t.dtype)  # [1, 1, W, 1, d/2]
            sw = sin_w[:, :img_w].unsqueeze(1).to(t.dtype)
            t_h = t_h * ch + _rotate_half(t_h) * sh
            t_w = t_w * cw + _rotate_half(t_w) * sw
            output = torch.cat([t_h, t_w], dim=-1)
        ctx.save_for_backward(cos_h, sin_h, cos_w, sin_w)
        ctx.img_h, ctx.img_w = img_h, img_w
        return output.view(b, img_h * img_w, t.shape[3], t.shape[4])

    @staticmethod
    def backward(ctx, grad_output):
        cos_h, sin_h, cos_w, sin_w = ctx.saved_tensors
        img_h, img_w = ctx.img_h, ctx.img_w
        b = grad_output.shape[0]
        g = grad_output.view(b, img_h, img_w, grad_output.shape[2], grad_output.shape[3])
        if g.is_cuda:
            ext = get_ext("rope")
            grad_input = ext.backward_2d(g.contiguous(), cos_h, sin_h, cos_w, sin_w)
        else:
            d = g.shape[-1]
            g_h, g_w = g[..., : d // 2], g[..., d // 2:]
            ch = cos_h[:, :img_h].unsqueeze(2).to(g.dtype)
            sh = sin_h[:, :img_h].unsqueeze(2).to(g.dtype)
            cw = cos_w[:, :img_w].unsqueeze(1).to(g.dtype)
            sw = sin_w[:, :img_w].unsqueeze(1).to(g.dtype)
            g_h = g_h * ch + _rotate_half(g_h) * (-sh)
            g_w = g_w * cw + _rotate_half(g_w) * (-sw)
            grad_input = torch.cat([g_h, g_w], dim=-1)
        return grad_input.view_as(grad_output), None, None, None, None, None, None


def fused_apply_rotary_pos_emb(t, freqs, transpose_output_memory=False):
    return FusedRoPEFunc.apply(t, freqs, transpose_output_memory)


def fused_apply_rotary_pos_emb_cached(t, cos_, sin_):
    return FusedRoPECachedFunc.apply(t, cos_, sin_)


def fused_apply_rotary_pos_emb_thd(t, cu_seqlens, freqs):
    return FusedRoPETHDFunc.apply(t, cu_seqlens, freqs)


def fused_apply_rotary_pos_emb_2d(t, img_h, img_w, cos_h, sin_h, cos_w, sin_w):
    return FusedRoPE2DFunc.apply(t, img_h, img_w, cos_h, sin_h, cos_w, sin_w)
