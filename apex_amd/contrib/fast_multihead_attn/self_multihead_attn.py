"""Fast self multi-head attention (the reference's historical
apex.contrib.fast_multihead_attn, named by BASELINE config #5).

MI355X composition: fused QKV projection (hipBLASLt bias epilogue) → batched
QK^T GEMM → fused scaled (causal/masked) wave64 softmax → PV GEMM → fused
output projection. Matches torch.nn.MultiheadAttention semantics for
self-attention with batch_first=False ([seq, batch, hidden]).
"""

import math

import torch

from ...fused_dense import fused_dense_function
from ...transformer import scaled_masked_softmax, scaled_softmax, scaled_upper_triang_masked_softmax


class SelfMultiheadAttn(torch.nn.Module):
    def __init__(self, embed_dim, num_heads, dropout=0.0, bias=True, include_norm_add=False,
                 impl="fast", separate_qkv_params=False, mask_additive=False):
        super().__init__()
        assert embed_dim % num_heads == 0
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.head_dim = embed_dim // num_heads
        self.dropout = dropout
        self.include_norm_add = include_norm_add
        self.impl = impl
        self.scaling = 1.0 / math.sqrt(self.head_dim)

        self.qkv_weight = torch.nn.Parameter(torch.empty(3 * embed_dim, embed_dim))
        self.qkv_bias = torch.nn.Parameter(torch.zeros(3 * embed_dim)) if bias else None
        self.out_proj_weight = torch.nn.Parameter(torch.empty(embed_dim, embed_dim))
        self.out_proj_bias = torch.nn.Parameter(torch.zeros(embed_dim)) if bias else None
        if include_norm_add:
            from ...normalization import FusedLayerNorm

            self.lyr_nrm = FusedLayerNorm(embed_dim)
        self.reset_parameters()

    def reset_parameters(self):
        torch.nn.init.xavier_uniform_(self.qkv_weight)
        torch.nn.init.xavier_uniform_(self.out_proj_weight)

    def forward(self, query, key=None, value=None, key_padding_mask=None, need_weights=False,
                attn_mask=None, is_training=True):
        """query: [seq, batch, hidden]; attn_mask='causal' or [b,1,sq,sk] bool."""
        x = query
        residual = x
        if self.include_norm_add:
            x = self.lyr_nrm(x)
        s, b, h = x.shape
        qkv = fused_dense_function(x.reshape(s * b, h), self.qkv_weight, self.qkv_bias)
        qkv = qkv.reshape(s, b, 3, self.num_heads, self.head_dim)
        q = qkv[:, :, 0].permute(1, 2, 0, 3)  # [b, nh, s, hd]
        k = qkv[:, :, 1].permute(1, 2, 0, 3)
        v = qkv[:, :, 2].permute(1, 2, 0, 3)

        from ...transformer import flash_attention, flash_attention_supported

        dropout_active = self.dropout if (is_training and self.training) else 0.0
        if (not need_weights and key_padding_mask is None
                and (attn_mask is None or attn_mask == "causal")
                and flash_attention_supported(q, dropout=dropout_active)):
            # MFMA flash path (fused philox attention dropout in-kernel)
            ctx = flash_attention(q, k, v,
                                  causal=(attn_mask == "causal"), scale=self.scaling,
                                  dropout_p=dropout_active)
            ctx = ctx.permute(2, 0, 1, 3).reshape(s * b, h)
            out = fused_dense_function(ctx, self.out_proj_weight, self.out_proj_bias)
            out = out.reshape(s, b, h)
            if self.include_norm_add:
                out = out + residual
            return out, None

        scores = torch.matmul(q, k.transpose(-2, -1))
        if attn_mask == "causal":
            probs = scaled_upper_triang_masked_softmax(
                scores.reshape(b * self.num_heads, s, s).contiguous(), self.scaling
            ).reshape(b, self.num_heads, s, s)
        elif attn_mask is not None:
            probs = scaled_masked_softmax(scores.contiguous(), attn_mask, self.scaling)
        elif key_padding_mask is not None:
            m = key_padding_mask[:, None, None, :].expand(b, 1, s, s).contiguous()
            probs = scaled_masked_softmax(scores.contiguous(), m, self.scaling)
        else:
            probs = scaled_softmax(scores.contiguous(), self.scaling)
        if self.dropout > 0 and is_training and self.training:
            probs = torch.nn.functional.dropout(probs, p=self.dropout)

        ctx = torch.matmul(probs, v)  # [b, nh, s, hd]
        ctx = ctx.permute(2, 0, 1, 3).reshape(s * b, h)
        out = fused_dense_function(ctx, self.out_proj_weight, self.out_proj_bias)
        out = out.reshape(s, b, h)
        if self.include_norm_add:
            out = out + residual
        if need_weights:
            return out, probs
        return out, None
