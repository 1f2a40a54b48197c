"""Encoder-decoder (cross) multi-head attention — companion of
SelfMultiheadAttn (reference surface: apex.contrib.fast_multihead_attn's
EncdecMultiheadAttn)."""

import math

import torch

from ...fused_dense import fused_dense_function
from ...transformer import scaled_masked_softmax, scaled_softmax


class EncdecMultiheadAttn(torch.nn.Module):
    def __init__(self, embed_dim, num_heads, dropout=0.0, bias=True, include_norm_add=False,
                 impl="fast"):
        super().__init__()
        assert embed_dim % num_heads == 0
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.head_dim = embed_dim // num_heads
        self.dropout = dropout
        self.include_norm_add = include_norm_add
        self.scaling = 1.0 / math.sqrt(self.head_dim)

        self.q_weight = torch.nn.Parameter(torch.empty(embed_dim, embed_dim))
        self.kv_weight = torch.nn.Parameter(torch.empty(2 * embed_dim, embed_dim))
        self.q_bias = torch.nn.Parameter(torch.zeros(embed_dim)) if bias else None
        self.kv_bias = torch.nn.Parameter(torch.zeros(2 * embed_dim)) if bias else None
        self.out_proj_weight = torch.nn.Parameter(torch.empty(embed_dim, embed_dim))
        self.out_proj_bias = torch.nn.Parameter(torch.zeros(embed_dim)) if bias else None
        if include_norm_add:
            from ...normalization import FusedLayerNorm

            self.lyr_nrm = FusedLayerNorm(embed_dim)
        self.reset_parameters()

    def reset_parameters(self):
        torch.nn.init.xavier_uniform_(self.q_weight)
        torch.nn.init.xavier_uniform_(self.kv_weight)
        torch.nn.init.xavier_uniform_(self.out_proj_weight)

    def forward(self, query, key, value=None, key_padding_mask=None, need_weights=False,
                attn_mask=None, is_training=True):
        """query: [sq, b, h]; key (encoder memory): [sk, b, h]."""
        x = query
        residual = x
        if self.include_norm_add:
            x = self.lyr_nrm(x)
        sq, b, h = x.shape
        sk = key.shape[0]
        q = fused_dense_function(x.reshape(sq * b, h), self.q_weight, self.q_bias)
        kv = fused_dense_function(key.reshape(sk * b, h), self.kv_weight, self.kv_bias)
        q = q.reshape(sq, b, self.num_heads, self.head_dim).permute(1, 2, 0, 3)
        kv = kv.reshape(sk, b, 2, self.num_heads, self.head_dim)
        k = kv[:, :, 0].permute(1, 2, 0, 3)
        v = kv[:, :, 1].permute(1, 2, 0, 3)

        from ...transformer import flash_attention, flash_attention_supported

        dropout_active = self.dropout if (is_training and self.training) else 0.0
        if (not need_weights and key_padding_mask is None and attn_mask is None
                and flash_attention_supported(q, dropout=dropout_active, k=k)):
            # MFMA flash cross-attention (Sq != Skv supported): no sq x sk
            # matrix, strided BSHD views pass copy-free, fused dropout
            ctx = flash_attention(q, k, v, causal=False, scale=self.scaling,
                                  dropout_p=dropout_active)
            ctx = ctx.permute(2, 0, 1, 3).reshape(sq * b, h)
            out = fused_dense_function(ctx, self.out_proj_weight, self.out_proj_bias)
            out = out.reshape(sq, b, h)
            if self.include_norm_add:
                out = out + residual
            return out, None

        scores = torch.matmul(q, k.transpose(-2, -1))  # [b, nh, sq, sk]
        if attn_mask is not None:
            probs = scaled_masked_softmax(scores.contiguous(), attn_mask, self.scaling)
        elif key_padding_mask is not None:
            m = key_padding_mask[:, None, None, :].expand(b, 1, sq, sk).contiguous()
            probs = scaled_masked_softmax(scores.contiguous(), m, self.scaling)
        else:
            probs = scaled_softmax(scores.contiguous(), self.scaling)
        if self.dropout > 0 and is_training and self.training:
            probs = torch.nn.functional.dropout(probs, p=self.dropout)
        ctx = torch.matmul(probs, v).permute(2, 0, 1, 3).reshape(sq * b, h)
        out = fused_dense_function(ctx, self.out_proj_weight, self.out_proj_bias)
        out = out.reshape(sq, b, h)
        if self.include_norm_add:
            out = out + residual
        if need_weights:
            return out, probs
        return out, None
