"""Transformer language models built from apex_amd fused ops.

These are the benchmark models for BASELINE configs #3-#5:
- BERT-base encoder: FusedLayerNorm + scaled_masked_softmax + FusedAdam.
- GPT-2 345M decoder: fused_dense (GEMM+bias+GELU) + FusedRMSNorm option +
  scaled_upper_triang_masked_softmax + FusedLAMB.

They exist to exercise the library the way Megatron-style trainers exercise
the reference kernels; synthetic-data training only.
"""

import math
from dataclasses import dataclass

import torch
import torch.nn as nn

from ..normalization import FusedLayerNorm, FusedRMSNorm
from ..fused_dense import fused_dense_function, fused_dense_gelu_dense_function
from ..transformer import (
    fused_apply_rotary_pos_emb,
    scaled_masked_softmax,
    scaled_softmax,
    scaled_upper_triang_masked_softmax,
)


@dataclass
class TransformerLMConfig:
    vocab_size: int = 50304
    hidden: int = 768
    layers: int = 12
    heads: int = 12
    seq_len: int = 512
    ffn_hidden: int = 0  # 0 → 4*hidden
    causal: bool = False
    norm: str = "layernorm"  # or "rmsnorm"

    def __post_init__(self):
        if self.ffn_hidden == 0:
            self.ffn_hidden = 4 * self.hidden


def bert_base_config(seq_len=512):
    return TransformerLMConfig(vocab_size=30528, hidden=768, layers=12, heads=12,
                               seq_len=seq_len, causal=False, norm="layernorm")


def gpt2_345m_config(seq_len=1024):
    return TransformerLMConfig(vocab_size=50304, hidden=1024, layers=24, heads=16,
                               seq_len=seq_len, causal=True, norm="rmsnorm")


class FusedSelfAttention(nn.Module):
    def __init__(self, cfg: TransformerLMConfig):
        super().__init__()
        self.h = cfg.hidden
        self.nh = cfg.heads
        self.hd = cfg.hidden // cfg.heads
        self.causal = cfg.causal
        self.qkv_w = nn.Parameter(torch.empty(3 * cfg.hidden, cfg.hidden))
        self.qkv_b = nn.Parameter(torch.zeros(3 * cfg.hidden))
        self.proj_w = nn.Parameter(torch.empty(cfg.hidden, cfg.hidden))
        self.proj_b = nn.Parameter(torch.zeros(cfg.hidden))
        nn.init.normal_(self.qkv_w, std=0.02)
        nn.init.normal_(self.proj_w, std=0.02)

    def forward(self, x, mask=None):
        from ..transformer import flash_attention, flash_attention_supported

        # x: [b, s, h]
        b, s, h = x.shape
        qkv = fused_dense_function(x, self.qkv_w, self.qkv_b)  # [b, s, 3h]
        qkv = qkv.view(b, s, 3, self.nh, self.hd).permute(2, 0, 3, 1, 4)  # [3, b, nh, s, hd]
        q, k, v = qkv[0], qkv[1], qkv[2]
        scale = 1.0 / math.sqrt(self.hd)
        if mask is None and flash_attention_supported(q):
            # MFMA flash kernel: no S x S matrix, no separate softmax pass
            # strided BSHD views pass straight through (the kernels read
            # through (b,h,s) strides — no .contiguous() copies)
            ctx = flash_attention(q, k, v, causal=self.causal, scale=scale)
        else:
            scores = torch.matmul(q, k.transpose(-2, -1))  # [b, nh, s, s]
            if self.causal:
                probs = scaled_upper_triang_masked_softmax(scores.view(b * self.nh, s, s), scale)
                probs = probs.view(b, self.nh, s, s)
            elif mask is not None:
                probs = scaled_masked_softmax(scores, mask, scale)
            else:
                probs = scaled_softmax(scores.contiguous(), scale)
            ctx = torch.matmul(probs, v)  # [b, nh, s, hd]
        ctx = ctx.transpose(1, 2).reshape(b, s, h)
        return fused_dense_function(ctx, self.proj_w, self.proj_b)


class FusedMLPBlock(nn.Module):
    def __init__(self, cfg: TransformerLMConfig):
        super().__init__()
        self.w1 = nn.Parameter(torch.empty(cfg.ffn_hidden, cfg.hidden))
        self.b1 = nn.Parameter(torch.zeros(cfg.ffn_hidden))
        self.w2 = nn.Parameter(torch.empty(cfg.hidden, cfg.ffn_hidden))
        self.b2 = nn.Parameter(torch.zeros(cfg.hidden))
        nn.init.normal_(self.w1, std=0.02)
        nn.init.normal_(self.w2, std=0.02)

    def forward(self, x):
        return fused_dense_gelu_dense_function(x, self.w1, self.b1, self.w2, self.b2)


class TransformerLayer(nn.Module):
    def __init__(self, cfg: TransformerLMConfig):
        super().__init__()
        norm_cls = FusedLayerNorm if cfg.norm == "layernorm" else FusedRMSNorm
        self.ln1 = norm_cls(cfg.hidden)
        self.attn = FusedSelfAttention(cfg)
        self.ln2 = norm_cls(cfg.hidden)
        self.mlp = FusedMLPBlock(cfg)

    def forward(self, x, mask=None):
        x = x + self.attn(self.ln1(x), mask)
        x = x + self.mlp(self.ln2(x))
        return x


class _TransformerLM(nn.Module):
    def __init__(self, cfg: TransformerLMConfig):
        super().__init__()
        self.cfg = cfg
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.pos_emb = nn.Embedding(cfg.seq_len, cfg.hidden)
        self.layers = nn.ModuleList([TransformerLayer(cfg) for _ in range(cfg.layers)])
        norm_cls = FusedLayerNorm if cfg.norm == "layernorm" else FusedRMSNorm
        self.final_norm = norm_cls(cfg.hidden)
        nn.init.normal_(self.tok_emb.weight, std=0.02)
        nn.init.normal_(self.pos_emb.weight, std=0.02)

    def forward(self, tokens, mask=None):
        from ..normalization import fused_add_norm

        b, s = tokens.shape
        pos = torch.arange(s, device=tokens.device).unsqueeze(0)
        x = self.tok_emb(tokens) + self.pos_emb(pos)
        # pre-LN residual stream with every add fused into the next norm:
        # each sublayer's output is carried as `delta` and folded into the
        # following norm's single-read kernel (z = x + delta written once)
        delta = None
        for layer in self.layers:
            if delta is None:
                n1 = layer.ln1(x)
            else:
                n1, x = fused_add_norm(x, delta, layer.ln1)
            delta = layer.attn(n1, mask)
            n2, x = fused_add_norm(x, delta, layer.ln2)
            delta = layer.mlp(n2)
        x, _ = fused_add_norm(x, delta, self.final_norm)
        # weight-tied LM head
        return torch.matmul(x, self.tok_emb.weight.t())


class BertModel(_TransformerLM):
    def __init__(self, cfg=None):
        super().__init__(cfg or bert_base_config())


class GPTModel(_TransformerLM):
    def __init__(self, cfg=None):
        super().__init__(cfg or gpt2_345m_config())


def transformer_large_config(seq_len=512):
    return TransformerLMConfig(vocab_size=32768, hidden=1024, layers=24, heads=16,
                               seq_len=seq_len, causal=True, norm="layernorm")


class TransformerLargeModel(nn.Module):
    """Transformer-large LM built on contrib.fast_multihead_attn (BASELINE
    config #5: fast-MHA equivalent + contrib.xentropy)."""

    def __init__(self, cfg=None):
        super().__init__()
        from ..contrib.fast_multihead_attn import SelfMultiheadAttn

        cfg = cfg or transformer_large_config()
        self.cfg = cfg
        norm_cls = FusedLayerNorm
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.pos_emb = nn.Embedding(cfg.seq_len, cfg.hidden)
        self.attns = nn.ModuleList([
            SelfMultiheadAttn(cfg.hidden, cfg.heads, dropout=0.0) for _ in range(cfg.layers)
        ])
        self.ln1 = nn.ModuleList([norm_cls(cfg.hidden) for _ in range(cfg.layers)])
        self.ln2 = nn.ModuleList([norm_cls(cfg.hidden) for _ in range(cfg.layers)])
        self.mlps = nn.ModuleList([FusedMLPBlock(cfg) for _ in range(cfg.layers)])
        self.final_norm = norm_cls(cfg.hidden)
        nn.init.normal_(self.tok_emb.weight, std=0.02)
        nn.init.normal_(self.pos_emb.weight, std=0.02)

    def forward(self, tokens):
        from ..normalization import fused_add_norm

        b, s = tokens.shape
        pos = torch.arange(s, device=tokens.device).unsqueeze(0)
        x = (self.tok_emb(tokens) + self.pos_emb(pos)).transpose(0, 1)  # [s, b, h]
        delta = None  # residual adds fused into the next norm's kernel
        for attn, l1, l2, mlp in zip(self.attns, self.ln1, self.ln2, self.mlps):
            if delta is None:
                n1 = l1(x)
            else:
                n1, x = fused_add_norm(x, delta, l1)
            delta, _ = attn(n1, attn_mask="causal")
            n2, x = fused_add_norm(x, delta, l2)
            delta = mlp(n2)
        x, _ = fused_add_norm(x, delta, self.final_norm)
        return torch.matmul(x.transpose(0, 1), self.tok_emb.weight.t())


# ---------------- LLaMA-style decoder (RMSNorm + RoPE + SwiGLU) ----------------

def llama_small_config(seq_len=512):
    """~400M-class LLaMA-shape config for the bench harness."""
    return TransformerLMConfig(vocab_size=32000, hidden=1024, layers=16, heads=16,
                               seq_len=seq_len, ffn_hidden=2816, causal=True,
                               norm="rmsnorm")


class LlamaAttention(nn.Module):
    """Causal self-attention with fused RoPE on q/k (no biases).

    ``use_flash=True`` (default; hardware-validated round 2) routes the
    attention core through ``transformer.flash_attention`` (no S x S matrix)
    instead of bmm + causal wave64 softmax whenever the kernel supports the
    shape/dtype; unsupported cases fall back automatically."""

    use_flash = True

    def __init__(self, cfg: TransformerLMConfig):
        super().__init__()
        self.nh = cfg.heads
        self.hd = cfg.hidden // cfg.heads
        self.qkv_w = nn.Parameter(torch.empty(3 * cfg.hidden, cfg.hidden))
        self.proj_w = nn.Parameter(torch.empty(cfg.hidden, cfg.hidden))
        nn.init.normal_(self.qkv_w, std=0.02)
        nn.init.normal_(self.proj_w, std=0.02)

    def forward(self, x, freqs):
        b, s, h = x.shape
        qkv = torch.nn.functional.linear(x, self.qkv_w)
        qkv = qkv.view(b, s, 3, self.nh, self.hd)
        # fused RoPE expects [s, b, h, d]
        q = fused_apply_rotary_pos_emb(qkv[:, :, 0].transpose(0, 1).contiguous(), freqs)
        k = fused_apply_rotary_pos_emb(qkv[:, :, 1].transpose(0, 1).contiguous(), freqs)
        q = q.permute(1, 2, 0, 3)                       # [b, nh, s, hd]
        k = k.permute(1, 2, 0, 3)
        v = qkv[:, :, 2].permute(0, 2, 1, 3)            # [b, nh, s, hd]
        from ..transformer import flash_attention, flash_attention_supported

        if LlamaAttention.use_flash and flash_attention_supported(q):
            ctx = flash_attention(q, k, v, causal=True, scale=1.0 / math.sqrt(self.hd))
        else:
            scores = torch.matmul(q, k.transpose(-2, -1))
            probs = scaled_upper_triang_masked_softmax(
                scores.reshape(b * self.nh, s, s), 1.0 / math.sqrt(self.hd))
            ctx = torch.matmul(probs.view(b, self.nh, s, s), v)
        ctx = ctx.transpose(1, 2).reshape(b, s, h)
        return torch.nn.functional.linear(ctx, self.proj_w)


class LlamaMLP(nn.Module):
    """SwiGLU: down( silu(gate(x)) * up(x) ), no biases."""

    def __init__(self, cfg: TransformerLMConfig):
        super().__init__()
        self.gate = nn.Parameter(torch.empty(cfg.ffn_hidden, cfg.hidden))
        self.up = nn.Parameter(torch.empty(cfg.ffn_hidden, cfg.hidden))
        self.down = nn.Parameter(torch.empty(cfg.hidden, cfg.ffn_hidden))
        for w in (self.gate, self.up, self.down):
            nn.init.normal_(w, std=0.02)

    def forward(self, x):
        f = torch.nn.functional
        return f.linear(f.silu(f.linear(x, self.gate)) * f.linear(x, self.up), self.down)


class LlamaModel(nn.Module):
    """Decoder-only LM: FusedRMSNorm (adds fused into the next norm), fused
    RoPE, causal wave64 softmax, SwiGLU MLP, tied LM head."""

    def __init__(self, cfg=None):
        super().__init__()
        from ..normalization import fused_add_norm

        self._fused_add_norm = fused_add_norm
        cfg = cfg or llama_small_config()
        self.cfg = cfg
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.attns = nn.ModuleList([LlamaAttention(cfg) for _ in range(cfg.layers)])
        self.mlps = nn.ModuleList([LlamaMLP(cfg) for _ in range(cfg.layers)])
        self.ln1 = nn.ModuleList([FusedRMSNorm(cfg.hidden) for _ in range(cfg.layers)])
        self.ln2 = nn.ModuleList([FusedRMSNorm(cfg.hidden) for _ in range(cfg.layers)])
        self.final_norm = FusedRMSNorm(cfg.hidden)
        nn.init.normal_(self.tok_emb.weight, std=0.02)
        hd = cfg.hidden // cfg.heads
        inv = 1.0 / (10000.0 ** (torch.arange(0, hd, 2).float() / hd))
        ang = torch.outer(torch.arange(cfg.seq_len).float(), inv)  # [s, hd/2]
        self.register_buffer("rope_freqs",
                             torch.cat([ang, ang], dim=-1).view(cfg.seq_len, 1, 1, hd),
                             persistent=False)

    def forward(self, tokens):
        b, s = tokens.shape
        freqs = self.rope_freqs[:s]
        x = self.tok_emb(tokens)
        delta = None
        for attn, mlp, l1, l2 in zip(self.attns, self.mlps, self.ln1, self.ln2):
            if delta is None:
                n1 = l1(x)
            else:
                n1, x = self._fused_add_norm(x, delta, l1)
            delta = attn(n1, freqs)
            n2, x = self._fused_add_norm(x, delta, l2)
            delta = mlp(n2)
        x, _ = self._fused_add_norm(x, delta, self.final_norm)
        return torch.matmul(x, self.tok_emb.weight.t())
