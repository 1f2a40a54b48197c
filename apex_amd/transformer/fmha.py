"""Flash-attention forward (EXPERIMENTAL — round-2 perf item).

``flash_attention_forward(q, k, v)`` runs the hand-written MFMA kernel
(csrc/fmha.hip) on GPU: Q K^T, online softmax, and P V in one pass — no
S x S attention matrix is ever materialized. Returns ``(out, lse)`` with
``lse = logsumexp`` rows for a later backward. bf16, head_dim 64/128,
seq_len % 32 == 0, layout [B, H, S, D].

Status: compile-checked and unit-testable against the eager composition
(tests/test_fmha_gpu.py — currently skipped pending on-hardware validation);
not wired into the bundled models yet.
"""

import math

import torch

from .._ext import get_ext


def eager_attention_reference(q, k, v, causal=False, scale=None):
    """fp32 eager composition used as the numerics reference."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if causal:
        S = q.shape[-2]
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), 1)
        s = s.masked_fill(mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, v.float()), lse


def flash_attention_forward(q, k, v, causal=False, scale=None):
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        ext = get_ext("mfma")
        out, lse = ext.fmha_fwd(q, k, v, causal, float(scale))
        return out, lse
    out, lse = eager_attention_reference(q, k, v, causal, scale)
    return out.to(q.dtype), lse
