"""CPU-path tests for FusedLayerNorm / FusedRMSNorm module fallbacks."""

import torch
import pytest

from apex_amd.normalization import FusedLayerNorm, FusedRMSNorm, MixedFusedLayerNorm
from apex_amd.normalization.fused_layer_norm import manual_rms_norm


def test_layer_norm_matches_torch():
    torch.manual_seed(0)
    ln = FusedLayerNorm(64)
    ref = torch.nn.LayerNorm(64)
    with torch.no_grad():
        ref.weight.copy_(ln.weight)
        ref.bias.copy_(ln.bias)
    x = torch.randn(8, 32, 64, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y = ln(x)
    y_ref = ref(x2)
    torch.testing.assert_close(y, y_ref)
    y.sum().backward()
    y_ref.sum().backward()
    torch.testing.assert_close(x.grad, x2.grad)
    torch.testing.assert_close(ln.weight.grad, ref.weight.grad)


def test_rms_norm_reference_math():
    torch.manual_seed(0)
    rms = FusedRMSNorm(48)
    x = torch.randn(4, 48)
    y = rms(x)
    expected = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + rms.eps) * rms.weight
    torch.testing.assert_close(y, expected)


def test_rms_norm_no_affine():
    rms = FusedRMSNorm(16, elementwise_affine=False)
    x = torch.randn(3, 16)
    y = rms(x)
    expected = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + rms.eps)
    torch.testing.assert_close(y, expected)


def test_mixed_fused_layer_norm_bf16_input():
    m = MixedFusedLayerNorm(32)
    x = torch.randn(4, 32, dtype=torch.bfloat16)
    y = m(x)
    assert y.dtype == torch.bfloat16


def test_manual_rms_norm_multidim_shape():
    x = torch.randn(2, 3, 4, 5)
    w = torch.ones(4, 5)
    y = manual_rms_norm(x, (4, 5), w, 1e-5)
    assert y.shape == x.shape


def test_fused_add_norm_cpu_fallback():
    from apex_amd.normalization import FusedLayerNorm, FusedRMSNorm, fused_add_norm

    torch.manual_seed(0)
    for norm in (FusedLayerNorm(64), FusedRMSNorm(64)):
        x = torch.randn(8, 64, requires_grad=True)
        d = torch.randn(8, 64, requires_grad=True)
        y, z = fused_add_norm(x, d, norm)
        torch.testing.assert_close(z, x + d)
        torch.testing.assert_close(y, norm(x + d))
        (y.sum() + z.sum()).backward()
        assert x.grad is not None and d.grad is not None


def test_transformer_model_fused_residual_stream_cpu():
    """Restructured pre-LN loop (adds folded into the next norm) must match
    the per-layer eager composition exactly on CPU."""
    from apex_amd.models.transformer import GPTModel, TransformerLMConfig

    cfg = TransformerLMConfig(vocab_size=128, hidden=64, layers=2, heads=4,
                              seq_len=16, causal=True, norm="layernorm")
    torch.manual_seed(0)
    lm = GPTModel(cfg)
    tokens = torch.randint(0, 128, (2, 16))
    out = lm(tokens)
    # eager reference: the original layer-local formulation
    x = lm.tok_emb(tokens) + lm.pos_emb(torch.arange(16).unsqueeze(0))
    for layer in lm.layers:
        x = layer(x)
    x = lm.final_norm(x)
    ref = torch.matmul(x, lm.tok_emb.weight.t())
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)
    out.sum().backward()  # autograd through the fused functional path works
