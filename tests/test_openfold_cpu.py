"""CPU tests for contrib.openfold and torchsched surfaces."""

import math

import torch
import pytest


def test_fused_adam_swa_cpu():
    from apex_amd.contrib.openfold import FusedAdamSWA

    torch.manual_seed(0)
    ps = [torch.randn(16, 16, requires_grad=True), torch.randn(8, requires_grad=True)]
    opt = FusedAdamSWA(ps, swa_decay_rate=0.9, lr=1e-2)
    swa_before = [s.clone() for s in opt.swa_params]
    for p in ps:
        p.grad = torch.randn_like(p)
    opt.step()
    for s, s0, p in zip(opt.swa_params, swa_before, ps):
        expected = 0.9 * s0 + 0.1 * p.detach()
        torch.testing.assert_close(s, expected, rtol=1e-5, atol=1e-6)


def test_openfold_mha_cpu():
    from apex_amd.contrib.openfold import AttnTri

    torch.manual_seed(1)
    b, h, s, d = 2, 4, 16, 32
    q = torch.randn(b, h, s, d)
    k = torch.randn(b, h, s, d)
    v = torch.randn(b, h, s, d)
    out = AttnTri(q, k, v, mask=None)
    ref = torch.softmax(q @ k.transpose(-2, -1) / math.sqrt(d), dim=-1) @ v
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)


def test_openfold_layer_norm_cpu():
    from apex_amd.contrib.openfold import LayerNormSmallShapeOptImpl

    # CPU path: module-level fallback exercises the same entry points
    from apex_amd.normalization import FusedLayerNorm

    ln = FusedLayerNorm(32)
    x = torch.randn(4, 32)
    ref = torch.nn.functional.layer_norm(x, (32,), ln.weight, ln.bias, ln.eps)
    torch.testing.assert_close(ln(x), ref)


def test_torchsched_backend_registered():
    from apex_amd.contrib import torchsched

    backend = torchsched.get_backend()
    assert callable(backend)
    assert torchsched.set_default_backend("torchsched") == "torchsched"


def test_torchsched_compiles_a_function():
    import torch._dynamo as dynamo

    dynamo.reset()

    def f(a, b):
        return torch.relu(a @ b) + 1

    compiled = torch.compile(f, backend="torchsched")
    a, b = torch.randn(4, 8), torch.randn(8, 4)
    torch.testing.assert_close(compiled(a, b), f(a, b))


def test_fused_adam_swa_from_optim():
    from apex_amd.contrib.openfold import FusedAdamSWA

    torch.manual_seed(0)
    fp32 = [torch.randn(8, 4, requires_grad=True), torch.randn(6, requires_grad=True)]
    adam = torch.optim.Adam(fp32, lr=1e-2, weight_decay=0.01)
    for _ in range(3):  # build up real Adam state
        for p in fp32:
            p.grad = torch.randn_like(p)
        adam.step()
    bf16 = [p.detach().to(torch.bfloat16) for p in fp32]
    swa = [p.detach().clone() for p in fp32]
    opt = FusedAdamSWA.from_optim(adam, fp32, bf16, swa, swa_decay_rate=0.9)
    assert opt.param_groups[0]["step"] == 3
    for src, dst in zip(fp32, opt.param_groups[0]["params"]):
        torch.testing.assert_close(opt.state[dst]["exp_avg"],
                                   adam.state[src]["exp_avg"])
    for p in fp32:
        p.grad = torch.randn_like(p)
    before = [s.clone() for s in swa]
    opt.step()
    for s, b, p in zip(swa, before, fp32):
        torch.testing.assert_close(s, 0.9 * b + 0.1 * p.detach(), rtol=1e-5, atol=1e-6)
    for c, p in zip(bf16, fp32):  # compute copy refreshed
        torch.testing.assert_close(c, p.detach().to(torch.bfloat16))


def test_openfold_attn_jit_variants_and_predicate():
    # reference openfold_triton surface: AttnBiasJIT / AttnNoBiasJIT /
    # CanSchTriMHA / enable-disable, and the alias module names
    from apex_amd.contrib.openfold import (
        AttnBiasJIT, AttnNoBiasJIT, AttnTri, CanSchTriMHA)
    from apex_amd.contrib.openfold import mha as mha_mod
    import apex_amd.contrib.openfold_triton as alias
    import apex_amd.contrib.cudnn_gbn as gbn_alias

    assert alias.AttnTri is AttnTri
    assert gbn_alias.GroupBatchNorm2d is not None

    torch.manual_seed(2)
    b, h, s, d = 2, 3, 8, 16
    q, k, v = (torch.randn(b, h, s, d) for _ in range(3))
    mask = (torch.rand(b, 1, 1, s) > 0.2).float()  # 1.0 keep / 0.0 drop
    bias = torch.randn(1, h, s, s)

    out_b = AttnBiasJIT(q, k, v, mask, bias, 1e9)
    out_nb = AttnNoBiasJIT(q, k, v, mask, 1e9)
    # eager reference: standard softmax attention with additive mask fold
    scale = 1.0 / math.sqrt(d)
    logits = q @ k.transpose(-2, -1) * scale + (mask - 1.0) * 1e9
    torch.testing.assert_close(
        out_nb, torch.softmax(logits, -1) @ v, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(
        out_b, torch.softmax(logits + bias, -1) @ v, rtol=1e-4, atol=1e-5)

    assert CanSchTriMHA([1, 128, 8, 256, 32])
    assert not CanSchTriMHA([1, 128, 8, 256, 32], inf=1e4)
    mha_mod.disable()
    try:
        assert not CanSchTriMHA([1, 128, 8, 256, 32])
    finally:
        mha_mod.enable()
