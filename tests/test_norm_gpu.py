"""GPU numerics for FusedLayerNorm/FusedRMSNorm vs fp32 torch references
(pattern: reference tests/L0/run_fused_layer_norm/test_fused_layer_norm.py)."""

import torch
import pytest

from apex_amd.normalization import (
    FusedLayerNorm,
    FusedRMSNorm,
    MixedFusedLayerNorm,
    MixedFusedRMSNorm,
)

pytestmark = pytest.mark.gpu

TOL = {
    torch.float32: dict(rtol=1e-5, atol=1e-5),
    torch.float16: dict(rtol=1e-3, atol=1e-3),
    torch.bfloat16: dict(rtol=1.6e-2, atol=1.6e-2),
}


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16, torch.bfloat16])
@pytest.mark.parametrize("hidden", [768, 1023, 4096, 12288])
@pytest.mark.parametrize("memory_efficient", [False, True])
def test_layer_norm_affine(dtype, hidden, memory_efficient):
    torch.manual_seed(0)
    rows = 64
    ln = FusedLayerNorm(hidden, memory_efficient=memory_efficient).cuda().to(dtype)
    x = torch.randn(rows, hidden, device="cuda", dtype=dtype, requires_grad=True)
    xr = x.detach().float().clone().requires_grad_(True)
    wr = ln.weight.detach().float().clone().requires_grad_(True)
    br = ln.bias.detach().float().clone().requires_grad_(True)

    y = ln(x)
    y_ref = torch.nn.functional.layer_norm(xr, (hidden,), wr, br, ln.eps)
    torch.testing.assert_close(y.float(), y_ref, **TOL[dtype])

    g = torch.randn_like(y)
    y.backward(g)
    y_ref.backward(g.float())
    torch.testing.assert_close(x.grad.float(), xr.grad, **TOL[dtype])
    # weight grads accumulate over rows — scale tolerance by sqrt(rows)
    wtol = {k: v * 8 for k, v in TOL[dtype].items()}
    torch.testing.assert_close(ln.weight.grad.float(), wr.grad, **wtol)
    torch.testing.assert_close(ln.bias.grad.float(), br.grad, **wtol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_layer_norm_no_affine(dtype):
    torch.manual_seed(1)
    ln = FusedLayerNorm(512, elementwise_affine=False).cuda().to(dtype)
    x = torch.randn(32, 512, device="cuda", dtype=dtype, requires_grad=True)
    xr = x.detach().float().clone().requires_grad_(True)
    y = ln(x)
    y_ref = torch.nn.functional.layer_norm(xr, (512,), None, None, ln.eps)
    torch.testing.assert_close(y.float(), y_ref, **TOL[dtype])
    g = torch.randn_like(y)
    y.backward(g)
    y_ref.backward(g.float())
    torch.testing.assert_close(x.grad.float(), xr.grad, **TOL[dtype])


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16, torch.bfloat16])
@pytest.mark.parametrize("hidden", [768, 1023, 8192])
@pytest.mark.parametrize("memory_efficient", [False, True])
def test_rms_norm_affine(dtype, hidden, memory_efficient):
    torch.manual_seed(2)
    rows = 48
    rms = FusedRMSNorm(hidden, memory_efficient=memory_efficient).cuda().to(dtype)
    x = torch.randn(rows, hidden, device="cuda", dtype=dtype, requires_grad=True)
    xr = x.detach().float().clone().requires_grad_(True)
    wr = rms.weight.detach().float().clone().requires_grad_(True)

    y = rms(x)
    y_ref = xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + rms.eps) * wr
    torch.testing.assert_close(y.float(), y_ref, **TOL[dtype])

    g = torch.randn_like(y)
    y.backward(g)
    y_ref.backward(g.float())
    torch.testing.assert_close(x.grad.float(), xr.grad, **TOL[dtype])
    wtol = {k: v * 8 for k, v in TOL[dtype].items()}
    torch.testing.assert_close(rms.weight.grad.float(), wr.grad, **wtol)


def test_rms_norm_no_affine_gpu():
    torch.manual_seed(3)
    rms = FusedRMSNorm(640, elementwise_affine=False).cuda()
    x = torch.randn(16, 640, device="cuda", requires_grad=True)
    xr = x.detach().clone().requires_grad_(True)
    y = rms(x)
    y_ref = xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + rms.eps)
    torch.testing.assert_close(y, y_ref, rtol=1e-5, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    y_ref.backward(g)
    torch.testing.assert_close(x.grad, xr.grad, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("cls", [MixedFusedLayerNorm, MixedFusedRMSNorm])
def test_mixed_dtype_norm(cls):
    torch.manual_seed(4)
    m = cls(1024).cuda()  # fp32 params
    x = torch.randn(8, 1024, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = m(x)
    assert y.dtype == torch.bfloat16
    assert m.weight.dtype == torch.float32
    y.float().sum().backward()
    assert x.grad is not None
    assert m.weight.grad is not None and m.weight.grad.dtype == torch.float32


def test_layer_norm_3d_input():
    torch.manual_seed(5)
    ln = FusedLayerNorm(256).cuda()
    x = torch.randn(4, 33, 256, device="cuda", requires_grad=True)
    xr = x.detach().clone().requires_grad_(True)
    y = ln(x)
    wr = ln.weight.detach().clone()
    br = ln.bias.detach().clone()
    y_ref = torch.nn.functional.layer_norm(xr, (256,), wr, br, ln.eps)
    torch.testing.assert_close(y, y_ref, rtol=1e-5, atol=1e-5)


def test_layer_norm_non_contiguous_input():
    torch.manual_seed(6)
    ln = FusedLayerNorm(128).cuda()
    base = torch.randn(16, 2, 128, device="cuda")
    x = base[:, 0, :]  # non-contiguous view
    y = ln(x)
    y_ref = torch.nn.functional.layer_norm(x.contiguous(), (128,), ln.weight, ln.bias, ln.eps)
    torch.testing.assert_close(y, y_ref, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("hidden", [768, 1024, 2048, 3000])
def test_fused_add_layer_norm(hidden):
    """Fused z = x + res; y = LN(z) vs eager fp32 composition, fwd + bwd
    including a downstream gradient arriving on z. hidden=3000 exercises the
    non-wave fallback path."""
    from apex_amd.normalization import fused_add_layer_norm_affine

    torch.manual_seed(7)
    n1 = 64
    x = torch.randn(n1, hidden, device="cuda", requires_grad=True)
    r = torch.randn(n1, hidden, device="cuda", requires_grad=True)
    w = torch.randn(hidden, device="cuda", requires_grad=True)
    b = torch.randn(hidden, device="cuda", requires_grad=True)
    y, z = fused_add_layer_norm_affine(x, r, w, b, (hidden,), 1e-5)

    xr = x.detach().clone().requires_grad_(True)
    rr = r.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    z_ref = xr + rr
    y_ref = torch.nn.functional.layer_norm(z_ref, (hidden,), wr, br, 1e-5)

    torch.testing.assert_close(z, z_ref, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(y, y_ref, rtol=1e-4, atol=1e-4)

    dy = torch.randn_like(y)
    dz = torch.randn_like(z)
    (y * dy + z * dz).sum().backward()
    (y_ref * dy + z_ref * dz).sum().backward()
    torch.testing.assert_close(x.grad, xr.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(r.grad, rr.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(w.grad, wr.grad, rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(b.grad, br.grad, rtol=1e-3, atol=1e-3)


def test_fused_add_rms_norm():
    from apex_amd.normalization import fused_add_rms_norm_affine

    torch.manual_seed(8)
    hidden = 1024
    x = torch.randn(32, hidden, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    r = torch.randn(32, hidden, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(hidden, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y, z = fused_add_rms_norm_affine(x, r, w, (hidden,), 1e-5)

    zf = (x.detach().float() + r.detach().float())
    y_ref = zf * torch.rsqrt(zf.pow(2).mean(-1, keepdim=True) + 1e-5) * w.detach().float()
    torch.testing.assert_close(z.float(), zf, rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(y.float(), y_ref, rtol=3e-2, atol=3e-2)
    y.float().sum().backward()
    assert torch.isfinite(x.grad).all() and torch.isfinite(w.grad).all()


def test_fused_add_norm_module_dispatch():
    from apex_amd.normalization import FusedLayerNorm, fused_add_norm

    ln = FusedLayerNorm(512).cuda()
    x = torch.randn(16, 512, device="cuda")
    d = torch.randn(16, 512, device="cuda")
    y, z = fused_add_norm(x, d, ln)
    torch.testing.assert_close(z, x + d, rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(y, ln(x + d), rtol=1e-5, atol=1e-5)
