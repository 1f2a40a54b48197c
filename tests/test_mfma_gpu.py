"""Hand-written MFMA GEMM tests: fragment-layout probe (asymmetric inputs,
transpose-detecting) + fused bias/GELU epilogue refcheck."""

import torch
import pytest

pytestmark = pytest.mark.gpu


def test_mfma_tile_layout():
    import apex_amd._mfma as mfma

    torch.manual_seed(0)
    A = torch.randn(16, 32, device="cuda", dtype=torch.bfloat16)
    B = torch.randn(32, 16, device="cuda", dtype=torch.bfloat16)
    D = mfma.mfma_tile_probe(A.view(torch.int16), B.view(torch.int16))
    ref = A.float() @ B.float()
    torch.testing.assert_close(D, ref, rtol=1e-2, atol=1e-2)


@pytest.mark.parametrize("shape", [(256, 128, 64), (512, 384, 768), (1024, 256, 96)])
def test_mfma_gemm_bias_gelu(shape):
    import apex_amd._mfma as mfma

    M, N, K = shape
    torch.manual_seed(1)
    X = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    W = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.1
    b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    out, gi = mfma.gemm_bias_gelu(X, W, b, True)
    z = X.float() @ W.float().t() + b.float()
    ref = torch.nn.functional.gelu(z, approximate="tanh")
    scale = z.abs().max()
    assert ((gi.float() - z).abs().max() / scale) < 2e-2
    assert ((out.float() - ref).abs().max() / scale) < 2e-2


def test_mfma_gemm_bias():
    import apex_amd._mfma as mfma

    torch.manual_seed(2)
    M, N, K = 384, 256, 128
    X = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    W = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.1
    b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    out = mfma.gemm_bias(X, W, b)
    ref = X.float() @ W.float().t() + b.float()
    assert ((out.float() - ref).abs().max() / ref.abs().max()) < 2e-2


@pytest.mark.parametrize("shape", [(256, 128, 128), (512, 384, 768)])
def test_mfma_gemm_v2(shape):
    import apex_amd._mfma as mfma

    M, N, K = shape
    torch.manual_seed(3)
    X = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    W = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.1
    b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    out = mfma.gemm_bias_v2(X, W, b)
    ref = X.float() @ W.float().t() + b.float()
    assert ((out.float() - ref).abs().max() / ref.abs().max()) < 2e-2
    og, gi = mfma.gemm_bias_gelu_v2(X, W, b, True)
    refg = torch.nn.functional.gelu(ref, approximate="tanh")
    scale = ref.abs().max()
    assert ((gi.float() - ref).abs().max() / scale) < 2e-2
    assert ((og.float() - refg).abs().max() / scale) < 2e-2
