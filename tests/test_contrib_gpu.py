"""GPU numerics for contrib kernels: xentropy, focal_loss, index_mul_2d,
clip_grad (fused path)."""

import torch
import pytest

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16, torch.bfloat16])
@pytest.mark.parametrize("smoothing", [0.0, 0.1])
def test_xentropy_gpu(dtype, smoothing):
    from apex_amd.contrib.xentropy import SoftmaxCrossEntropyLoss

    torch.manual_seed(0)
    N, C = 128, 1000
    logits = torch.randn(N, C, device="cuda", dtype=dtype, requires_grad=True)
    labels = torch.randint(1, C, (N,), device="cuda")
    losses = SoftmaxCrossEntropyLoss.apply(logits, labels, smoothing, 0, True)

    xf = logits.detach().float().requires_grad_(True)
    lse = torch.logsumexp(xf, -1)
    xy = xf.gather(-1, labels.unsqueeze(-1)).squeeze(-1)
    ref = lse - (1 - smoothing) * xy - smoothing * xf.mean(-1)
    tol = dict(rtol=1e-5, atol=1e-5) if dtype == torch.float32 else dict(rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(losses.float(), ref, **tol)

    g = torch.randn(N, device="cuda")
    losses.backward(g.to(losses.dtype))
    ref.backward(g)
    torch.testing.assert_close(logits.grad.float(), xf.grad, **tol)


def test_xentropy_padding_idx_gpu():
    from apex_amd.contrib.xentropy import SoftmaxCrossEntropyLoss

    torch.manual_seed(1)
    N, C = 32, 64
    logits = torch.randn(N, C, device="cuda", requires_grad=True)
    labels = torch.randint(0, C, (N,), device="cuda")
    labels[:5] = 3  # padding idx
    losses = SoftmaxCrossEntropyLoss.apply(logits, labels, 0.0, 3, False)
    assert (losses[labels == 3] == 0).all()
    losses.sum().backward()
    assert (logits.grad[labels == 3] == 0).all()


@pytest.mark.parametrize("smoothing", [0.0, 0.1])
def test_focal_loss_gpu(smoothing):
    from apex_amd.contrib.focal_loss import focal_loss
    from apex_amd.contrib.focal_loss.focal_loss import _ref_focal

    torch.manual_seed(2)
    A, C = 256, 64  # anchors, padded classes
    C_real = 60
    x = torch.randn(A, C, device="cuda", requires_grad=True)
    y = torch.randint(-2, C_real, (A,), device="cuda")
    nps = torch.tensor([float((y >= 0).sum().clamp(min=1))], device="cuda")

    loss = focal_loss(x, y, nps, C_real, 0.25, 2.0, smoothing)
    ref = _ref_focal(x.detach().cpu(), y.cpu(), nps.cpu(), C_real, 0.25, 2.0, smoothing)
    torch.testing.assert_close(loss.cpu(), ref, rtol=1e-4, atol=1e-4)

    # backward vs autograd through the reference expression
    x2 = x.detach().cpu().requires_grad_(True)
    ref2 = _ref_focal(x2, y.cpu(), nps.cpu(), C_real, 0.25, 2.0, smoothing)
    loss.backward()
    ref2.backward()
    torch.testing.assert_close(x.grad.cpu(), x2.grad, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16])
def test_index_mul_2d_gpu(dtype):
    from apex_amd.contrib.index_mul_2d import index_mul_2d

    torch.manual_seed(3)
    src, n, d = 50, 200, 64
    in1 = torch.randn(src, d, device="cuda", dtype=dtype, requires_grad=True)
    in2 = torch.randn(n, d, device="cuda", dtype=dtype, requires_grad=True)
    idx = torch.randint(0, src, (n,), device="cuda")
    out = index_mul_2d(in1, in2, idx)
    ref_in1 = in1.detach().float().requires_grad_(True)
    ref_in2 = in2.detach().float().requires_grad_(True)
    ref = ref_in1.index_select(0, idx) * ref_in2
    tol = dict(rtol=1e-5, atol=1e-5) if dtype == torch.float32 else dict(rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(out.float(), ref, **tol)
    g = torch.randn_like(ref)
    out.backward(g.to(dtype))
    ref.backward(g)
    torch.testing.assert_close(in1.grad.float(), ref_in1.grad, **tol)
    torch.testing.assert_close(in2.grad.float(), ref_in2.grad, **tol)


def test_clip_grad_fused_gpu():
    from apex_amd.contrib.clip_grad import clip_grad_norm_

    torch.manual_seed(4)
    ps1 = [torch.randn(100, device="cuda", requires_grad=True) for _ in range(4)]
    ps2 = [p.detach().clone().requires_grad_(True) for p in ps1]
    for p1, p2 in zip(ps1, ps2):
        g = torch.randn_like(p1) * 5
        p1.grad = g.clone()
        p2.grad = g.clone()
    n1 = clip_grad_norm_(ps1, 1.0)
    n2 = torch.nn.utils.clip_grad_norm_(ps2, 1.0)
    torch.testing.assert_close(n1, n2, rtol=1e-5, atol=1e-6)
    for p1, p2 in zip(ps1, ps2):
        torch.testing.assert_close(p1.grad, p2.grad, rtol=1e-5, atol=1e-6)


def test_index_mul_2d_double_backward_gpu():
    from apex_amd.contrib.index_mul_2d import index_mul_2d

    torch.manual_seed(0)
    in1 = torch.randn(6, 8, device="cuda", requires_grad=True)
    in2 = torch.randn(4, 8, device="cuda", requires_grad=True)
    idx = torch.tensor([0, 2, 2, 5], device="cuda")
    out = index_mul_2d(in1, in2, idx)
    g1, = torch.autograd.grad(out.sum(), in1, create_graph=True)
    gg, = torch.autograd.grad(g1.sum(), in2)
    torch.testing.assert_close(gg, torch.ones_like(in2))
    # first-order grads still match the eager composition
    o_ref = in1.index_select(0, idx) * in2
    torch.testing.assert_close(out, o_ref)


def test_permutation_search_gpu_kernel():
    """GPU stripe-pair scoring: exhaustive search on device must be monotone
    and match the CPU implementation's kept magnitude."""
    from apex_amd.contrib.sparsity.permutation_search import (
        _group_kept_sum, exhaustive_search)

    torch.manual_seed(0)
    w = torch.randn(96, 64, device="cuda")
    base = float(_group_kept_sum(w.abs().float()))
    perm_gpu = exhaustive_search(w)
    kept_gpu = float(_group_kept_sum(w.abs().float()[:, perm_gpu.to(w.device)]))
    assert kept_gpu >= base - 1e-4

    perm_cpu = exhaustive_search(w.cpu())
    kept_cpu = float(_group_kept_sum(w.abs().float().cpu()[:, perm_cpu]))
    # both searches are greedy sweeps; the GPU variant applies batched
    # non-overlapping improvements — allow tiny slack either way
    assert kept_gpu >= kept_cpu * 0.995


def test_permutation_scores_kernel_matches_torch():
    """stripe_pair_scores == the torch topk composition, exactly."""
    from apex_amd._ext import get_ext
    from apex_amd.contrib.sparsity.permutation_search import _stripe_pair_partitions

    ps = get_ext("permutation_search")
    torch.manual_seed(1)
    rows, cols, m = 64, 32, 4
    w = torch.rand(rows, cols, device="cuda")
    parts = _stripe_pair_partitions(m).cuda()
    nstripes = cols // m
    pairs = [(i, j) for i in range(nstripes - 1) for j in range(i + 1, nstripes)]
    perm = torch.arange(cols, device="cuda").view(nstripes, m)
    cols8 = torch.cat([perm[torch.tensor([p[0] for p in pairs], device="cuda")],
                       perm[torch.tensor([p[1] for p in pairs], device="cuda")]], dim=1)
    scores = ps.stripe_pair_scores(w, cols8, parts)
    # torch reference
    for k in (0, len(pairs) // 2, len(pairs) - 1):
        idx = cols8[k]
        wp = w[:, idx][:, parts]              # [rows, P, 8]
        g = wp.reshape(rows, parts.shape[0], 2, m)
        ref = g.topk(2, dim=3).values.sum(dim=(0, 2, 3))
        torch.testing.assert_close(scores[k], ref, rtol=1e-4, atol=1e-3)


def test_clip_grad_mixed_dtypes_gpu():
    """clip_grad_norm_ over a MIXED bf16+fp32 grad list (the conv+BN case)
    must match torch's reference clip (per-dtype multi-tensor grouping)."""
    from apex_amd.contrib.clip_grad import clip_grad_norm_

    torch.manual_seed(0)
    ps = [torch.randn(256, device="cuda", dtype=torch.bfloat16, requires_grad=True),
          torch.randn(128, device="cuda", dtype=torch.float32, requires_grad=True),
          torch.randn(64, device="cuda", dtype=torch.bfloat16, requires_grad=True)]
    for p in ps:
        p.grad = torch.randn_like(p) * 3.0
    ref_grads = [p.grad.clone() for p in ps]

    total = clip_grad_norm_(ps, max_norm=1.0)
    ref_total = torch.norm(torch.stack([g.float().norm() for g in ref_grads]))
    torch.testing.assert_close(total.float(), ref_total, rtol=1e-2, atol=1e-3)
    coef = 1.0 / (ref_total + 1e-6)
    for p, g0 in zip(ps, ref_grads):
        torch.testing.assert_close(p.grad.float(), (g0.float() * coef),
                                   rtol=1e-2, atol=1e-3)
