"""CPU tests for contrib: sparsity ASP, xentropy fallback, group_norm
fallback, focal loss reference, fast_layer_norm fallback."""

import torch
import pytest


def test_create_mask_2to4():
    from apex_amd.contrib.sparsity import create_mask

    torch.manual_seed(0)
    w = torch.randn(16, 32)
    mask = create_mask(w, "m4n2_1d")
    assert mask.shape == w.shape
    groups = mask.reshape(-1, 4).sum(dim=1)
    assert (groups == 2).all()
    # kept entries are the 2 largest per group
    wg = w.abs().reshape(-1, 4)
    kept = wg[mask.reshape(-1, 4)].reshape(-1, 2).min(dim=1).values
    dropped = wg[~mask.reshape(-1, 4)].reshape(-1, 2).max(dim=1).values
    assert (kept >= dropped - 1e-6).all()


def test_create_mask_2d():
    from apex_amd.contrib.sparsity import create_mask

    torch.manual_seed(1)
    w = torch.randn(8, 8)
    mask = create_mask(w, "m4n2_2d_best")
    m = mask.reshape(2, 4, 2, 4)
    # every 4-row and 4-col of each 4x4 block has exactly 2 kept
    assert (mask.reshape(8, 2, 4).sum(-1) == 2).all()


def test_asp_prune_and_step_reapplies_mask():
    from apex_amd.contrib.sparsity import ASP

    ASP._reset()
    torch.manual_seed(2)
    model = torch.nn.Sequential(torch.nn.Linear(16, 8), torch.nn.ReLU(), torch.nn.Linear(8, 4))
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    ASP.prune_trained_model(model, opt)
    w = model[0].weight
    assert (w.detach().reshape(-1, 4) != 0).sum(dim=1).max() <= 2
    # train a step; mask must be re-applied
    x = torch.randn(4, 16)
    loss = model(x).sum()
    loss.backward()
    opt.step()
    assert ((w.detach().reshape(-1, 4) != 0).sum(dim=1) <= 2).all()
    ASP._reset()


def test_group_norm_cpu_fallback():
    from apex_amd.contrib.group_norm import GroupNorm

    torch.manual_seed(3)
    gn = GroupNorm(4, 32)
    ref = torch.nn.GroupNorm(4, 32)
    with torch.no_grad():
        ref.weight.copy_(gn.weight)
        ref.bias.copy_(gn.bias)
    x = torch.randn(2, 32, 8, 8)
    torch.testing.assert_close(gn(x), ref(x))


def test_group_norm_silu_cpu():
    from apex_amd.contrib.group_norm import GroupNorm

    gn = GroupNorm(2, 8, act="silu")
    x = torch.randn(2, 8, 4, 4)
    expected = torch.nn.functional.silu(
        torch.nn.functional.group_norm(x, 2, gn.weight, gn.bias, gn.eps)
    )
    torch.testing.assert_close(gn(x), expected)


def test_fast_layer_norm_cpu_fallback():
    from apex_amd.contrib.layer_norm import FastLayerNorm

    ln = FastLayerNorm(64)
    x = torch.randn(4, 64)
    ref = torch.nn.functional.layer_norm(x, (64,), ln.weight, ln.bias, ln.epsilon)
    torch.testing.assert_close(ln(x), ref)


def test_xentropy_cpu_fallback():
    from apex_amd.contrib.xentropy import SoftmaxCrossEntropyLoss

    torch.manual_seed(4)
    logits = torch.randn(16, 32, requires_grad=True)
    labels = torch.randint(1, 32, (16,))
    losses = SoftmaxCrossEntropyLoss.apply(logits, labels, 0.1, 0, False)
    ref = torch.nn.functional.cross_entropy(logits.detach(), labels, reduction="none",
                                            label_smoothing=0.1)
    torch.testing.assert_close(losses, ref, rtol=1e-5, atol=1e-6)
    losses.sum().backward()
    x2 = logits.detach().clone().requires_grad_(True)
    torch.nn.functional.cross_entropy(x2, labels, reduction="none", label_smoothing=0.1).sum().backward()
    torch.testing.assert_close(logits.grad, x2.grad, rtol=1e-5, atol=1e-6)


def test_focal_loss_reference_math():
    from apex_amd.contrib.focal_loss import focal_loss

    torch.manual_seed(5)
    x = torch.randn(32, 16)
    y = torch.randint(-2, 16, (32,))
    nps = torch.tensor([max(float((y >= 0).sum()), 1.0)])
    loss = focal_loss(x, y, nps, 16, 0.25, 2.0, 0.0)
    assert torch.isfinite(loss)


def test_groupbn_nhwc_cpu():
    from apex_amd.contrib.groupbn import BatchNorm2d_NHWC

    torch.manual_seed(6)
    bn = BatchNorm2d_NHWC(8)
    bn.train()
    x = torch.randn(4, 5, 5, 8)
    y = bn(x)
    ref_bn = torch.nn.BatchNorm2d(8)
    ref_bn.train()
    with torch.no_grad():
        ref_bn.weight.copy_(bn.weight)
        ref_bn.bias.copy_(bn.bias)
    ref = ref_bn(x.permute(0, 3, 1, 2)).permute(0, 2, 3, 1)
    torch.testing.assert_close(y, ref, rtol=1e-4, atol=1e-5)
    # add+relu fusion path
    z = torch.randn_like(y)
    y2 = BatchNorm2d_NHWC(8, fuse_relu=True)
    y2.train()
    out = y2(x, z)
    assert (out >= 0).all()


def test_permutation_search_improves_kept_magnitude():
    from apex_amd.contrib.sparsity.permutation_search import (
        efficacy, search_for_good_permutation, apply_permutation_in_place,
    )

    torch.manual_seed(7)
    # adversarial: each 4-group has correlated magnitudes so permutation helps
    w = torch.randn(32, 64)
    w[:, ::4] *= 5.0  # big channels clustered into the same group positions
    before = efficacy(w)
    perm = search_for_good_permutation(w, max_iters=20)
    after = efficacy(w[:, perm])
    assert after >= before  # never worse
    lin = torch.nn.Linear(64, 32)
    with torch.no_grad():
        lin.weight.copy_(w)
    apply_permutation_in_place(lin, perm)
    torch.testing.assert_close(lin.weight.detach(), w[:, perm])


def test_exhaustive_permutation_search():
    from apex_amd.contrib.sparsity.permutation_search import (
        efficacy, exhaustive_search, search_for_good_permutation,
    )

    torch.manual_seed(3)
    # craft a matrix where the identity grouping is pessimal: large-magnitude
    # columns packed into the same stripes so 2:4 must drop half of them
    rows, cols = 16, 32
    w = torch.rand(rows, cols) * 0.01
    w[:, 0:4] += 10.0  # stripe 0 all-large; optimal splits them across stripes
    w[:, 4:8] += 10.0
    base = efficacy(w)
    perm = exhaustive_search(w)
    assert sorted(perm.tolist()) == list(range(cols))  # valid permutation
    improved = efficacy(w[:, perm])
    assert improved > base * 1.5  # spreading the 8 big cols ~doubles kept mass
    # strategy dispatch matches
    perm2 = search_for_good_permutation(w, strategy="exhaustive")
    assert torch.equal(perm, perm2)
    # monotone: re-running on the permuted matrix cannot reduce efficacy
    perm3 = exhaustive_search(w[:, perm])
    assert efficacy(w[:, perm][:, perm3]) >= improved - 1e-5


def test_permutation_propagation_preserves_function():
    """FX-walked group permutation is a pure re-parameterization: model
    outputs are unchanged while 2:4 kept magnitude improves."""
    from apex_amd.contrib.sparsity.permutation_lib import (
        find_permutation_groups, permute_model_for_sparsity,
    )
    from apex_amd.contrib.sparsity.permutation_search import efficacy

    class Res(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.lin1 = torch.nn.Linear(16, 32)
            self.ln = torch.nn.LayerNorm(32)
            self.lin2 = torch.nn.Linear(32, 32)
            self.lin3 = torch.nn.Linear(32, 8)

        def forward(self, x):
            h = self.ln(self.lin1(x))
            h = h + torch.relu(self.lin2(h))
            return self.lin3(h)

    torch.manual_seed(11)
    m = Res()
    with torch.no_grad():  # make some channels dominant so permutation helps
        m.lin1.weight[:4] *= 10
    groups = find_permutation_groups(m)
    assert len(groups) == 1
    g = groups[0]
    assert len(g.consumers) == 2 and len(g.producers) == 2 and len(g.norms) == 1

    x = torch.randn(8, 16)
    before_out = m(x)
    before_eff = efficacy(m.lin3.weight.detach()) + efficacy(m.lin2.weight.detach())
    n = permute_model_for_sparsity(m)
    assert n == 1
    after_out = m(x)
    after_eff = efficacy(m.lin3.weight.detach()) + efficacy(m.lin2.weight.detach())
    torch.testing.assert_close(after_out, before_out, rtol=1e-4, atol=1e-5)
    assert after_eff >= before_eff - 1e-4


def test_permutation_propagation_rejects_leaky_group():
    """If the stream feeding a consumer is also returned directly, permuting
    it would change the model output — the group must be dropped."""
    from apex_amd.contrib.sparsity.permutation_lib import find_permutation_groups

    class Leaky(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.lin1 = torch.nn.Linear(16, 32)
            self.lin2 = torch.nn.Linear(32, 8)

        def forward(self, x):
            h = self.lin1(x)
            return self.lin2(h), h  # h escapes un-permuted

    assert find_permutation_groups(Leaky()) == []


def test_asp_allow_permutation_end_to_end():
    from apex_amd.contrib.sparsity import ASP

    torch.manual_seed(2)
    model = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 8))
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    x = torch.randn(4, 16)
    before = model(x)
    ASP._reset()
    ASP.init_model_for_pruning(model, mask_calculator="m4n2_1d", verbosity=0,
                               allow_permutation=True)
    ASP.init_optimizer_for_pruning(opt)
    ASP.compute_sparse_masks()
    # weights are now 2:4 masked; each 4-group along dim1 has >=2 zeros
    w = model[2].weight.detach()
    zeros_per_group = (w.reshape(w.shape[0], -1, 4) == 0).sum(-1)
    assert (zeros_per_group >= 2).all()
    ASP._reset()


def test_index_mul_2d_double_backward_cpu():
    from apex_amd.contrib.index_mul_2d import index_mul_2d

    torch.manual_seed(0)
    in1 = torch.randn(6, 8, requires_grad=True)
    in2 = torch.randn(4, 8, requires_grad=True)
    idx = torch.tensor([0, 2, 2, 5])
    out = index_mul_2d(in1, in2, idx)
    g1, = torch.autograd.grad(out.sum(), in1, create_graph=True)
    gg, = torch.autograd.grad(g1.sum(), in2)
    torch.testing.assert_close(gg, torch.ones_like(in2))


def test_exhaustive_search_monotone_random():
    """Property: the found permutation never reduces 2:4 kept magnitude, on
    arbitrary random matrices (not just crafted ones)."""
    from apex_amd.contrib.sparsity.permutation_search import efficacy, exhaustive_search

    for seed in range(5):
        torch.manual_seed(seed)
        rows = 8 + seed * 3
        cols = 16 + 8 * (seed % 3)
        w = torch.randn(rows, cols) * torch.rand(1, cols).exp()
        base = efficacy(w)
        perm = exhaustive_search(w)
        assert sorted(perm.tolist()) == list(range(cols))
        assert efficacy(w[:, perm]) >= base - 1e-5


def test_xentropy_padding_idx_masks_loss_and_grad():
    # every padding_idx target contributes zero loss AND zero grad
    from apex_amd.contrib.xentropy import SoftmaxCrossEntropyLoss

    torch.manual_seed(6)
    logits = torch.randn(12, 20, requires_grad=True)
    labels = torch.randint(0, 20, (12,))
    labels[::3] = 7  # padding rows
    losses = SoftmaxCrossEntropyLoss.apply(logits, labels, 0.1, 7, False)
    assert torch.all(losses[::3] == 0)
    losses.sum().backward()
    assert torch.all(logits.grad[::3] == 0)
    # non-padding rows match torch CE
    keep = torch.ones(12, dtype=torch.bool)
    keep[::3] = False
    ref = torch.nn.functional.cross_entropy(
        logits.detach()[keep], labels[keep], reduction="none", label_smoothing=0.1)
    torch.testing.assert_close(losses[keep], ref, rtol=1e-5, atol=1e-6)


def test_xentropy_full_smoothing_uniform_target():
    # smoothing=1.0: the target is uniform over classes — loss equals the
    # mean negative log prob; grads are softmax(p) - 1/K
    from apex_amd.contrib.xentropy import SoftmaxCrossEntropyLoss

    torch.manual_seed(7)
    K = 10
    logits = torch.randn(6, K, requires_grad=True)
    labels = torch.randint(1, K, (6,))
    losses = SoftmaxCrossEntropyLoss.apply(logits, labels, 1.0, 0, False)
    ref = torch.nn.functional.cross_entropy(logits.detach(), labels,
                                            reduction="none", label_smoothing=1.0)
    torch.testing.assert_close(losses, ref, rtol=1e-5, atol=1e-6)
    losses.sum().backward()
    expected = torch.softmax(logits.detach(), -1) - 1.0 / K
    torch.testing.assert_close(logits.grad, expected, rtol=1e-5, atol=1e-6)


def test_focal_loss_gamma0_matches_weighted_bce():
    # gamma=0, alpha=0.5 reduces focal loss to (0.5x) sigmoid BCE summed over
    # real classes / num_positives
    from apex_amd.contrib.focal_loss import focal_loss

    torch.manual_seed(8)
    N, K = 24, 12
    x = torch.randn(N, K, requires_grad=True)
    y = torch.randint(-1, K, (N,))  # -1 = negative sample rows
    nps = torch.tensor([max(float((y >= 0).sum()), 1.0)])
    loss = focal_loss(x, y, nps, K, 0.5, 0.0, 0.0)

    onehot = torch.zeros(N, K)
    valid = y >= 0
    onehot[valid] = torch.nn.functional.one_hot(y[valid], K).float()
    bce = torch.nn.functional.binary_cross_entropy_with_logits(
        x.detach(), onehot, reduction="none")
    expected = 0.5 * bce.sum() / nps
    torch.testing.assert_close(loss.reshape(()), expected.reshape(()),
                               rtol=1e-4, atol=1e-5)
