"""Channel-permutation search for 2:4 structured sparsity.

Counterpart of the reference ``apex.contrib.sparsity.permutation_lib`` /
``permutation_search_kernels`` (channel_swap strategy): permute the INPUT
channels of a weight matrix so that the magnitude kept by the N:M mask is
maximized, improving pruned-model accuracy. The permutation must then be
applied consistently to the producing layer's output channels (handled by
the caller / ASP integration).

Implemented strategy: greedy bounded channel-swap (the reference's
``channel_swap.py`` approach): repeatedly try swapping channel pairs across
stripe groups, keeping swaps that increase the total kept magnitude, until
convergence or the escape budget runs out. All math is torch ops (runs on
GPU tensors directly).
"""

import torch

from .sparse_masklib import create_mask


def efficacy(weight2d, pattern="m4n2_1d"):
    """Total |weight| kept by the mask."""
    mask = create_mask(weight2d, pattern)
    return float((weight2d.detach().abs() * mask).sum())


def _group_kept_sum(w_abs, m=4, n=2):
    """Sum of the n largest |w| in every m-wide group along dim 1, per
    group-column block: returns total kept magnitude."""
    cols = w_abs.shape[1]
    g = w_abs.reshape(w_abs.shape[0], cols // m, m)
    top = g.topk(n, dim=2).values
    return top.sum()


def search_for_good_permutation(weight2d, m=4, n=2, max_iters=100, escape_attempts=10,
                                seed=0):
    """Return a permutation of the input channels (dim 1) improving the 2:4
    kept magnitude. Greedy channel-swap with random restarts."""
    w = weight2d.detach().abs().float()
    rows, cols = w.shape
    assert cols % m == 0
    gen = torch.Generator(device="cpu").manual_seed(seed)
    perm = torch.arange(cols)
    best_total = float(_group_kept_sum(w, m, n))

    def kept_for_groups(wp, gidx):
        g = wp[:, gidx * m:(gidx + 1) * m]
        return g.topk(n, dim=1).values.sum()

    improved = True
    iters = 0
    while improved and iters < max_iters:
        improved = False
        iters += 1
        # sample candidate swap pairs across different groups
        ncand = min(256, cols * 2)
        ca = torch.randint(0, cols, (ncand,), generator=gen)
        cb = torch.randint(0, cols, (ncand,), generator=gen)
        for a, b in zip(ca.tolist(), cb.tolist()):
            ga, gb = a // m, b // m
            if ga == gb:
                continue
            wp = w[:, perm]
            before = kept_for_groups(wp, ga) + kept_for_groups(wp, gb)
            perm[a], perm[b] = perm[b].item(), perm[a].item()
            wp = w[:, perm]
            after = kept_for_groups(wp, ga) + kept_for_groups(wp, gb)
            if float(after) > float(before) + 1e-7:
                improved = True
            else:
                perm[a], perm[b] = perm[b].item(), perm[a].item()  # revert
    return perm


def apply_permutation_in_place(module, perm):
    """Permute a Linear/Conv1x1 weight's input channels by ``perm``."""
    with torch.no_grad():
        if module.weight.dim() == 2:
            module.weight.copy_(module.weight[:, perm])
        elif module.weight.dim() == 4 and module.weight.shape[2:] == (1, 1):
            module.weight.copy_(module.weight[:, perm, :, :])
        else:
            raise RuntimeError("permutation only applies to Linear / 1x1 Conv weights")
    return module
