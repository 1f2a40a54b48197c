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
                                seed=0, strategy="channel_swap"):
    """Return a permutation of the input channels (dim 1) improving the 2:4
    kept magnitude.

    strategy="channel_swap": greedy sampled channel-pair swaps (fast, larger
    matrices). strategy="exhaustive": stripe-pair exhaustive repartitioning
    (deterministic, strictly monotone — the reference's Exhaustive_Search)."""
    if strategy == "exhaustive":
        return exhaustive_search(weight2d, m=m, n=n)
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


def _stripe_pair_partitions(m=4):
    """All unique ways to partition 2*m columns into two unordered m-wide
    stripes (C(2m-1, m-1) of them: column 0 pinned to the first stripe).
    For m=4 that is 35 candidate layouts per stripe pair — the reference's
    Exhaustive_Search unit (permutation_search_kernels/exhaustive_search)."""
    import itertools

    two_m = 2 * m
    perms = []
    for rest in itertools.combinations(range(1, two_m), m - 1):
        a = (0,) + rest
        b = tuple(c for c in range(two_m) if c not in a)
        perms.append(a + b)
    return torch.tensor(perms, dtype=torch.long)


def exhaustive_search(weight2d, m=4, n=2, max_sweeps=8):
    """Stripe-pair exhaustive permutation search.

    For every pair of m-wide stripes, evaluates ALL unique repartitions of
    their 2m columns and keeps the best; sweeps until a full pass makes no
    improvement. Deterministic, no randomness; strictly monotone in kept
    magnitude.

    On GPU (m=4, n=2) each sweep scores EVERY stripe pair x every
    repartition in ONE kernel launch (apex_amd._permutation_search — the
    MI355X analogue of the reference's permutation_search_kernels.cu) and
    greedily applies the best non-overlapping improvements; one
    device-to-host copy per sweep instead of one sync per pair.

    Returns the permutation of the input channels (dim 1).
    """
    w = weight2d.detach().abs().float()
    rows, cols = w.shape
    assert cols % m == 0
    nstripes = cols // m
    parts = _stripe_pair_partitions(m).to(w.device)  # [P, 2m]
    perm = torch.arange(cols, device=w.device)

    if w.is_cuda and m == 4 and n == 2 and nstripes >= 2:
        return _exhaustive_search_gpu(w, perm, parts, m, nstripes, max_sweeps)

    def pair_kept(cols2m):
        # cols2m: [rows, 2m] -> kept magnitude per candidate partition [P]
        wp = cols2m[:, parts]                       # [rows, P, 2m]
        g = wp.reshape(rows, parts.shape[0], 2, m)  # two stripes of m
        return g.topk(n, dim=3).values.sum(dim=(0, 2, 3))

    for _ in range(max_sweeps):
        improved = False
        for i in range(nstripes - 1):
            for j in range(i + 1, nstripes):
                idx = torch.cat([perm[i * m:(i + 1) * m], perm[j * m:(j + 1) * m]])
                kept = pair_kept(w[:, idx])
                best = int(kept.argmax())
                if best != 0 and float(kept[best]) > float(kept[0]) + 1e-7:
                    new = idx[parts[best]]
                    perm[i * m:(i + 1) * m] = new[:m]
                    perm[j * m:(j + 1) * m] = new[m:]
                    improved = True
        if not improved:
            break
    return perm.cpu()


def _exhaustive_search_gpu(w, perm, parts, m, nstripes, max_sweeps):
    from ..._ext import get_ext

    ps = get_ext("permutation_search")
    pairs = [(i, j) for i in range(nstripes - 1) for j in range(i + 1, nstripes)]
    pair_t = torch.tensor(pairs, dtype=torch.long, device=w.device)  # [np, 2]
    for _ in range(max_sweeps):
        # gather the 8 permuted column ids of every pair: [np, 8]
        stripes = perm.view(nstripes, m)
        cols8 = torch.cat([stripes[pair_t[:, 0]], stripes[pair_t[:, 1]]], dim=1)
        scores = ps.stripe_pair_scores(w, cols8, parts)        # [np, 35]
        best_val, best_p = scores.max(dim=1)
        gain = best_val - scores[:, 0]
        order = torch.argsort(gain, descending=True)
        # one host copy per sweep
        order_l = order.cpu().tolist()
        gain_l = gain.cpu().tolist()
        bp_l = best_p.cpu().tolist()
        used = set()
        improved = False
        for k in order_l:
            if gain_l[k] <= 1e-6:
                break
            i, j = pairs[k]
            if i in used or j in used:
                continue  # stripes already rewritten this sweep
            used.add(i)
            used.add(j)
            idx = cols8[k]
            new = idx[parts[bp_l[k]]]
            perm[i * m:(i + 1) * m] = new[:m]
            perm[j * m:(j + 1) * m] = new[m:]
            improved = True
        if not improved:
            break
    return perm.cpu()


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
