"""N:M structured-sparsity mask generation.

API parity with the reference ``apex.contrib.sparsity.sparse_masklib``
(create_mask(tensor, pattern="m4n2_1d")): masks keep the n largest-magnitude
elements of every m-wide group along the last dimension. Supported patterns:
``m<M>n<N>_1d`` (e.g. m4n2_1d — the 2:4 pattern MFMA sparsity consumes) and
``m4n2_2d_best`` (greedy 4x4 block refinement, matching the reference's 2d
option).
"""

import re

import torch


def _mn_1d_mask(weight2d, m, n):
    numel = weight2d.numel()
    assert numel % m == 0, f"tensor numel {numel} not divisible by group size {m}"
    groups = weight2d.detach().abs().reshape(-1, m)
    idx = torch.argsort(groups, dim=1, descending=True)[:, :n]
    mask = torch.zeros_like(groups, dtype=torch.bool)
    mask.scatter_(1, idx, True)
    return mask.reshape(weight2d.shape)


_VALID_2D_MASKS = {}


def _valid_2d_masks(m, n, device):
    """All m x m binary matrices with EXACTLY n per row and n per column
    (for 4:2 there are 90). Computed once, cached per device."""
    key = (m, n, str(device))
    if key in _VALID_2D_MASKS:
        return _VALID_2D_MASKS[key]
    import itertools

    row_choices = list(itertools.combinations(range(m), n))
    masks = []
    for rows in itertools.product(row_choices, repeat=m):
        col_cnt = [0] * m
        for r in rows:
            for c in r:
                col_cnt[c] += 1
        if all(c == n for c in col_cnt):
            mk = torch.zeros(m, m, dtype=torch.float32)
            for i, r in enumerate(rows):
                for c in r:
                    mk[i, c] = 1.0
            masks.append(mk)
    out = torch.stack(masks).to(device)
    _VALID_2D_MASKS[key] = out
    return out


def _mn_2d_best_mask(weight2d, m, n):
    """Exact 2d best: for each m x m block pick the doubly-n:m mask with the
    largest kept magnitude (exhaustive over the valid mask set, vectorized
    over blocks — reference semantics of m4n2_2d_best)."""
    rows, cols = weight2d.shape
    assert rows % m == 0 and cols % m == 0
    w = weight2d.detach().abs().float()
    valid = _valid_2d_masks(m, n, w.device)  # [V, m, m]
    blocks = w.reshape(rows // m, m, cols // m, m).permute(0, 2, 1, 3).reshape(-1, m, m)
    scores = torch.einsum("bij,vij->bv", blocks, valid)
    best = scores.argmax(dim=1)
    bmasks = valid[best].bool()  # [B, m, m]
    mask = bmasks.reshape(rows // m, cols // m, m, m).permute(0, 2, 1, 3).reshape(rows, cols)
    return mask


def create_mask(tensor, pattern="m4n2_1d"):
    """Return a bool mask with the same shape as ``tensor``."""
    shape = tensor.shape
    t2d = tensor.reshape(-1, shape[-1]) if tensor.dim() != 2 else tensor
    match = re.fullmatch(r"m(\d+)n(\d+)_(1d|2d_best|2d_greedy)", pattern)
    if not match:
        raise ValueError(f"unsupported sparsity pattern {pattern}")
    m, n, kind = int(match.group(1)), int(match.group(2)), match.group(3)
    if kind == "1d":
        mask = _mn_1d_mask(t2d, m, n)
    else:
        mask = _mn_2d_best_mask(t2d, m, n)
    return mask.reshape(shape).to(tensor.device)
