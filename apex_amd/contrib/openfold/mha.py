"""OpenFold multi-head attention (reference: openfold_triton's AttnTri /
AttnBiasJIT / AttnNoBiasJIT / CanSchTriMHA, mha.py:36-469), composed from
the library's fused softmax and hipBLASLt GEMMs."""

import math

import torch

from ...transformer import scaled_masked_softmax

_MHA_ENABLED = True


def enable():
    global _MHA_ENABLED
    _MHA_ENABLED = True


def disable():
    global _MHA_ENABLED
    _MHA_ENABLED = False


def CanSchTriMHA(in_shape, has_bias=True, inf=1e9, training=True):
    """Accelerated-path predicate (reference mha.py:36: a hard-coded list of
    OpenFold shapes the Triton kernel was tuned for). The MI355X composition
    is shape-generic — any [*, Q|S, H, S, D] attention runs on the fused
    wave64 softmax kernels — so this answers only the semantic constraints:
    the module-level enable()/disable() switch and the reference's
    ``inf == 1e9`` contract (the fused kernel folds the mask additively)."""
    if not _MHA_ENABLED:
        return False
    if inf != 1e9:
        return False
    return len(in_shape) >= 4


def _attention_bias(query, key, value, mask, bias, inf):
    """Eager reference attention, OpenFold float-mask convention
    (mask 1.0 = keep, 0.0 = drop; additive -inf fold)."""
    scale = 1.0 / math.sqrt(query.size(-1))
    logits = torch.matmul(query * scale, torch.swapdims(key, -2, -1))
    logits = logits + (mask - 1.0) * inf
    if bias is not None:
        logits = logits + bias
    return torch.matmul(torch.softmax(logits, dim=-1), value)


def _attention_no_bias(query, key, value, mask, inf):
    return _attention_bias(query, key, value, mask, None, inf)


try:  # compiled variants (reference: AttnBiasJIT/AttnNoBiasJIT, mha.py:468)
    AttnBiasJIT = torch.compile(_attention_bias)
    AttnNoBiasJIT = torch.compile(_attention_no_bias)
except Exception:  # pragma: no cover - inductor unavailable
    AttnBiasJIT = _attention_bias
    AttnNoBiasJIT = _attention_no_bias


def AttnTri(q, k, v, mask, bias=None, inf=1e9):
    """q, k, v: [..., heads, seq, dim]; mask: broadcastable bool (True = keep
    in OpenFold convention — converted to the masked-softmax convention)."""
    scale = 1.0 / math.sqrt(q.shape[-1])
    scores = torch.matmul(q, k.transpose(-2, -1))
    if bias is not None:
        # fused kernel computes softmax(scale * x); we need
        # softmax(scale * qk + bias), so fold bias as bias/scale into x
        scores = scores + bias / scale
    # fold arbitrary leading dims into [b, np, sq, sk] for the fused kernel
    shape = scores.shape
    s4 = scores.reshape(-1, shape[-3], shape[-2], shape[-1])
    if mask is not None:
        m = (~mask.to(torch.bool)).expand(shape).reshape(s4.shape)
        m = m[:, :1].contiguous()  # kernel broadcasts over heads
        probs = scaled_masked_softmax(s4.contiguous(), m, scale)
    else:
        from ...transformer import scaled_softmax

        probs = scaled_softmax(s4.contiguous(), scale)
    probs = probs.reshape(shape)
    return torch.matmul(probs, v)
