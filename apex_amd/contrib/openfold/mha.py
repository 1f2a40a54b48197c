"""OpenFold multi-head attention (reference: openfold_triton's AttnTri),
composed from the library's fused softmax and hipBLASLt GEMMs."""

import math

import torch

from ...transformer import scaled_masked_softmax


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
