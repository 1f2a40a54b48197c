"""Fused softmax cross-entropy with label smoothing.

API parity with the reference ``apex.contrib.xentropy.SoftmaxCrossEntropyLoss``
(apex/contrib/xentropy/softmax_xentropy.py:6-33): saves only logits +
``max_log_sum_exp`` (the probabilities are recomputed in backward), with
``padding_idx`` masking applied on the host side.
"""

import torch

from ..._ext import get_ext


class SoftmaxCrossEntropyLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, smoothing=0.0, padding_idx=0, half_to_float=False):
        if logits.is_cuda:
            ext = get_ext("xentropy")
            losses, max_log_sum_exp = ext.forward(logits, labels, smoothing, half_to_float)
        else:
            xf = logits.float()
            max_log_sum_exp = torch.logsumexp(xf, dim=-1)
            xy = xf.gather(-1, labels.unsqueeze(-1)).squeeze(-1)
            losses = max_log_sum_exp - (1.0 - smoothing) * xy - smoothing * xf.mean(-1)
            if not half_to_float:
                losses = losses.to(logits.dtype)
        losses.masked_fill_(labels == padding_idx, 0)
        ctx.save_for_backward(logits, max_log_sum_exp, labels,
                              torch.FloatTensor([smoothing]),
                              torch.LongTensor([padding_idx]))
        return losses

    @staticmethod
    def backward(ctx, grad_loss):
        logits, max_log_sum_exp, labels, smoothing, padding_idx = ctx.saved_tensors
        smoothing = smoothing.item()
        padding_idx = padding_idx.item()
        if not grad_loss.is_contiguous():
            grad_loss = grad_loss.contiguous()
        grad_loss = grad_loss.clone()
        grad_loss.masked_fill_(labels == padding_idx, 0)
        if logits.is_cuda:
            ext = get_ext("xentropy")
            grad_logits = ext.backward(grad_loss, logits, max_log_sum_exp, labels, smoothing)
        else:
            C = logits.size(-1)
            p = torch.exp(logits.float() - max_log_sum_exp.unsqueeze(-1))
            target = torch.full_like(p, smoothing / C)
            target.scatter_add_(
                -1, labels.unsqueeze(-1), torch.full_like(labels, 1.0 - smoothing, dtype=p.dtype).unsqueeze(-1)
            )
            grad_logits = (grad_loss.float().unsqueeze(-1) * (p - target)).to(logits.dtype)
        return grad_logits, None, None, None, None
