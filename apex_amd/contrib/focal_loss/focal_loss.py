"""Fused sigmoid focal loss (loss + partial gradient in one kernel).

API parity with the reference ``apex.contrib.focal_loss``
(apex/contrib/focal_loss/focal_loss.py:6-60). Targets ``y`` per anchor:
``y == -2`` ignores the anchor, ``y >= 0`` marks class ``y`` positive and all
other real classes negative; classes ``>= num_real_classes`` are padding.
"""

import torch

from ..._ext import get_ext


def _ref_focal(cls_output, targets, num_positives_sum, num_real_classes, alpha, gamma, smoothing):
    x = cls_output.float()
    C = x.size(-1)
    y = targets.long()
    half_s = smoothing / 2.0
    t = torch.full_like(x, half_s)
    pos_mask = y >= 0
    t1 = t.reshape(-1, C)
    flat_y = y.reshape(-1)
    rows = torch.arange(flat_y.numel(), device=x.device)
    sel = pos_mask.reshape(-1)
    t1[rows[sel], flat_y[sel].clamp(min=0)] = 1.0 - half_s
    t = t1.reshape(x.shape)
    if smoothing == 0.0:
        t = (t > 0.5).float()
    sigma = torch.sigmoid(x)
    ce = torch.nn.functional.binary_cross_entropy_with_logits(x, t, reduction="none")
    is_pos = torch.zeros_like(x, dtype=torch.bool)
    p1 = is_pos.reshape(-1, C)
    p1[rows[sel], flat_y[sel].clamp(min=0)] = True
    is_pos = p1.reshape(x.shape)
    pt = torch.where(is_pos, sigma, 1 - sigma)
    a_t = torch.where(is_pos, torch.full_like(x, alpha), torch.full_like(x, 1 - alpha))
    loss = a_t * (1 - pt) ** gamma * ce
    # mask ignored anchors and pad classes
    ignore = (y == -2).unsqueeze(-1).expand_as(loss)
    loss = loss.masked_fill(ignore, 0.0)
    if num_real_classes < C:
        loss[..., num_real_classes:] = 0.0
    return loss.sum() / num_positives_sum.float().squeeze()


class FocalLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, cls_output, cls_targets_at_level, num_positives_sum, num_real_classes,
                alpha, gamma, label_smoothing=0.0):
        ext = get_ext("focal_loss")
        loss, partial_grad = ext.forward(
            cls_output, cls_targets_at_level, num_positives_sum, num_real_classes,
            alpha, gamma, label_smoothing,
        )
        ctx.save_for_backward(partial_grad, num_positives_sum)
        return loss

    @staticmethod
    def backward(ctx, grad_loss):
        partial_grad, num_positives_sum = ctx.saved_tensors
        ext = get_ext("focal_loss")
        # in-place: partial_grad IS grad_input (reference behavior)
        grad_input = ext.backward(grad_loss.contiguous(), partial_grad, num_positives_sum)
        return grad_input, None, None, None, None, None, None


def focal_loss(cls_output, cls_targets_at_level, num_positives_sum, num_real_classes,
               alpha, gamma, label_smoothing=0.0):
    """Fused focal loss (falls back to reference torch math on CPU)."""
    if not cls_output.is_cuda:
        return _ref_focal(cls_output, cls_targets_at_level, num_positives_sum,
                          num_real_classes, alpha, gamma, label_smoothing)
    return FocalLoss.apply(cls_output, cls_targets_at_level, num_positives_sum,
                           num_real_classes, alpha, gamma, label_smoothing)
