"""RNN-T transducer joint and loss.

API parity with the reference ``apex.contrib.transducer``
(apex/contrib/transducer/transducer.py: TransducerJoint:6, TransducerLoss:88).
Notes:
- packed layouts are supported (``pack_output`` with ``batch_offset`` =
  inclusive cumsum of f_len*g_len, and ``packed_input`` for the loss);
- dropout is FUSED into the joint kernels (round 2): a Philox counter keyed
  on the flat output index generates the mask inside forward AND both
  backward reductions — mask-free backward, exactly the reference's
  philox.cuh design (apex/contrib/csrc/transducer/philox.cuh);
- ``fuse_softmax_backward`` is accepted; the loss consumes log-probs and
  returns grads w.r.t. them (the log_softmax backward is chained by autograd
  rather than fused into the loss kernel).
"""

import torch

from ..._ext import get_ext


class TransducerJointFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, f, g, f_len, g_len, relu, batch_offset=None, packed_batch=0,
                dropout_prob=0.0):
        ext = get_ext("transducer")
        # philox seed drawn from torch's CPU RNG: deterministic under
        # torch.manual_seed, shared by fwd and the mask-free backward
        seed = int(torch.randint(0, 2 ** 62, (1,)).item()) if dropout_prob > 0 else 0
        if batch_offset is not None:
            # exclusive offsets from the reference's inclusive cumsum
            off = torch.cat([batch_offset.new_zeros(1), batch_offset[:-1]])
            (out,) = ext.joint_forward_packed(f, g, f_len, g_len, off, packed_batch, relu,
                                              dropout_prob, seed)
            ctx.off = off
        else:
            (out,) = ext.joint_forward(f, g, f_len, g_len, relu, dropout_prob, seed)
            ctx.off = None
        ctx.save_for_backward(out, f_len, g_len)
        ctx.dims = (f.size(0), f.size(1), g.size(1), f.size(2))
        ctx.relu = relu
        ctx.dropout = (dropout_prob, seed)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        ext = get_ext("transducer")
        out, f_len, g_len = ctx.saved_tensors
        B, T, U, H = ctx.dims
        p, seed = ctx.dropout
        if ctx.off is not None:
            df, dg = ext.joint_backward_packed(grad_out, out, f_len, g_len, ctx.off, B, T, U, H,
                                               ctx.relu, p, seed)
        else:
            df, dg = ext.joint_backward(grad_out, out, f_len, g_len, B, T, U, H, ctx.relu,
                                        p, seed)
        return df, dg, None, None, None, None, None, None


class TransducerJoint(torch.nn.Module):
    def __init__(self, pack_output=False, relu=False, dropout=False, opt=1, fwd_tile_size=4,
                 dropout_prob=0.0, probe_mask=False):
        super().__init__()
        self.pack_output = pack_output
        self.relu = relu
        self.dropout = dropout
        self.dropout_prob = dropout_prob
        self.mask_probe = [] if (relu or dropout) and probe_mask else None

    def forward(self, f, g, f_len, g_len, batch_offset=None, packed_batch=0):
        p = self.dropout_prob if (self.dropout and self.training) else 0.0
        if self.pack_output:
            assert batch_offset is not None and packed_batch > 0, \
                "pack_output needs batch_offset (cumsum of f_len*g_len) and packed_batch"
            return TransducerJointFunc.apply(f, g, f_len, g_len, self.relu, batch_offset,
                                             packed_batch, p)
        if not f.is_cuda:
            # reference math on CPU
            out = f.unsqueeze(2) + g.unsqueeze(1)
            mask_t = torch.arange(f.size(1), device=f.device)[None, :, None, None] < f_len[:, None, None, None]
            mask_u = torch.arange(g.size(1), device=g.device)[None, None, :, None] < g_len[:, None, None, None]
            out = out * (mask_t & mask_u)
            if self.relu:
                out = torch.relu(out)
            if p > 0:
                out = torch.nn.functional.dropout(out, p=p)
            return out
        return TransducerJointFunc.apply(f, g, f_len, g_len, self.relu, None, 0, p)


class TransducerLossFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, label, f_len, y_len, blank_idx, batch_offset=None, max_f_len=0):
        ext = get_ext("transducer")
        off = None
        if batch_offset is not None:
            off = torch.cat([batch_offset.new_zeros(1), batch_offset[:-1]])
        losses, alpha = ext.loss_forward(x, label, f_len, y_len, blank_idx, off, max_f_len)
        ctx.save_for_backward(x, label, alpha, f_len, y_len)
        ctx.blank_idx = blank_idx
        ctx.off = off
        ctx.max_f_len = max_f_len
        return losses

    @staticmethod
    def backward(ctx, grad_loss):
        ext = get_ext("transducer")
        x, label, alpha, f_len, y_len = ctx.saved_tensors
        dx = ext.loss_backward(x, label, alpha, grad_loss.contiguous(), f_len, y_len,
                               ctx.blank_idx, ctx.off, ctx.max_f_len)
        return dx, None, None, None, None, None, None


def _ref_rnnt_loss(x, label, f_len, y_len, blank):
    """Pure-python alpha DP on log-probs (CPU fallback + numerics oracle)."""
    B = x.shape[0]
    losses = []
    for b in range(B):
        T = int(f_len[b])
        U = int(y_len[b]) + 1
        xb = x[b].float()
        alpha = torch.full((T, U), float("-inf"))
        alpha[0, 0] = 0.0
        for t in range(T):
            for u in range(U):
                if t == 0 and u == 0:
                    continue
                cands = []
                if t > 0:
                    cands.append(alpha[t - 1, u] + xb[t - 1, u, blank])
                if u > 0:
                    cands.append(alpha[t, u - 1] + xb[t, u - 1, label[b, u - 1]])
                alpha[t, u] = torch.logsumexp(torch.stack(cands), 0)
        losses.append(-(alpha[T - 1, U - 1] + xb[T - 1, U - 1, blank]))
    return torch.stack(losses)


class TransducerLoss(torch.nn.Module):
    def __init__(self, fuse_softmax_backward=True, opt=1, packed_input=False):
        super().__init__()
        self.packed_input = packed_input
        self.fuse_softmax_backward = fuse_softmax_backward

    def forward(self, x, label, f_len, y_len, blank_idx=0, batch_offset=None, max_f_len=None,
                debug_list=None):
        """x: log-probs [B, T, U, V] (dense) or [packed_rows, V] with
        batch_offset = cumsum(f_len*(y_len+1)); label: [B, U-1]."""
        if self.packed_input:
            assert batch_offset is not None and max_f_len is not None, \
                "packed_input needs batch_offset (cumsum of f_len*(y_len+1)) and max_f_len"
            return TransducerLossFunc.apply(x, label, f_len, y_len, blank_idx, batch_offset,
                                            max_f_len)
        if not x.is_cuda:
            return _ref_rnnt_loss(x, label, f_len, y_len, blank_idx)
        return TransducerLossFunc.apply(x, label, f_len, y_len, blank_idx, None, 0)
