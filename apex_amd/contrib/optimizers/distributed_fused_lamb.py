"""DistributedFusedLAMB — ZeRO-style sharded LAMB over RCCL/xGMI.

API and behavior parity with the reference
``apex.contrib.optimizers.DistributedFusedLAMB``
(apex/contrib/optimizers/distributed_fused_lamb.py:26-1333, the MLPerf-BERT
pipeline). Round-2 deep form — every step-phase is a fused launch:

* grad-norm partials accumulate PER BUCKET as each reduce-scatter finishes
  (``_on_bucket_grad_synced``, running in stream-order behind the comm
  stream) — the reference's dedicated ``_l2_grad_norm_st`` norm stream,
  overlapped with the tail of backward;
* stage 1 (Adam-style update with in-kernel global-norm clip) is ONE
  ``multi_tensor_lamb_stage1`` launch per param group over all its bucket
  shards — the global grad norm stays a device tensor end to end (no host
  sync on the norm path);
* per-tensor trust-ratio norms: each rank's shard holds one contiguous
  SEGMENT of each param; one ``multi_tensor_l2norm(per_tensor=True)`` over
  all segments (params) + one over all updates, squared and scattered into a
  [n_tensors, 2] vector, ONE all_reduce, sqrt — the cross-rank norm exchange
  is a single small collective per step (reference: chunked reductions
  :943-1008);
* stage 2 (trust-ratio apply) is one ``multi_tensor_lamb_stage2`` launch per
  group over the same segments with the reduced norms;
* updated shards all-gather per bucket on the comm stream, pipelined behind
  the stage-2 launches (same machinery as DistributedFusedAdam).

``clip_after_ar=True`` (default) clips by the norm of the REDUCED grads;
``clip_after_ar=False`` uses the norm of the local pre-reduction grads
(accumulated in the backward hooks before the collective) — the reference's
pre-AR clip fused into the pipeline.
"""

import torch
import torch.distributed as dist

from ..._ext import get_ext
from ...multi_tensor_apply import multi_tensor_applier
from .distributed_fused_adam import DistributedFusedAdam


class DistributedFusedLAMB(DistributedFusedAdam):
    def __init__(
        self,
        params,
        lr=1e-3,
        bias_correction=True,
        betas=(0.9, 0.999),
        eps=1e-6,
        weight_decay=0.01,
        max_grad_norm=1.0,
        adam_w_mode=True,
        grad_averaging=True,
        use_nvlamb=False,
        clip_after_ar=True,
        **kwargs,
    ):
        for unsupported in ("store_param_remainders", "with_scaled_states", "capturable"):
            if kwargs.get(unsupported):
                raise ValueError(f"DistributedFusedLAMB does not support {unsupported}")
        super().__init__(
            params, lr=lr, bias_correction=bias_correction, betas=betas, eps=eps,
            adam_w_mode=adam_w_mode, weight_decay=weight_decay, **kwargs,
        )
        self.max_grad_norm = max_grad_norm
        self.grad_averaging = grad_averaging
        self.use_nvlamb = use_nvlamb
        self.clip_after_ar = clip_after_ar
        self._global_scale = 1.0
        # device-side grad-norm accumulator (squared); partials land here as
        # buckets finish their reductions
        self._gnorm_sq = torch.zeros(1, dtype=torch.float32, device=self.device)
        self._lamb_plan = None

    def set_global_scale(self, global_scale):
        """External loss-scaler hookup (reference :1222)."""
        self._global_scale = float(global_scale)

    def complete_reductions(self):
        """Reference :1235 — finish all grad reductions."""
        self.grad_sync()

    # ---- overlapped grad-norm partials (reference _l2_grad_norm_st) ----
    def _on_bucket_grad_synced(self, b):
        if self.clip_after_ar:
            # stream-ordered behind the comm stream (the caller just waited
            # it), so this overlaps the NEXT buckets' reductions
            self._gnorm_sq += b.grad_shard.float().pow(2).sum()

    def _on_bucket_pre_reduce(self, b):
        if not self.clip_after_ar:
            # pre-AR norm: the local accumulated grads, before the collective
            self._gnorm_sq += b.grad_data.float().pow(2).sum()

    # ---- fused plan over bucket shards ----
    def _build_lamb_plan(self):
        n_tensors = 0
        tensor_index = {}
        for b in self.buckets:
            for p, _ in b.params:
                if id(p) not in tensor_index:
                    tensor_index[id(p)] = n_tensors
                    n_tensors += 1
        by_group = {}
        for b in self.buckets:
            by_group.setdefault(id(b.group), {"group": b.group, "buckets": [],
                                              "seg_p": [], "seg_u": [], "idx": []})
            ent = by_group[id(b.group)]
            ent["buckets"].append(b)
            lo = self.rank * b.shard_size
            hi = lo + b.shard_size
            for p, offset in b.params:
                s0, s1 = max(offset, lo), min(offset + p.numel(), hi)
                if s0 >= s1:
                    continue
                seg = slice(s0 - lo, s1 - lo)
                ent["seg_p"].append(b.master_shard[seg])
                ent["seg_u"].append(b.grad_shard[seg])
                ent["idx"].append(tensor_index[id(p)])
        for ent in by_group.values():
            ent["idx_t"] = torch.tensor(ent["idx"], dtype=torch.long, device=self.device)
        self._lamb_plan = list(by_group.values())
        self._lamb_n_tensors = n_tensors

    @torch.no_grad()
    def step(self, closure=None, grad_scaler=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        self.grad_sync()
        self._step += 1

        inv_scale = 1.0 / self._global_scale
        if grad_scaler is not None:
            inv_scale *= float(grad_scaler._get_scale_async().double().reciprocal())
        if inv_scale != 1.0:
            for b in self.buckets:
                b.grad_shard.mul_(inv_scale)

        if self.device.type == "cuda":
            self._step_fused(inv_scale)
        else:
            self._step_ref(inv_scale)

        for b in self.buckets:
            self._issue_bucket_param_sync(b)
        if self.world_size > 1 and not self.overlap_param_sync:
            for b in self.buckets:
                self._finish_param_sync_bucket(b)
            if self._use_stream:
                torch.cuda.current_stream().wait_stream(self._comm_stream)
        self._gnorm_sq.zero_()
        self._reset_buckets_after_step()
        return loss

    def _reduced_gnorm(self, inv_scale):
        """Device tensor: global grad norm (post-unscale)."""
        gn = self._gnorm_sq.clone()
        if self.clip_after_ar:
            if self.world_size > 1:
                dist.all_reduce(gn, group=self.process_group)
        else:
            # pre-AR partials are per-rank sums over LOCAL grads; scale to
            # the averaged-grad magnitude: ||1/W sum g_r||^2 ~ 1/W^2 sum
            # ||g_r||^2 (the reference's pre-AR clip is the same
            # approximation, fused into its reductions)
            if self.world_size > 1:
                dist.all_reduce(gn, group=self.process_group)
            gn /= float(self.world_size) ** 2
        return (gn * (inv_scale * inv_scale) if inv_scale != 1.0 else gn).sqrt()

    def _step_fused(self, inv_scale):
        amp_C = get_ext("amp_C")
        if self._lamb_plan is None:
            self._build_lamb_plan()
        # _gnorm_sq accumulated over SCALED grads as buckets synced; the
        # shards were unscaled above, so fold inv_scale into the norm here
        gnorm = self._reduced_gnorm(inv_scale)

        for ent in self._lamb_plan:
            group = ent["group"]
            beta1, beta2 = group["betas"]
            multi_tensor_applier(
                amp_C.multi_tensor_lamb_stage1, self._noop,
                [[b.grad_shard for b in ent["buckets"]],
                 [b.master_shard for b in ent["buckets"]],
                 [b.exp_avg for b in ent["buckets"]],
                 [b.exp_avg_sq for b in ent["buckets"]]],
                beta1, beta2, group["eps"], self._step,
                1 if group["bias_correction"] else 0, group["weight_decay"],
                1 if self.grad_averaging else 0, self.adam_w_mode,
                gnorm, self.max_grad_norm,
            )

        # per-tensor segment norms -> one cross-rank reduce -> trust ratios
        for ent in self._lamb_plan:
            full = torch.zeros(self._lamb_n_tensors, 2, device=self.device)
            if ent["seg_p"]:
                pn = multi_tensor_applier(amp_C.multi_tensor_l2norm, self._noop,
                                          [ent["seg_p"]], True)[1]
                un = multi_tensor_applier(amp_C.multi_tensor_l2norm, self._noop,
                                          [ent["seg_u"]], True)[1]
                full.index_put_((ent["idx_t"],),
                                torch.stack([pn.pow(2), un.pow(2)], dim=1), accumulate=True)
            # every rank MUST join the reduce even with an all-padding shard
            # (an empty seg list on one rank would desync the collective
            # sequence and hang the group)
            if self.world_size > 1:
                dist.all_reduce(full, group=self.process_group)
            if not ent["seg_p"]:
                continue
            norms = full.sqrt()
            seg_pn = norms.index_select(0, ent["idx_t"])[:, 0].contiguous()
            seg_un = norms.index_select(0, ent["idx_t"])[:, 1].contiguous()
            multi_tensor_applier(
                amp_C.multi_tensor_lamb_stage2, self._noop,
                [ent["seg_p"], ent["seg_u"]],
                seg_pn, seg_un, ent["group"]["lr"], ent["group"]["weight_decay"],
                self.use_nvlamb,
            )

    # ---- CPU reference path (gloo CI) ----
    def _step_ref(self, inv_scale):
        gnorm = float(self.grad_norm_from_shards())
        clip = gnorm / self.max_grad_norm if (self.max_grad_norm > 0 and gnorm > self.max_grad_norm) else 1.0
        for b in self.buckets:
            group = b.group
            beta1, beta2 = group["betas"]
            step = self._step
            bc1 = 1 - beta1 ** step if group["bias_correction"] else 1.0
            bc2 = 1 - beta2 ** step if group["bias_correction"] else 1.0
            beta3 = 1 - beta1 if self.grad_averaging else 1.0
            wd, eps, lr = group["weight_decay"], group["eps"], group["lr"]

            g = b.grad_shard.div_(clip)
            if self.adam_w_mode == 0 and wd != 0:
                g = g.add_(b.master_shard, alpha=wd)
            b.exp_avg.mul_(beta1).add_(g, alpha=beta3)
            b.exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
            update = (b.exp_avg / bc1) / ((b.exp_avg_sq / bc2).sqrt() + eps)
            if self.adam_w_mode == 1 and wd != 0:
                update = update + wd * b.master_shard
            b.grad_shard.copy_(update)

            lo = self.rank * b.shard_size
            hi = lo + b.shard_size
            partials = torch.zeros(len(b.params), 2, device=self.device)
            for i, (p, offset) in enumerate(b.params):
                s0, s1 = max(offset, lo), min(offset + p.numel(), hi)
                if s0 >= s1:
                    continue
                seg = slice(s0 - lo, s1 - lo)
                partials[i, 0] = b.master_shard[seg].pow(2).sum()
                partials[i, 1] = update[seg].pow(2).sum()
            if self.world_size > 1:
                dist.all_reduce(partials, group=self.process_group)
            norms = partials.sqrt()
            for i, (p, offset) in enumerate(b.params):
                s0, s1 = max(offset, lo), min(offset + p.numel(), hi)
                if s0 >= s1:
                    continue
                seg = slice(s0 - lo, s1 - lo)
                pn, un = float(norms[i, 0]), float(norms[i, 1])
                if (self.use_nvlamb or wd != 0) and pn != 0 and un != 0:
                    ratio = lr * pn / un
                else:
                    ratio = lr
                b.master_shard[seg].add_(update[seg], alpha=-ratio)

    def grad_norm_from_shards(self):
        local_sq = sum(float(b.grad_shard.pow(2).sum()) for b in self.buckets)
        t = torch.tensor([local_sq], device=self.device)
        if self.world_size > 1:
            dist.all_reduce(t, group=self.process_group)
        return t.sqrt().squeeze()
