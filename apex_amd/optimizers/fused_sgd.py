"""FusedSGD — multi-tensor fused SGD with momentum for MI355X.

API parity with the reference ``apex.optimizers.FusedSGD``
(apex/optimizers/fused_sgd.py:7-284), including the amp-O2 master-weight
contract: when ``amp.initialize(..., opt_level="O2")`` has installed an
``_amp_stash`` on this optimizer, ``step()`` pairs each fp16 model param with
its fp32 master and launches the 4-list kernel that updates the master and
writes the fp16 copy in one pass (``materialize_master_grads``).
"""

import torch

from .._ext import get_ext
from ..multi_tensor_apply import multi_tensor_applier


class FusedSGD(torch.optim.Optimizer):
    def __init__(
        self,
        params,
        lr=None,
        momentum=0.0,
        dampening=0.0,
        weight_decay=0.0,
        nesterov=False,
        wd_after_momentum=False,
        materialize_master_grads=True,
        set_grad_none=False,
    ):
        if lr is None:
            raise ValueError("lr is required for FusedSGD")
        if lr < 0.0:
            raise ValueError(f"Invalid learning rate: {lr}")
        if momentum < 0.0:
            raise ValueError(f"Invalid momentum value: {momentum}")
        if weight_decay < 0.0:
            raise ValueError(f"Invalid weight_decay value: {weight_decay}")
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("Nesterov momentum requires a momentum and zero dampening")

        defaults = dict(lr=lr, momentum=momentum, dampening=dampening, weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)

        self.wd_after_momentum = wd_after_momentum
        self.materialize_master_grads = materialize_master_grads
        self.set_grad_none = set_grad_none
        self.most_recent_scale = 1.0
        self.scale_set_by_backward = False
        self._dummy_overflow_buf = None

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault("nesterov", False)

    def zero_grad(self, set_to_none: bool = False):
        if self.set_grad_none or set_to_none:
            for group in self.param_groups:
                for p in group["params"]:
                    p.grad = None
        else:
            super().zero_grad(set_to_none=False)

    def get_momentums(self, params):
        momentums = []
        first_run = False
        for p in params:
            state = self.state[p]
            if "momentum_buffer" not in state:
                first_run = True
                state["momentum_buffer"] = torch.zeros_like(p)
            momentums.append(state["momentum_buffer"])
        return momentums, first_run

    def _noop_buf(self, device):
        if self._dummy_overflow_buf is None or self._dummy_overflow_buf.device != device:
            self._dummy_overflow_buf = torch.zeros(1, dtype=torch.int32, device=device)
        return self._dummy_overflow_buf

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        explicit_master_params = hasattr(self, "_amp_stash") and hasattr(self._amp_stash, "fp32_from_fp16_groups")

        for gid, group in enumerate(self.param_groups):
            weight_decay = group["weight_decay"]
            momentum = group["momentum"]
            dampening = group["dampening"]
            nesterov = group["nesterov"]
            launches = []  # (lists, num_lists)

            if explicit_master_params:
                stash = self._amp_stash
                fp32_params = [p for p in stash.fp32_from_fp32_groups[gid] if p.grad is not None]
                fp32_grads = [p.grad for p in stash.fp32_from_fp32_groups[gid] if p.grad is not None]
                fp32_momentums, first_run = self.get_momentums(fp32_params)
                if fp32_params:
                    launches.append(((fp32_grads, fp32_params, fp32_momentums), first_run, 3))

                if self.materialize_master_grads:
                    fp16_model_params = [
                        p for i, p in enumerate(stash.fp16_groups[gid])
                        if stash.fp32_from_fp16_groups[gid][i].grad is not None
                    ]
                    fp32_from_fp16_params = [p for p in stash.fp32_from_fp16_groups[gid] if p.grad is not None]
                    fp32_from_fp16_grads = [p.grad for p in stash.fp32_from_fp16_groups[gid] if p.grad is not None]
                    fp32_from_fp16_momentums, first_run = self.get_momentums(fp32_from_fp16_params)
                    if fp32_from_fp16_params:
                        launches.append(
                            ((fp32_from_fp16_grads, fp32_from_fp16_params, fp32_from_fp16_momentums, fp16_model_params), first_run, 4)
                        )
                else:
                    fp16_model_params = [p for p in stash.fp16_groups[gid] if p.grad is not None]
                    fp16_model_grads = [p.grad for p in stash.fp16_groups[gid] if p.grad is not None]
                    fp32_from_fp16_params = [
                        mp for p, mp in zip(stash.fp16_groups[gid], stash.fp32_from_fp16_groups[gid])
                        if p.grad is not None
                    ]
                    fp32_from_fp16_momentums, first_run = self.get_momentums(fp32_from_fp16_params)
                    if fp16_model_params:
                        launches.append(
                            ((fp16_model_grads, fp32_from_fp16_params, fp32_from_fp16_momentums, fp16_model_params), first_run, 4)
                        )
            else:
                params = [p for p in group["params"] if p.grad is not None]
                grads = [p.grad for p in params]
                momentums, first_run = self.get_momentums(params)
                if params:
                    launches.append(((grads, params, momentums), first_run, 3))

            for lists, first_run, num_lists in launches:
                device = lists[1][0].device
                if device.type == "cuda":
                    amp_C = get_ext("amp_C")
                    multi_tensor_applier(
                        amp_C.multi_tensor_sgd,
                        self._noop_buf(device),
                        list(lists),
                        weight_decay,
                        momentum,
                        dampening,
                        group["lr"],
                        nesterov,
                        first_run,
                        self.wd_after_momentum,
                        1.0 / self.most_recent_scale,
                    )
                else:
                    self._step_ref(lists, num_lists, weight_decay, momentum, dampening, group["lr"], nesterov,
                                   first_run, 1.0 / self.most_recent_scale)

        self.most_recent_scale = 1.0
        self.scale_set_by_backward = False
        return loss

    def _step_ref(self, lists, num_lists, wd, momentum, dampening, lr, nesterov, first_run, scale):
        grads, params, momentums = lists[0], lists[1], lists[2]
        out_copy = lists[3] if num_lists == 4 else None
        for i, (g, p, m) in enumerate(zip(grads, params, momentums)):
            gf = g.float() * scale
            pf = p.float()
            if wd != 0 and not self.wd_after_momentum:
                gf = gf + wd * pf
            if momentum != 0:
                if first_run:
                    m.copy_(gf.to(m.dtype))
                else:
                    m.mul_(momentum).add_(gf, alpha=1 - dampening)
                gf = gf + momentum * m.float() if nesterov else m.float().clone()
            if wd != 0 and self.wd_after_momentum:
                gf = gf + wd * pf
            pf = pf - lr * gf
            p.copy_(pf.to(p.dtype))
            if out_copy is not None:
                out_copy[i].copy_(pf.to(out_copy[i].dtype))
