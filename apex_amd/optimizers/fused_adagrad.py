"""FusedAdagrad — multi-tensor fused Adagrad for MI355X.

API parity with the reference ``apex.optimizers.FusedAdagrad``
(apex/optimizers/fused_adagrad.py:5-131): ``adagrad_w_mode`` selects
decoupled weight decay; tensors grouped fp16/bf16 vs fp32.
"""

import torch

from .._ext import get_ext
from ..multi_tensor_apply import multi_tensor_applier


class FusedAdagrad(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-2, eps=1e-10, weight_decay=0.0, set_grad_none=True, adagrad_w_mode=False):
        defaults = dict(lr=lr, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.adagrad_w_mode = 1 if adagrad_w_mode else 0
        self.set_grad_none = set_grad_none
        self._dummy_overflow_buf = None

    def zero_grad(self, set_to_none: bool = True):
        if self.set_grad_none or set_to_none:
            for group in self.param_groups:
                for p in group["params"]:
                    p.grad = None
        else:
            super().zero_grad(set_to_none=False)

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

        for group in self.param_groups:
            g16, p16, h16 = [], [], []
            g32, p32, h32 = [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                if p.grad.is_sparse:
                    raise RuntimeError("FusedAdagrad does not support sparse gradients")
                state = self.state[p]
                if len(state) == 0:
                    state["sum"] = torch.zeros_like(p, dtype=torch.float32)
                if p.dtype in (torch.float16, torch.bfloat16):
                    g16.append(p.grad)
                    p16.append(p)
                    h16.append(state["sum"])
                elif p.dtype == torch.float32:
                    g32.append(p.grad)
                    p32.append(p)
                    h32.append(state["sum"])
                else:
                    raise RuntimeError("FusedAdagrad only supports fp16/bf16/fp32")

            device = group["params"][0].device
            for lists in [[g16, p16, h16], [g32, p32, h32]]:
                if not lists[0]:
                    continue
                if device.type == "cuda":
                    amp_C = get_ext("amp_C")
                    multi_tensor_applier(
                        amp_C.multi_tensor_adagrad, self._noop_buf(device), lists,
                        group["lr"], group["eps"], self.adagrad_w_mode, group["weight_decay"],
                    )
                else:
                    self._step_ref(group, *lists)
        return loss

    def _step_ref(self, group, g, p, h):
        lr, eps, wd = group["lr"], group["eps"], group["weight_decay"]
        for gi, pi, hi in zip(g, p, h):
            gf = gi.float()
            pf = pi.float()
            if self.adagrad_w_mode == 0 and wd != 0:
                gf = gf + wd * pf
            hi.add_(gf * gf)
            update = gf / (hi.sqrt() + eps)
            if self.adagrad_w_mode == 1 and wd != 0:
                update = update + wd * pf
            pf = pf - lr * update
            pi.copy_(pf.to(pi.dtype))
