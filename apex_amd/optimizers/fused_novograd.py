"""FusedNovoGrad — multi-tensor fused NovoGrad for MI355X.

API parity with the reference ``apex.optimizers.FusedNovoGrad``
(apex/optimizers/fused_novograd.py:5-255). NovoGrad keeps a per-TENSOR
second moment: ``exp_avg_sq`` is stored as two flat group-level vectors
(fp16-set / fp32-set) updated from per-tensor grad norms each step; the
multi-tensor kernel consumes the norm vector directly.
"""

import torch

from .._ext import get_ext
from ..multi_tensor_apply import multi_tensor_applier


class FusedNovoGrad(torch.optim.Optimizer):
    def __init__(
        self,
        params,
        lr=1e-3,
        bias_correction=True,
        betas=(0.95, 0.98),
        eps=1e-8,
        weight_decay=0.0,
        amsgrad=False,
        reg_inside_moment=False,
        grad_averaging=True,
        norm_type=2,
        init_zero=False,
        set_grad_none=True,
    ):
        if amsgrad:
            raise RuntimeError("FusedNovoGrad does not support the AMSGrad variant.")
        if norm_type not in (0, 2):
            raise RuntimeError("FusedNovoGrad only supports l2/inf norm (norm_type 2 or 0)")
        defaults = dict(
            lr=lr,
            bias_correction=bias_correction,
            betas=betas,
            eps=eps,
            weight_decay=weight_decay,
            grad_averaging=grad_averaging,
            norm_type=norm_type,
            init_zero=init_zero,
        )
        super().__init__(params, defaults)
        # moment_mode 0: wd outside the moment (L2-style); 1: wd inside.
        self.moment_mode = 0 if reg_inside_moment else 1
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

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        # Relocate group-level norm vectors to the param device (reference :149-155).
        for group in self.param_groups:
            if "exp_avg_sq" in group and group["params"]:
                device = group["params"][0].device
                group["exp_avg_sq"] = [t.to(device) for t in group["exp_avg_sq"]]

    def _per_tensor_norms(self, grads, norm_type, device):
        if device.type == "cuda":
            amp_C = get_ext("amp_C")
            noop = self._noop_buf(device)
            # returns (global_norm, per_tensor_norms)
            _, per_tensor = multi_tensor_applier(amp_C.multi_tensor_l2norm, noop, [grads], True)
            if norm_type == 0:
                per_tensor = torch.stack([g.float().abs().max() for g in grads])
            return per_tensor.flatten()
        if norm_type == 2:
            return torch.stack([g.float().norm() for g in grads])
        return torch.stack([g.float().abs().max() for g in grads])

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            bias_correction = 1 if group["bias_correction"] else 0
            beta1, beta2 = group["betas"]
            grad_averaging = 1 if group["grad_averaging"] else 0
            group["step"] = group.get("step", 0) + 1

            g16, p16, m16 = [], [], []
            g32, p32, m32 = [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                if p.dtype in (torch.float16, torch.bfloat16):
                    g16.append(p.grad)
                    p16.append(p)
                    m16.append(state["exp_avg"])
                elif p.dtype == torch.float32:
                    g32.append(p.grad)
                    p32.append(p)
                    m32.append(state["exp_avg"])
                else:
                    raise RuntimeError("FusedNovoGrad only supports fp16/bf16/fp32")

            device = group["params"][0].device

            # Group-level per-tensor second-moment vectors.
            if "exp_avg_sq" not in group:
                group["exp_avg_sq"] = [None, None]
                if group["init_zero"]:
                    group["exp_avg_sq"][0] = torch.zeros(len(g16), dtype=torch.float32, device=device)
                    group["exp_avg_sq"][1] = torch.zeros(len(g32), dtype=torch.float32, device=device)
                else:
                    group["exp_avg_sq"][0] = (
                        self._per_tensor_norms(g16, group["norm_type"], device) ** 2 if g16
                        else torch.zeros(0, dtype=torch.float32, device=device)
                    )
                    group["exp_avg_sq"][1] = (
                        self._per_tensor_norms(g32, group["norm_type"], device) ** 2 if g32
                        else torch.zeros(0, dtype=torch.float32, device=device)
                    )
            else:
                if g16:
                    n16 = self._per_tensor_norms(g16, group["norm_type"], device)
                    group["exp_avg_sq"][0].mul_(beta2).add_(n16 ** 2, alpha=1 - beta2)
                if g32:
                    n32 = self._per_tensor_norms(g32, group["norm_type"], device)
                    group["exp_avg_sq"][1].mul_(beta2).add_(n32 ** 2, alpha=1 - beta2)

            for lists, v in [((g16, p16, m16), group["exp_avg_sq"][0]), ((g32, p32, m32), group["exp_avg_sq"][1])]:
                if not lists[0]:
                    continue
                if device.type == "cuda":
                    amp_C = get_ext("amp_C")
                    multi_tensor_applier(
                        amp_C.multi_tensor_novograd, self._noop_buf(device), list(lists), v,
                        group["lr"], beta1, beta2, group["eps"], group["step"],
                        bias_correction, group["weight_decay"], grad_averaging,
                        self.moment_mode, group["norm_type"],
                    )
                else:
                    self._step_ref(group, bias_correction, beta1, beta2, grad_averaging, v, *lists)
        return loss

    def _step_ref(self, group, bias_correction, beta1, beta2, grad_averaging, v, g, p, m):
        step = group["step"]
        bc1 = 1.0 - beta1 ** step if bias_correction else 1.0
        bc2 = 1.0 - beta2 ** step if bias_correction else 1.0
        lr, wd, eps = group["lr"], group["weight_decay"], group["eps"]
        beta3 = 1.0 - beta1 if grad_averaging else 1.0
        for i, (gi, pi, mi) in enumerate(zip(g, p, m)):
            gf = gi.float()
            pf = pi.float()
            denom = (v[i] / bc2).sqrt() + eps
            gf = gf / denom
            if wd != 0 and self.moment_mode == 1:
                gf = gf + wd * pf
            mi.mul_(beta1).add_(gf, alpha=beta3)
            update = mi / bc1
            if wd != 0 and self.moment_mode == 0:
                update = update + wd * pf
            pf = pf - lr * update
            pi.copy_(pf.to(pi.dtype))
