"""FusedLAMB — multi-tensor fused LAMB for MI355X.

API parity with the reference ``apex.optimizers.FusedLAMB``
(apex/optimizers/fused_lamb.py:5-244). Step structure (reference :145-241):
per-dtype ``multi_tensor_l2norm`` → fused global grad norm → one
``multi_tensor_lamb`` launch per dtype group (stage-1 Adam-style update with
global-norm clipping, per-tensor trust-ratio apply in stage 2, both inside
the one kernel sequence).
"""

import torch

from .._ext import get_ext
from ..multi_tensor_apply import multi_tensor_applier
from ..tracing import traced


class FusedLAMB(torch.optim.Optimizer):
    def __init__(
        self,
        params,
        lr=1e-3,
        bias_correction=True,
        betas=(0.9, 0.999),
        eps=1e-6,
        weight_decay=0.01,
        amsgrad=False,
        adam_w_mode=True,
        grad_averaging=True,
        set_grad_none=True,
        max_grad_norm=1.0,
        use_nvlamb=False,
        capturable=False,
    ):
        if amsgrad:
            raise RuntimeError("FusedLAMB does not support the AMSGrad variant.")
        defaults = dict(
            lr=lr,
            bias_correction=bias_correction,
            betas=betas,
            eps=eps,
            weight_decay=weight_decay,
            grad_averaging=grad_averaging,
            max_grad_norm=max_grad_norm,
        )
        super().__init__(params, defaults)
        self.adam_w_mode = 1 if adam_w_mode else 0
        self.set_grad_none = set_grad_none
        self.use_nvlamb = use_nvlamb
        self._dummy_overflow_buf = None
        # capturable: lr/step live on device and bias corrections are
        # computed in-kernel (multi_tensor_lamb_capturable), so the whole
        # step records into a hipGraph and replays with an advancing step.
        # Note: graph REPLAYS advance only the device counter (_step_t);
        # read it (not group["step"]) for the true step count, and update
        # _lr_t in place for lr schedules under capture.
        self.capturable = capturable
        self._lr_t = None
        self._step_t = None

    def zero_grad(self, set_to_none: bool = True):
        if self.set_grad_none or set_to_none:
            for group in self.param_groups:
                for p in group["params"]:
                    p.grad = None
        else:
            super().zero_grad(set_to_none=False)

    def state_dict(self):
        if self.capturable and self._step_t is not None:
            # graph replays advance only the device counter; sync the host
            # per-group counts so checkpoints carry the true step
            true_step = int(self._step_t.item())
            for group in self.param_groups:
                group["step"] = true_step
        return super().state_dict()

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        if self.capturable and self._step_t is not None:
            self._step_t.fill_(int(self.param_groups[0].get("step", 0)))

    def _noop_buf(self, device):
        if self._dummy_overflow_buf is None or self._dummy_overflow_buf.device != device:
            self._dummy_overflow_buf = torch.zeros(1, dtype=torch.int32, device=device)
        return self._dummy_overflow_buf

    @torch.no_grad()
    @traced("FusedLAMB.step")
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        # Global grad norm over ALL groups (reference computes one per step).
        device = self.param_groups[0]["params"][0].device
        if (self.capturable and len(self.param_groups) > 1
                and device.type == "cuda"
                and torch.cuda.is_current_stream_capturing()
                and len({float(g["lr"]) for g in self.param_groups}) > 1):
            # one shared _lr_t cannot carry per-group lrs inside a graph
            # (eager steps refresh it per group, stream-ordered)
            raise RuntimeError(
                "capturable FusedLAMB: cannot capture with divergent "
                "per-group lrs; use one lr or capture per group")
        g_all_16, g_all_32 = [], []
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                if p.dtype in (torch.float16, torch.bfloat16):
                    g_all_16.append(p.grad)
                elif p.dtype == torch.float32:
                    g_all_32.append(p.grad)
                else:
                    raise RuntimeError("FusedLAMB only supports fp16/bf16/fp32")

        if device.type == "cuda":
            amp_C = get_ext("amp_C")
            noop = self._noop_buf(device)
            norms = []
            if g_all_16:
                norms.append(multi_tensor_applier(amp_C.multi_tensor_l2norm, noop, [g_all_16], False)[0])
            if g_all_32:
                norms.append(multi_tensor_applier(amp_C.multi_tensor_l2norm, noop, [g_all_32], False)[0])
            global_grad_norm = torch.norm(torch.stack([n.squeeze() for n in norms])) if norms else torch.zeros((), device=device)
        else:
            sq = sum(float(g.float().pow(2).sum()) for g in g_all_16 + g_all_32)
            global_grad_norm = sq ** 0.5

        for group in self.param_groups:
            bias_correction = 1 if group["bias_correction"] else 0
            beta1, beta2 = group["betas"]
            grad_averaging = 1 if group["grad_averaging"] else 0
            group["step"] = group.get("step", 0) + 1

            g16, p16, m16, v16 = [], [], [], []
            g32, p32, m32, v32 = [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                if p.dtype in (torch.float16, torch.bfloat16):
                    g16.append(p.grad)
                    p16.append(p)
                    m16.append(state["exp_avg"])
                    v16.append(state["exp_avg_sq"])
                else:
                    g32.append(p.grad)
                    p32.append(p)
                    m32.append(state["exp_avg"])
                    v32.append(state["exp_avg_sq"])

            if device.type == "cuda":
                amp_C = get_ext("amp_C")
                noop = self._noop_buf(device)
                if self.capturable:
                    if self._lr_t is None:
                        self._lr_t = torch.full((1,), float(group["lr"]),
                                                dtype=torch.float32, device=device)
                        # resume-aware: group["step"] was already advanced
                        # for THIS step above, so seed with step-1 and let
                        # the add_(1) below land on the true count
                        self._step_t = torch.full((1,), int(group["step"]) - 1,
                                                  dtype=torch.int32, device=device)
                    elif not torch.cuda.is_current_stream_capturing():
                        # eager steps track group["lr"] (lr schedules); the
                        # refresh is skipped DURING capture so replays read
                        # whatever the user writes into _lr_t on device
                        self._lr_t.fill_(float(group["lr"]))
                    if group is self.param_groups[0]:
                        self._step_t.add_(1)  # device op: advances per replay
                for lists in [[g16, p16, m16, v16], [g32, p32, m32, v32]]:
                    if not lists[0]:
                        continue
                    if self.capturable:
                        multi_tensor_applier(
                            amp_C.multi_tensor_lamb_capturable, noop, lists,
                            self._lr_t, beta1, beta2, group["eps"], self._step_t,
                            bias_correction, group["weight_decay"], grad_averaging,
                            self.adam_w_mode, global_grad_norm, group["max_grad_norm"],
                            self.use_nvlamb,
                        )
                    else:
                        multi_tensor_applier(
                            amp_C.multi_tensor_lamb, noop, lists,
                            group["lr"], beta1, beta2, group["eps"], group["step"],
                            bias_correction, group["weight_decay"], grad_averaging,
                            self.adam_w_mode, global_grad_norm, group["max_grad_norm"],
                            self.use_nvlamb,
                        )
            else:
                for lists in [[g16, p16, m16, v16], [g32, p32, m32, v32]]:
                    if lists[0]:
                        self._step_ref(group, bias_correction, beta1, beta2, grad_averaging,
                                       float(global_grad_norm), *lists)
        return loss

    def _step_ref(self, group, bias_correction, beta1, beta2, grad_averaging, global_grad_norm, g, p, m, v):
        step = group["step"]
        bc1 = 1.0 - beta1 ** step if bias_correction else 1.0
        bc2 = 1.0 - beta2 ** step if bias_correction else 1.0
        lr, wd, eps = group["lr"], group["weight_decay"], group["eps"]
        max_grad_norm = group["max_grad_norm"]
        clip = global_grad_norm / max_grad_norm if (max_grad_norm > 0 and global_grad_norm > max_grad_norm) else 1.0
        beta3 = 1.0 - beta1 if grad_averaging else 1.0
        for gi, pi, mi, vi in zip(g, p, m, v):
            gf = gi.float() / clip
            pf = pi.float()
            if self.adam_w_mode == 0 and wd != 0:
                gf = gf + wd * pf
            mi.mul_(beta1).add_(gf, alpha=beta3)
            vi.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
            update = (mi / bc1) / ((vi / bc2).sqrt() + eps)
            if self.adam_w_mode == 1 and wd != 0:
                update = update + wd * pf
            p_norm = pf.norm()
            u_norm = update.norm()
            if (self.use_nvlamb or wd != 0) and p_norm != 0 and u_norm != 0:
                ratio = lr * (p_norm / u_norm)
            else:
                ratio = lr
            pf = pf - ratio * update
            pi.copy_(pf.to(pi.dtype))
