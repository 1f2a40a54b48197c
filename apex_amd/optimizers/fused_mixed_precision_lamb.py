"""FusedMixedPrecisionLamb — graph-safe LAMB with low-precision param copies.

API parity with the reference ``apex.optimizers.FusedMixedPrecisionLamb``
(apex/optimizers/fused_mixed_precision_lamb.py:9-291): ``lr``/``step`` are
device tensors (sync-free, hipGraph-safe), fp32 masters are the optimizer's
params, and ``reduced_precision_dtype`` maintains bf16/fp16 model copies
updated inside the fused kernel. Integrates with ``torch.amp.GradScaler``
(``_step_supports_amp_scaling``).
"""

import torch

from .._ext import get_ext
from ..multi_tensor_apply import multi_tensor_applier


class FusedMixedPrecisionLamb(torch.optim.Optimizer):
    def __init__(
        self,
        params,
        lr=1e-3,
        step=0,
        bias_correction=True,
        betas=(0.9, 0.999),
        eps=1e-6,
        weight_decay=0.01,
        amsgrad=False,
        grad_averaging=True,
        max_grad_norm=1.0,
        use_nvlamb=False,
        reduced_precision_dtype=None,
    ):
        if amsgrad:
            raise RuntimeError("FusedMixedPrecisionLamb does not support the AMSGrad variant.")

        defaults = dict(
            lr=torch.tensor(lr, dtype=torch.float32),
            step=torch.tensor([step], dtype=torch.int32),
            bias_correction=bias_correction,
            betas=betas,
            eps=eps,
            weight_decay=weight_decay,
            grad_averaging=grad_averaging,
            max_grad_norm=max_grad_norm,
        )
        tensor_state = ["lr", "step"]
        super().__init__(params, defaults)

        # Move tensor-state to the params' device.
        device = self.param_groups[0]["params"][0].device
        for idx, group in enumerate(self.param_groups):
            for item in tensor_state:
                self.param_groups[idx][item] = group[item].to(device=device)

        # Build fp32 masters; keep model (possibly reduced-precision) params.
        self.reduced_precision_dtype = reduced_precision_dtype
        self.param_groups_full_precision = []
        for group in self.param_groups:
            full = []
            for p in group["params"]:
                if reduced_precision_dtype is not None and p.dtype == reduced_precision_dtype:
                    full.append(p.detach().clone().float())
                else:
                    full.append(None)  # param already fp32 — updated in place
            self.param_groups_full_precision.append({"params": full})

        self.use_nvlamb = use_nvlamb
        self._step_supports_amp_scaling = True
        self._dummy_overflow_buf = None

    def _noop_buf(self, device):
        if self._dummy_overflow_buf is None or self._dummy_overflow_buf.device != device:
            self._dummy_overflow_buf = torch.zeros(1, dtype=torch.int32, device=device)
        return self._dummy_overflow_buf

    @torch.no_grad()
    def step(self, closure=None, grad_scaler=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        device = self.param_groups[0]["params"][0].device

        # Scale handling (device tensors; graph-safe).
        if grad_scaler is not None:
            found_inf = grad_scaler._check_inf_per_device(self)[device]
            scale = grad_scaler._get_scale_async()
            inv_scale = scale.double().reciprocal().float()
        else:
            found_inf = torch.zeros(1, dtype=torch.float32, device=device)
            inv_scale = torch.ones(1, dtype=torch.float32, device=device)

        # Collect all grads for global norm.
        all_grads = []
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is not None:
                    all_grads.append(p.grad)

        if device.type == "cuda":
            amp_C = get_ext("amp_C")
            noop = self._noop_buf(device)
            g_norm = multi_tensor_applier(amp_C.multi_tensor_l2norm_mp, noop, [all_grads], False)[0]
        else:
            sq = sum(float(g.float().pow(2).sum()) for g in all_grads)
            g_norm = torch.tensor(sq ** 0.5, device=device)

        for gi, group in enumerate(self.param_groups):
            bias_correction = 1 if group["bias_correction"] else 0
            beta1, beta2 = group["betas"]
            grad_averaging = 1 if group["grad_averaging"] else 0
            group["step"] += 1

            g_list, p_list, m_list, v_list, full_list = [], [], [], [], []
            for pi, p in enumerate(group["params"]):
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                g_list.append(p.grad)
                p_list.append(p)
                m_list.append(state["exp_avg"])
                v_list.append(state["exp_avg_sq"])
                full_list.append(self.param_groups_full_precision[gi]["params"][pi])

            if not g_list:
                continue

            if device.type == "cuda":
                amp_C = get_ext("amp_C")
                noop = self._noop_buf(device)
                # Split into reduced-precision (with master) and fp32 sets.
                rp = [(g, p, m, v, f) for g, p, m, v, f in zip(g_list, p_list, m_list, v_list, full_list) if f is not None]
                fp = [(g, p, m, v) for g, p, m, v, f in zip(g_list, p_list, m_list, v_list, full_list) if f is None]
                if rp:
                    gs, ps, ms, vs, fs = (list(t) for t in zip(*rp))
                    multi_tensor_applier(
                        amp_C.multi_tensor_lamb_mp, noop, [gs, fs, ms, vs, ps],
                        group["lr"], beta1, beta2, group["eps"], group["step"],
                        bias_correction, group["weight_decay"], grad_averaging, 1,
                        g_norm, group["max_grad_norm"], self.use_nvlamb, found_inf, inv_scale,
                    )
                if fp:
                    gs, ps, ms, vs = (list(t) for t in zip(*fp))
                    multi_tensor_applier(
                        amp_C.multi_tensor_lamb_mp, noop, [gs, ps, ms, vs, ps],
                        group["lr"], beta1, beta2, group["eps"], group["step"],
                        bias_correction, group["weight_decay"], grad_averaging, 1,
                        g_norm, group["max_grad_norm"], self.use_nvlamb, found_inf, inv_scale,
                    )
            else:
                self._step_ref(group, bias_correction, beta1, beta2, grad_averaging,
                               float(g_norm), float(inv_scale), g_list, p_list, m_list, v_list, full_list)
        return loss

    def _step_ref(self, group, bias_correction, beta1, beta2, grad_averaging,
                  global_grad_norm, inv_scale, g, p, m, v, full):
        step = int(group["step"].item()) if torch.is_tensor(group["step"]) else group["step"]
        bc1 = 1.0 - beta1 ** step if bias_correction else 1.0
        bc2 = 1.0 - beta2 ** step if bias_correction else 1.0
        lr = float(group["lr"].item()) if torch.is_tensor(group["lr"]) else group["lr"]
        wd, eps = group["weight_decay"], group["eps"]
        max_grad_norm = group["max_grad_norm"]
        global_grad_norm = global_grad_norm * inv_scale
        clip = global_grad_norm / max_grad_norm if (max_grad_norm > 0 and global_grad_norm > max_grad_norm) else 1.0
        beta3 = 1.0 - beta1 if grad_averaging else 1.0
        for gi, pi, mi, vi, fi in zip(g, p, m, v, full):
            master = fi if fi is not None else pi
            gf = gi.float() * inv_scale / clip
            pf = master.float()
            mi.mul_(beta1).add_(gf, alpha=beta3)
            vi.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
            update = (mi / bc1) / ((vi / bc2).sqrt() + eps)
            if wd != 0:
                update = update + wd * pf
            p_norm = pf.norm()
            u_norm = update.norm()
            if (self.use_nvlamb or wd != 0) and p_norm != 0 and u_norm != 0:
                ratio = lr * (p_norm / u_norm)
            else:
                ratio = lr
            pf = pf - ratio * update
            master.copy_(pf)
            if fi is not None:
                pi.copy_(pf.to(pi.dtype))
