"""FusedAdam — multi-tensor fused Adam/AdamW for MI355X.

API parity with the reference ``apex.optimizers.FusedAdam``
(apex/optimizers/fused_adam.py:5-355): dtype-grouped tensor lists, fp32
``exp_avg``/``exp_avg_sq`` state, optional *capturable* mode (device-tensor
``lr``/``step``, GradScaler ``found_inf``/``inv_scale`` read on-device so the
step is hipGraph-capturable) and ``master_weights``.

Device path: one HIP multi-tensor kernel per dtype group
(csrc/multi_tensor_adam.hip) through the binary-search launcher — typically
2-3 launches for an entire network. CPU path: reference torch math (used by
the no-GPU CI and as the numerics oracle).
"""


import torch

from .._ext import get_ext
from ..multi_tensor_apply import multi_tensor_applier
from ..tracing import traced


class FusedAdam(torch.optim.Optimizer):
    """Implements Adam/AdamW with a single fused multi-tensor HIP kernel.

    Arguments mirror the reference FusedAdam:

    - adam_w_mode (bool): True → decoupled weight decay (AdamW), False → L2.
    - capturable (bool): keep ``step``/``lr`` as device tensors and read
      GradScaler state on-device so ``step()`` can be captured in a hipGraph.
    - master_weights (bool): fp32 master copies updated alongside low-precision
      params (requires capturable, as in the reference :84-87).
    """

    def __init__(
        self,
        params,
        lr=1e-3,
        bias_correction=True,
        betas=(0.9, 0.999),
        eps=1e-8,
        adam_w_mode=True,
        weight_decay=0.0,
        amsgrad=False,
        set_grad_none=True,
        capturable=False,
        master_weights=False,
    ):
        if amsgrad:
            raise RuntimeError("FusedAdam does not support the AMSGrad variant.")
        if master_weights and not capturable:
            raise RuntimeError("master_weights requires capturable=True")

        # If capturable, LR must live on device so hipGraph replay sees updates.
        if capturable:
            lr = torch.tensor(lr, dtype=torch.float32)
            if torch.cuda.is_available():
                lr = lr.cuda()

        defaults = dict(lr=lr, bias_correction=bias_correction, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.adam_w_mode = 1 if adam_w_mode else 0
        self.set_grad_none = set_grad_none
        self.capturable = capturable
        self.master_weights = master_weights

        # GradScaler integration (torch.amp) — scaler calls into us with
        # per-device scale/found_inf tensors.
        self._step_supports_amp_scaling = capturable

        if master_weights:
            self.param_groups_master = []
            for group in self.param_groups:
                self.param_groups_master.append(
                    {"params": [p.detach().clone().float() if p.requires_grad else None for p in group["params"]]}
                )

        self._dummy_overflow_buf = None

    def zero_grad(self, set_to_none: bool = True):
        if self.set_grad_none or set_to_none:
            for group in self.param_groups:
                for p in group["params"]:
                    p.grad = None
        else:
            super().zero_grad(set_to_none=False)

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        if not self.capturable:
            return
        # capturable kernels read lr/step from DEVICE pointers; a checkpoint
        # loaded without map_location leaves the group tensors on CPU (and a
        # CPU lr inside a later capture records a pageable H2D that faults
        # on replay) — pin them to the params' device here
        for group in self.param_groups:
            if not group["params"]:
                continue
            device = group["params"][0].device
            for key in ("lr", "step"):
                v = group.get(key)
                if torch.is_tensor(v) and v.device != device:
                    group[key] = v.to(device)

    def _noop_buf(self, device):
        if self._dummy_overflow_buf is None or self._dummy_overflow_buf.device != device:
            self._dummy_overflow_buf = torch.zeros(1, dtype=torch.int32, device=device)
        return self._dummy_overflow_buf

    @torch.no_grad()
    @traced("FusedAdam.step")
    def step(self, closure=None, grads=None, output_params=None, scale=None, grad_norms=None, grad_scaler=None):
        if any(p is not None for p in [grads, output_params, scale, grad_norms]):
            raise RuntimeError(
                "FusedAdam has been updated: 'grads'/'output_params'/'scale'/'grad_norms' "
                "are no longer supported (matches reference behavior)."
            )
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for gi, group in enumerate(self.param_groups):
            if len(group["params"]) == 0:
                continue
            device = group["params"][0].device
            bias_correction = 1 if group["bias_correction"] else 0
            beta1, beta2 = group["betas"]

            # Host-side step counter (single int per group, as in reference).
            if "step" in group:
                if self.capturable:
                    group["step"].add_(1)  # in-place: hipGraph replay must see it
                else:
                    group["step"] += 1
            else:
                group["step"] = (
                    torch.zeros(1, dtype=torch.int32, device=device) + 1 if self.capturable else 1
                )

            # Group tensors by dtype.
            g16, p16, m16, v16 = [], [], [], []
            gbf, pbf, mbf, vbf = [], [], [], []
            g32, p32, m32, v32 = [], [], [], []
            p16_master, pbf_master = [], []

            master_group = self.param_groups_master[gi]["params"] if self.master_weights else None

            for pi, p in enumerate(group["params"]):
                if p.grad is None:
                    continue
                if p.grad.is_sparse:
                    raise RuntimeError("FusedAdam does not support sparse gradients")
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                if p.dtype == torch.float16:
                    g16.append(p.grad)
                    p16.append(p)
                    m16.append(state["exp_avg"])
                    v16.append(state["exp_avg_sq"])
                    if master_group is not None:
                        p16_master.append(master_group[pi])
                elif p.dtype == torch.bfloat16:
                    gbf.append(p.grad)
                    pbf.append(p)
                    mbf.append(state["exp_avg"])
                    vbf.append(state["exp_avg_sq"])
                    if master_group is not None:
                        pbf_master.append(master_group[pi])
                elif p.dtype == torch.float32:
                    g32.append(p.grad)
                    p32.append(p)
                    m32.append(state["exp_avg"])
                    v32.append(state["exp_avg_sq"])
                else:
                    raise RuntimeError("FusedAdam only supports fp16/bf16/fp32 params")

            if device.type == "cuda":
                self._step_cuda(
                    group, bias_correction, beta1, beta2, grad_scaler,
                    [(g16, p16, m16, v16, p16_master), (gbf, pbf, mbf, vbf, pbf_master), (g32, p32, m32, v32, None)],
                )
            else:
                for lists in [(g16, p16, m16, v16), (gbf, pbf, mbf, vbf), (g32, p32, m32, v32)]:
                    self._step_ref(group, bias_correction, beta1, beta2, *lists)

        return loss

    # ----- device path -----
    def _step_cuda(self, group, bias_correction, beta1, beta2, grad_scaler, dtype_lists):
        amp_C = get_ext("amp_C")
        device = group["params"][0].device
        noop = self._noop_buf(device)

        if self.capturable:
            if grad_scaler is not None:
                scale = grad_scaler._get_scale_async()
                found_inf = grad_scaler._check_inf_per_device(self)[device]
                noop = found_inf.to(torch.int32) if found_inf.dtype != torch.int32 else found_inf
                inv_scale = scale.double().reciprocal().float()
            else:
                inv_scale = torch.ones(1, dtype=torch.float32, device=device)
            lr = group["lr"].to(device) if group["lr"].device != device else group["lr"]
            step_t = group["step"]
            for lists in dtype_lists:
                g, p, m, v, p_master = lists
                if not g:
                    continue
                if self.master_weights and p_master:
                    multi_tensor_applier(
                        amp_C.multi_tensor_adam_capturable_master, noop,
                        [g, p, m, v, p_master],
                        lr, beta1, beta2, group["eps"], step_t, self.adam_w_mode,
                        bias_correction, group["weight_decay"], inv_scale,
                    )
                else:
                    multi_tensor_applier(
                        amp_C.multi_tensor_adam_capturable, noop,
                        [g, p, m, v],
                        lr, beta1, beta2, group["eps"], step_t, self.adam_w_mode,
                        bias_correction, group["weight_decay"], inv_scale,
                    )
        else:
            for lists in dtype_lists:
                g, p, m, v, _ = lists
                if not g:
                    continue
                multi_tensor_applier(
                    amp_C.multi_tensor_adam, noop,
                    [g, p, m, v],
                    group["lr"], beta1, beta2, group["eps"], group["step"],
                    self.adam_w_mode, bias_correction, group["weight_decay"],
                )

    # ----- reference path (CPU CI + numerics oracle) -----
    def _step_ref(self, group, bias_correction, beta1, beta2, g, p, m, v):
        if not g:
            return
        step = group["step"] if isinstance(group["step"], int) else int(group["step"].item())
        bc1 = 1.0 - beta1 ** step if bias_correction else 1.0
        bc2 = 1.0 - beta2 ** step if bias_correction else 1.0
        lr = group["lr"] if not torch.is_tensor(group["lr"]) else group["lr"].item()
        wd = group["weight_decay"]
        eps = group["eps"]
        for gi, pi, mi, vi in zip(g, p, m, v):
            gf = gi.float()
            pf = pi.float()
            if self.adam_w_mode == 0 and wd != 0:
                gf = gf + wd * pf
            mi.mul_(beta1).add_(gf, alpha=1 - beta1)
            vi.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
            denom = (vi / bc2).sqrt_().add_(eps)
            update = (mi / bc1) / denom
            if self.adam_w_mode == 1 and wd != 0:
                update = update + wd * pf
            pf = pf - lr * update
            pi.copy_(pf.to(pi.dtype))
