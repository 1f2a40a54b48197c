"""Loss scaler with on-device dynamic scale update.

Reconstructs the removed apex amp loss scaler, with the dynamic-scale update
running entirely on-device via the ``update_scale_hysteresis`` HIP kernel
(reference kernel: csrc/update_scale_hysteresis.cu:5-41 — single-workitem
kernel: backoff on inf only after hysteresis is exhausted, growth every
``growth_interval`` clean steps, never grows to inf).

The per-iteration overflow *decision* (skip optimizer.step or not) is a host
decision, so one device→host read of the overflow flag per ``scale_loss``
exit is inherent — same as the reference's dynamic scaler.
"""

import torch

from .._ext import get_ext
from ..multi_tensor_apply import multi_tensor_applier


class LossScaler:
    def __init__(
        self,
        loss_scale,
        init_scale=2.0 ** 16,
        scale_factor=2.0,
        scale_window=2000,
        min_loss_scale=None,
        max_loss_scale=2.0 ** 24,
        hysteresis=1,
    ):
        self.dynamic = loss_scale == "dynamic"
        self._loss_scale = min(max_loss_scale, init_scale) if self.dynamic else float(loss_scale)
        self._scale_factor = scale_factor
        self._scale_window = scale_window
        self._min_loss_scale = min_loss_scale if min_loss_scale is not None else 1.0
        self._max_loss_scale = max_loss_scale
        self._hysteresis = hysteresis
        self._unskipped = 0
        self._has_overflow = False
        # device-side state (lazily created)
        self._scale_t = None
        self._growth_tracker_t = None
        self._hysteresis_t = None
        self._overflow_buf = None

    # ----- public -----
    def loss_scale(self):
        return self._loss_scale

    def update_scale(self):
        """Host-side dynamic update (CPU path / after host overflow check)."""
        if not self.dynamic:
            return
        if self._has_overflow:
            self._hysteresis_left = getattr(self, "_hysteresis_left", self._hysteresis) - 1
            if self._hysteresis_left <= 0:
                self._loss_scale = max(self._min_loss_scale, self._loss_scale / self._scale_factor)
                self._hysteresis_left = self._hysteresis
            self._unskipped = 0
        else:
            self._unskipped += 1
            self._hysteresis_left = self._hysteresis
            if self._unskipped == self._scale_window:
                self._loss_scale = min(self._max_loss_scale, self._loss_scale * self._scale_factor)
                self._unskipped = 0

    def _ensure_device_state(self, device):
        if self._scale_t is None or self._scale_t.device != device:
            self._scale_t = torch.tensor([self._loss_scale], dtype=torch.float32, device=device)
            self._growth_tracker_t = torch.tensor([self._unskipped], dtype=torch.int32, device=device)
            self._hysteresis_t = torch.tensor([self._hysteresis], dtype=torch.int32, device=device)
            self._overflow_buf = torch.zeros(1, dtype=torch.int32, device=device)

    # When several grad sets are unscaled in one optimizer iteration
    # (multiple optimizers, O2 masters + fp32 group), the dynamic-scale state
    # must tick exactly ONCE per iteration (round-1 advisor finding).
    # scale_loss brackets the calls with begin_unscale()/finish_unscale():
    # between them unscale_grads only accumulates into the shared overflow
    # flag; finish_unscale runs the hysteresis update once and does the one
    # device→host read. A bare unscale_grads call (no bracket) keeps the old
    # self-contained behavior.
    _in_iteration = False

    def begin_unscale(self):
        self._in_iteration = True
        self._iter_overflow_cpu = False
        if self._overflow_buf is not None:
            self._overflow_buf.zero_()

    def finish_unscale(self):
        """Tick the dynamic scale once and return the iteration's overflow."""
        self._in_iteration = False
        if self._overflow_buf is not None and self._overflow_buf.is_cuda:
            if self.dynamic:
                amp_C = get_ext("amp_C")
                amp_C.update_scale_hysteresis(
                    self._scale_t, self._growth_tracker_t, self._hysteresis_t,
                    self._overflow_buf, self._scale_factor,
                    1.0 / self._scale_factor, self._scale_window, self._hysteresis,
                )
            self._has_overflow = bool(self._overflow_buf.item()) or self._iter_overflow_cpu
            if self.dynamic:
                self._loss_scale = float(self._scale_t.item())
                self._loss_scale = min(self._max_loss_scale, max(self._min_loss_scale, self._loss_scale))
                self._scale_t.fill_(self._loss_scale)
        else:
            self._has_overflow = self._iter_overflow_cpu
            if self.dynamic:
                self.update_scale()
        return self._has_overflow

    def unscale_grads(self, grads_in, grads_out, scale_override=None, check=True):
        """out = in * (1/scale), with isfinite check setting the overflow flag.

        Standalone call: returns True if an overflow was detected
        (host-synchronizing on GPU) and ticks the dynamic scale iff
        ``scale_override`` is None. Inside a begin_unscale()/finish_unscale()
        bracket: only accumulates the overflow flag; no tick, no host sync.
        ``check=False`` (static scale): skip the device-to-host overflow read
        entirely — the launch stays async/hipGraph-capturable.
        """
        bracketed = self._in_iteration
        update_state = (scale_override is None) and not bracketed
        scale = self._loss_scale if scale_override is None else scale_override
        if len(grads_in) == 0:
            if not bracketed:
                self._has_overflow = False
            return False
        device = grads_in[0].device
        if device.type == "cuda":
            amp_C = get_ext("amp_C")
            self._ensure_device_state(device)
            if not bracketed:
                self._overflow_buf.zero_()
            multi_tensor_applier(
                amp_C.multi_tensor_scale,
                self._overflow_buf,
                [grads_in, grads_out],
                1.0 / scale,
            )
            if bracketed:
                return False  # decision deferred to finish_unscale
            if not check:
                self._has_overflow = False
                return False
            if self.dynamic and update_state:
                # on-device scale update; host reads only the skip decision
                amp_C.update_scale_hysteresis(
                    self._scale_t,
                    self._growth_tracker_t,
                    self._hysteresis_t,
                    self._overflow_buf,
                    self._scale_factor,
                    1.0 / self._scale_factor,
                    self._scale_window,
                    self._hysteresis,
                )
            self._has_overflow = bool(self._overflow_buf.item())
            if self.dynamic and update_state:
                self._loss_scale = float(self._scale_t.item())
                self._loss_scale = min(self._max_loss_scale, max(self._min_loss_scale, self._loss_scale))
                self._scale_t.fill_(self._loss_scale)
        else:
            inv = 1.0 / scale
            overflow = False
            for gi, go in zip(grads_in, grads_out):
                gf = gi.float() * inv
                if not torch.isfinite(gf).all():
                    overflow = True
                go.copy_(gf.to(go.dtype))
            if bracketed:
                self._iter_overflow_cpu = self._iter_overflow_cpu or overflow
                return False
            self._has_overflow = overflow
            if update_state:
                self.update_scale()
        return self._has_overflow

    def has_overflow(self):
        return self._has_overflow

    def state_dict(self):
        return {
            "loss_scale": self._loss_scale,
            "unskipped": self._unskipped,
            "dynamic": self.dynamic,
        }

    def load_state_dict(self, sd):
        self._loss_scale = sd["loss_scale"]
        self._unskipped = sd.get("unskipped", 0)
        self.dynamic = sd.get("dynamic", self.dynamic)
        self._scale_t = None  # recreate lazily on the right device
