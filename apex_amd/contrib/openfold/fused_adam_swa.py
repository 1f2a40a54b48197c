"""FusedAdamSWA — Adam step + stochastic weight averaging in fused launches.

Counterpart of the reference ``apex.contrib.openfold_triton.FusedAdamSWA``
(apex/contrib/openfold_triton/fused_adam_swa.py): maintains a parallel set
of SWA params updated as ``swa += (p - swa) * decay_factor`` after each Adam
step. Device path: one multi_tensor_adam launch + one multi_tensor_axpby
launch for the running average.
"""

import torch

from ..._ext import get_ext
from ...multi_tensor_apply import multi_tensor_applier
from ...optimizers import FusedAdam


class FusedAdamSWA(FusedAdam):
    def __init__(self, params, swa_params=None, swa_decay_rate=0.9, lr=1e-3,
                 bias_correction=True, betas=(0.9, 0.999), eps=1e-8, adam_w_mode=True,
                 weight_decay=0.0):
        super().__init__(params, lr=lr, bias_correction=bias_correction, betas=betas, eps=eps,
                         adam_w_mode=adam_w_mode, weight_decay=weight_decay)
        self.swa_decay_rate = swa_decay_rate
        if swa_params is not None:
            self.swa_params = list(swa_params)
        else:
            self.swa_params = [p.detach().clone() for g in self.param_groups for p in g["params"]]
        self._model_params = [p for g in self.param_groups for p in g["params"]]
        assert len(self.swa_params) == len(self._model_params)

    @torch.no_grad()
    def step(self, closure=None):
        loss = super().step(closure)
        d = self.swa_decay_rate
        device = self._model_params[0].device
        if device.type == "cuda":
            amp_C = get_ext("amp_C")
            noop = torch.zeros(1, dtype=torch.int32, device=device)
            # swa = d * swa + (1-d) * p
            multi_tensor_applier(
                amp_C.multi_tensor_axpby, noop,
                [self.swa_params, self._model_params, self.swa_params], d, 1.0 - d, -1,
            )
        else:
            for s, p in zip(self.swa_params, self._model_params):
                s.mul_(d).add_(p.to(s.dtype), alpha=1.0 - d)
        return loss
