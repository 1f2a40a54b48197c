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
    def __init__(self, params, swa_params=None, compute_params=None, swa_decay_rate=0.9,
                 lr=1e-3, bias_correction=True, betas=(0.9, 0.999), eps=1e-8,
                 adam_w_mode=True, weight_decay=0.0):
        super().__init__(params, lr=lr, bias_correction=bias_correction, betas=betas, eps=eps,
                         adam_w_mode=adam_w_mode, weight_decay=weight_decay)
        self.swa_decay_rate = swa_decay_rate
        if swa_params is not None:
            self.swa_params = list(swa_params)
        else:
            self.swa_params = [p.detach().clone() for g in self.param_groups for p in g["params"]]
        self._model_params = [p for g in self.param_groups for p in g["params"]]
        assert len(self.swa_params) == len(self._model_params)
        # reference contract: fp32 params are the optimizer state, an
        # optional low-precision compute copy is refreshed after each step
        self.compute_params = list(compute_params) if compute_params is not None else None
        if self.compute_params is not None:
            assert len(self.compute_params) == len(self._model_params)

    @classmethod
    def from_optim(cls, adam_optimizer, fp32_params, bf16_params, swa_params,
                   swa_decay_rate):
        """Build from an existing ``torch.optim.Adam``, inheriting its
        hyperparameters and per-param moments (reference:
        openfold_triton/fused_adam_swa.py from_optim)."""
        assert len(adam_optimizer.param_groups) == 1
        g = adam_optimizer.param_groups[0]
        opt = cls(params=list(fp32_params), compute_params=list(bf16_params),
                  swa_params=list(swa_params), swa_decay_rate=swa_decay_rate,
                  lr=g["lr"], betas=g["betas"], eps=g["eps"],
                  weight_decay=g["weight_decay"])
        steps = set()
        for src, dst in zip(g["params"], opt.param_groups[0]["params"]):
            st = adam_optimizer.state.get(src)
            if not st:
                continue
            opt.state[dst]["exp_avg"] = st["exp_avg"].detach().clone().float()
            opt.state[dst]["exp_avg_sq"] = st["exp_avg_sq"].detach().clone().float()
            s = st.get("step", 0)
            steps.add(int(s.item()) if torch.is_tensor(s) else int(s))
        if len(steps) > 1:
            raise ValueError("FusedAdamSWA requires all params updated by the same steps")
        if steps:
            opt.param_groups[0]["step"] = steps.pop()
        return opt

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
        if self.compute_params is not None:
            if device.type == "cuda":
                amp_C = get_ext("amp_C")
                noop = torch.zeros(1, dtype=torch.int32, device=device)
                multi_tensor_applier(
                    amp_C.multi_tensor_scale, noop,
                    [self._model_params, self.compute_params], 1.0,
                )
            else:
                for c, p in zip(self.compute_params, self._model_params):
                    c.copy_(p.to(c.dtype))
        return loss
