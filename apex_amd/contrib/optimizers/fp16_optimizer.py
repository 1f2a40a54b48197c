"""FP16_Optimizer — legacy fp16 master-weight wrapper.

API parity with the reference ``apex.contrib.optimizers.FP16_Optimizer``
(apex/contrib/optimizers/fp16_optimizer.py:5-248): flat fp32 master groups,
``backward(loss)`` applies the loss scale, ``step()`` unscales with overflow
check and dynamic loss-scale update. Superseded by amp O2 but kept for
surface parity.
"""

import torch

from ...amp.scaler import LossScaler


class FP16_Optimizer:
    def __init__(self, init_optimizer, static_loss_scale=1.0, dynamic_loss_scale=False,
                 dynamic_loss_args=None, verbose=False):
        self.optimizer = init_optimizer
        self.fp16_groups = []
        self.fp32_groups = []
        for group in self.optimizer.param_groups:
            fp16 = [p for p in group["params"]]
            fp32 = [p.detach().clone().float().requires_grad_(True) for p in fp16]
            self.fp16_groups.append(fp16)
            self.fp32_groups.append(fp32)
            group["params"] = fp32
        if dynamic_loss_scale:
            self.scaler = LossScaler("dynamic", **(dynamic_loss_args or {}))
        else:
            self.scaler = LossScaler(static_loss_scale)
        self.verbose = verbose

    @property
    def loss_scale(self):
        return self.scaler.loss_scale()

    def zero_grad(self, set_grads_to_None=True):
        for group in self.fp16_groups:
            for p in group:
                p.grad = None
        for group in self.fp32_groups:
            for p in group:
                p.grad = None

    def backward(self, loss):
        (loss.float() * self.scaler.loss_scale()).backward()

    def step(self, closure=None):
        # unscale fp16 grads into fp32 masters
        model_grads, master_grads = [], []
        for g16, g32 in zip(self.fp16_groups, self.fp32_groups):
            for p16, p32 in zip(g16, g32):
                if p16.grad is not None:
                    if p32.grad is None:
                        p32.grad = torch.empty_like(p32)
                    model_grads.append(p16.grad)
                    master_grads.append(p32.grad)
        overflow = self.scaler.unscale_grads(model_grads, master_grads)
        if overflow:
            if self.verbose:
                print(f"Gradient overflow. Skipping step, reducing loss scale to {self.loss_scale}")
            return
        self.optimizer.step()
        with torch.no_grad():
            for g16, g32 in zip(self.fp16_groups, self.fp32_groups):
                for p16, p32 in zip(g16, g32):
                    p16.copy_(p32.to(p16.dtype))

    def state_dict(self):
        return {
            "optimizer_state_dict": self.optimizer.state_dict(),
            "scaler": self.scaler.state_dict(),
            "fp32_groups": self.fp32_groups,
        }

    def load_state_dict(self, sd):
        self.optimizer.load_state_dict(sd["optimizer_state_dict"])
        self.scaler.load_state_dict(sd["scaler"])
        for cur, saved in zip(self.fp32_groups, sd["fp32_groups"]):
            for c, s in zip(cur, saved):
                c.data.copy_(s.data)
