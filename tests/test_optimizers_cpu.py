"""CPU-path optimizer tests: apex_amd fused optimizers vs torch references.

Pattern mirrors the reference L0 suite
(tests/L0/run_optimizers/test_fused_optimizer.py): identical param sets,
identical grads, step both, compare.
"""

import torch
import pytest

from apex_amd.optimizers import FusedAdam, FusedSGD, FusedLAMB, FusedAdagrad, FusedNovoGrad


def make_params(seed=0, shapes=((64, 64), (128,), (33, 7))):
    torch.manual_seed(seed)
    ps_a, ps_b = [], []
    for s in shapes:
        t = torch.randn(*s)
        a = t.clone().requires_grad_(True)
        b = t.clone().requires_grad_(True)
        ps_a.append(a)
        ps_b.append(b)
    return ps_a, ps_b


def set_same_grads(ps_a, ps_b, seed):
    torch.manual_seed(seed)
    for a, b in zip(ps_a, ps_b):
        g = torch.randn_like(a)
        a.grad = g.clone()
        b.grad = g.clone()


@pytest.mark.parametrize("adam_w_mode", [True, False])
def test_fused_adam_matches_torch(adam_w_mode):
    ps_ref, ps_tst = make_params()
    wd = 0.01
    if adam_w_mode:
        ref_opt = torch.optim.AdamW(ps_ref, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=wd)
    else:
        ref_opt = torch.optim.Adam(ps_ref, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=wd)
    tst_opt = FusedAdam(ps_tst, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=wd, adam_w_mode=adam_w_mode)
    for i in range(10):
        set_same_grads(ps_ref, ps_tst, seed=100 + i)
        ref_opt.step()
        tst_opt.step()
        for a, b in zip(ps_ref, ps_tst):
            torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_fused_adam_multi_group():
    ps_ref, ps_tst = make_params()
    ref_opt = torch.optim.AdamW(
        [{"params": ps_ref[:1], "lr": 1e-3}, {"params": ps_ref[1:], "lr": 2e-4}], weight_decay=0.0
    )
    tst_opt = FusedAdam(
        [{"params": ps_tst[:1], "lr": 1e-3}, {"params": ps_tst[1:], "lr": 2e-4}], weight_decay=0.0
    )
    for i in range(5):
        set_same_grads(ps_ref, ps_tst, seed=i)
        ref_opt.step()
        tst_opt.step()
    for a, b in zip(ps_ref, ps_tst):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize("momentum,nesterov,wd", [(0.9, False, 0.0), (0.9, True, 1e-4), (0.0, False, 0.0)])
def test_fused_sgd_matches_torch(momentum, nesterov, wd):
    ps_ref, ps_tst = make_params()
    ref_opt = torch.optim.SGD(ps_ref, lr=0.1, momentum=momentum, nesterov=nesterov, weight_decay=wd)
    tst_opt = FusedSGD(ps_tst, lr=0.1, momentum=momentum, nesterov=nesterov, weight_decay=wd)
    for i in range(10):
        set_same_grads(ps_ref, ps_tst, seed=i)
        ref_opt.step()
        tst_opt.step()
        for a, b in zip(ps_ref, ps_tst):
            torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_fused_adagrad_matches_torch():
    ps_ref, ps_tst = make_params()
    ref_opt = torch.optim.Adagrad(ps_ref, lr=1e-2, eps=1e-10, weight_decay=1e-4, lr_decay=0.0)
    tst_opt = FusedAdagrad(ps_tst, lr=1e-2, eps=1e-10, weight_decay=1e-4)
    for i in range(10):
        set_same_grads(ps_ref, ps_tst, seed=i)
        ref_opt.step()
        tst_opt.step()
        for a, b in zip(ps_ref, ps_tst):
            torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-6)


class RefLAMB(torch.optim.Optimizer):
    """Hand-written LAMB reference (pattern of reference tests/L0
    test_lamb.py:11-171) matching the apex kernel semantics."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-6, weight_decay=0.01,
                 max_grad_norm=1.0, bias_correction=True, grad_averaging=True, adam_w_mode=True,
                 use_nvlamb=False):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.max_grad_norm = max_grad_norm
        self.bias_correction = bias_correction
        self.grad_averaging = grad_averaging
        self.adam_w_mode = adam_w_mode
        self.use_nvlamb = use_nvlamb
        self._step = 0

    @torch.no_grad()
    def step(self):
        self._step += 1
        sq = 0.0
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is not None:
                    sq += float(p.grad.float().pow(2).sum())
        gnorm = sq ** 0.5
        clip = gnorm / self.max_grad_norm if (self.max_grad_norm > 0 and gnorm > self.max_grad_norm) else 1.0
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            bc1 = 1 - beta1 ** self._step if self.bias_correction else 1.0
            bc2 = 1 - beta2 ** self._step if self.bias_correction else 1.0
            beta3 = 1 - beta1 if self.grad_averaging else 1.0
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["m"] = torch.zeros_like(p)
                    state["v"] = torch.zeros_like(p)
                g = p.grad / clip
                if not self.adam_w_mode and group["weight_decay"] != 0:
                    g = g + group["weight_decay"] * p
                state["m"].mul_(beta1).add_(g, alpha=beta3)
                state["v"].mul_(beta2).addcmul_(g, g, value=1 - beta2)
                update = (state["m"] / bc1) / ((state["v"] / bc2).sqrt() + group["eps"])
                if self.adam_w_mode and group["weight_decay"] != 0:
                    update = update + group["weight_decay"] * p
                pn, un = p.norm(), update.norm()
                if (self.use_nvlamb or group["weight_decay"] != 0) and pn != 0 and un != 0:
                    ratio = group["lr"] * pn / un
                else:
                    ratio = group["lr"]
                p.add_(update, alpha=-float(ratio))


def test_fused_lamb_matches_reference():
    ps_ref, ps_tst = make_params()
    ref_opt = RefLAMB(ps_ref, lr=1e-3, weight_decay=0.01)
    tst_opt = FusedLAMB(ps_tst, lr=1e-3, weight_decay=0.01)
    for i in range(10):
        set_same_grads(ps_ref, ps_tst, seed=i)
        ref_opt.step()
        tst_opt.step()
        for a, b in zip(ps_ref, ps_tst):
            torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-6)


def test_fused_novograd_decreases_loss():
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Tanh(), torch.nn.Linear(32, 1))
    opt = FusedNovoGrad(model.parameters(), lr=5e-2)
    x = torch.randn(64, 16)
    y = torch.randn(64, 1)
    losses = []
    for _ in range(50):
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.7


def test_optimizer_state_dict_roundtrip():
    ps_a, ps_b = make_params()
    opt_a = FusedAdam(ps_a, lr=1e-3)
    for i in range(3):
        set_same_grads(ps_a, ps_a, seed=i)
        opt_a.step()
    sd = opt_a.state_dict()
    opt_b = FusedAdam(ps_b, lr=1e-3)
    opt_b.load_state_dict(sd)
    set_same_grads(ps_a, ps_a, seed=99)
    set_same_grads(ps_b, ps_b, seed=99)
    opt_a.step()
    opt_b.step()
    # states should evolve identically after load (params started equal)
    for a, b in zip(ps_a, ps_b):
        torch.testing.assert_close(opt_a.state[a]["exp_avg"], opt_b.state[b]["exp_avg"])


def test_fused_mixed_precision_lamb_masters_track_fp32():
    # bf16 params + reduced_precision_dtype=bf16 keep fp32 masters; given
    # identical fp32 grads the master trajectory must match a pure-fp32 run,
    # and the model params must equal the masters cast down each step
    # (reference apex/optimizers/fused_mixed_precision_lamb.py semantics)
    from apex_amd.optimizers import FusedMixedPrecisionLamb

    torch.manual_seed(0)
    base = [torch.randn(7, 5), torch.randn(11)]
    p32 = [b.clone().requires_grad_(True) for b in base]
    pbf = [b.clone().bfloat16().requires_grad_(True) for b in base]

    o32 = FusedMixedPrecisionLamb(p32, lr=1e-2, weight_decay=0.01)
    obf = FusedMixedPrecisionLamb(pbf, lr=1e-2, weight_decay=0.01,
                                  reduced_precision_dtype=torch.bfloat16)
    masters = obf.param_groups_full_precision[0]["params"]
    assert all(m is not None and m.dtype == torch.float32 for m in masters)

    for i in range(5):
        torch.manual_seed(100 + i)
        for a, b in zip(p32, pbf):
            g = torch.randn_like(a)
            a.grad = g.clone()
            b.grad = g.clone().bfloat16()
        o32.step()
        obf.step()

    assert int(o32.param_groups[0]["step"].item()) == 5
    for m, p in zip(masters, pbf):
        torch.testing.assert_close(p.detach(), m.to(torch.bfloat16))
    # masters track the fp32 run up to bf16 gradient rounding
    for a, m in zip(p32, masters):
        torch.testing.assert_close(a.detach(), m, rtol=3e-2, atol=3e-2)


def test_fused_mixed_precision_lamb_converges():
    from apex_amd.optimizers import FusedMixedPrecisionLamb

    torch.manual_seed(3)
    model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Tanh(),
                                torch.nn.Linear(32, 1))
    opt = FusedMixedPrecisionLamb(model.parameters(), lr=5e-2)
    x, y = torch.randn(64, 16), torch.randn(64, 1)
    losses = []
    for _ in range(50):
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.7


def test_fused_lamb_capturable_checkpoint_step_sync():
    # graph replays advance only the device _step_t; state_dict must carry
    # the true count and load must restore it (unit-level: _step_t stubbed
    # on CPU — the cuda branch creates it device-side from group["step"]-1)
    from apex_amd.optimizers import FusedLAMB

    ps = [torch.randn(4, requires_grad=True)]
    ps[0].grad = torch.randn(4)
    opt = FusedLAMB(ps, lr=1e-3, capturable=True)
    opt.step()  # CPU ref path: group["step"] = 1
    opt._step_t = torch.tensor([7], dtype=torch.int32)  # pretend 6 replays
    sd = opt.state_dict()
    assert all(g["step"] == 7 for g in sd["param_groups"])

    ps2 = [torch.randn(4, requires_grad=True)]
    opt2 = FusedLAMB(ps2, lr=1e-3, capturable=True)
    opt2._step_t = torch.tensor([0], dtype=torch.int32)
    opt2.load_state_dict(sd)
    assert int(opt2._step_t) == 7
    # fresh optimizer without a live _step_t: the loaded host count is the
    # seed for device-side creation (group["step"] round-tripped)
    opt3 = FusedLAMB([torch.randn(4, requires_grad=True)], lr=1e-3, capturable=True)
    opt3.load_state_dict(sd)
    assert opt3.param_groups[0]["step"] == 7 and opt3._step_t is None


def test_fused_adam_capturable_load_normalizes_group_tensors():
    # capturable FusedAdam keeps lr/step as tensors inside param_groups;
    # after load_state_dict they must sit on the params' device (CPU here —
    # the override is a no-op move, but must not crash and must keep values)
    from apex_amd.optimizers import FusedAdam

    ps = [torch.randn(4, requires_grad=True)]
    ps[0].grad = torch.randn(4)
    opt = FusedAdam(ps, lr=1e-3, capturable=True)
    opt.step()
    sd = opt.state_dict()
    ps2 = [torch.randn(4, requires_grad=True)]
    opt2 = FusedAdam(ps2, lr=1e-3, capturable=True)
    opt2.load_state_dict(sd)
    g = opt2.param_groups[0]
    assert torch.is_tensor(g["lr"]) and g["lr"].device == ps2[0].device
    assert float(g["lr"]) == pytest.approx(1e-3, rel=1e-6)
    step = g["step"] if not torch.is_tensor(g["step"]) else int(g["step"].item())
    assert int(step) == 1


def test_fused_optimizers_frozen_params():
    # reference frozen-model test (L0 test_fused_optimizer.py:201): params
    # with no grad are skipped, not zero-updated
    from apex_amd.optimizers import FusedAdam, FusedLAMB, FusedSGD, FusedNovoGrad

    for cls, kw in ((FusedAdam, {}), (FusedLAMB, {}), (FusedSGD, {"momentum": 0.9}),
                    (FusedNovoGrad, {})):
        torch.manual_seed(0)
        live = torch.randn(6, requires_grad=True)
        frozen = torch.randn(6, requires_grad=False)
        frozen_before = frozen.detach().clone()
        opt = cls([live, frozen], lr=1e-2, **kw)
        for i in range(3):
            torch.manual_seed(i)
            live.grad = torch.randn(6)
            opt.step()
        assert torch.equal(frozen, frozen_before), cls.__name__
        assert not torch.equal(live.detach(), frozen_before), cls.__name__
