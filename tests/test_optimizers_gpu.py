"""GPU optimizer tests: fused HIP kernels vs torch fp32 references on-device."""

import torch
import pytest

from apex_amd.optimizers import FusedAdam, FusedSGD, FusedLAMB, FusedAdagrad, FusedNovoGrad

pytestmark = pytest.mark.gpu


def make_params(dtype=torch.float32, seed=0, shapes=((64, 64), (129,), (33, 37), (2048, 11))):
    torch.manual_seed(seed)
    ps_a, ps_b = [], []
    for s in shapes:
        t = torch.randn(*s, device="cuda").to(dtype)
        ps_a.append(t.clone().requires_grad_(True))
        ps_b.append(t.clone().requires_grad_(True))
    return ps_a, ps_b


def set_same_grads(ps_a, ps_b, seed, scale=1.0):
    torch.manual_seed(seed)
    for a, b in zip(ps_a, ps_b):
        g = (torch.randn_like(a.float()) * scale).to(a.dtype)
        a.grad = g.clone()
        b.grad = g.clone()


@pytest.mark.parametrize("adam_w_mode", [True, False])
def test_fused_adam_vs_torch_gpu(adam_w_mode):
    ps_ref, ps_tst = make_params()
    cls = torch.optim.AdamW if adam_w_mode else torch.optim.Adam
    ref = cls(ps_ref, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=0.01)
    tst = FusedAdam(ps_tst, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=0.01,
                    adam_w_mode=adam_w_mode)
    for i in range(10):
        set_same_grads(ps_ref, ps_tst, seed=i)
        ref.step()
        tst.step()
    torch.cuda.synchronize()
    for a, b in zip(ps_ref, ps_tst):
        torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("dtype", [torch.float16, torch.bfloat16])
def test_fused_adam_low_precision_params(dtype):
    # fp16/bf16 params with fp32 state vs fp32 reference math
    ps, _ = make_params(dtype=dtype, shapes=((128, 65), (77,)))
    masters = [p.detach().float().clone() for p in ps]
    tst = FusedAdam(ps, lr=1e-3, weight_decay=0.0)
    m = [torch.zeros_like(x) for x in masters]
    v = [torch.zeros_like(x) for x in masters]
    for step in range(1, 4):
        set_same_grads(ps, ps, seed=step)
        tst.step()
        # fp32 reference on masters using the low-precision grads
        for i, p in enumerate(masters):
            g = ps[i].grad.float()
            m[i].mul_(0.9).add_(g, alpha=0.1)
            v[i].mul_(0.999).addcmul_(g, g, value=0.001)
            bc1 = 1 - 0.9 ** step
            bc2 = 1 - 0.999 ** step
            p.sub_(1e-3 * (m[i] / bc1) / ((v[i] / bc2).sqrt() + 1e-8))
    torch.cuda.synchronize()
    tol = 1e-2 if dtype == torch.float16 else 5e-2
    for p, master in zip(ps, masters):
        torch.testing.assert_close(p.float(), master, rtol=tol, atol=tol)


def test_fused_adam_capturable_matches_plain():
    ps_a, ps_b = make_params()
    a = FusedAdam(ps_a, lr=1e-3, weight_decay=0.01)
    b = FusedAdam(ps_b, lr=1e-3, weight_decay=0.01, capturable=True)
    for i in range(5):
        set_same_grads(ps_a, ps_b, seed=i)
        a.step()
        b.step()
    torch.cuda.synchronize()
    for x, y in zip(ps_a, ps_b):
        torch.testing.assert_close(x, y, rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize("momentum,nesterov,wd", [(0.9, False, 0.0), (0.9, True, 1e-4)])
def test_fused_sgd_vs_torch_gpu(momentum, nesterov, wd):
    ps_ref, ps_tst = make_params()
    ref = torch.optim.SGD(ps_ref, lr=0.1, momentum=momentum, nesterov=nesterov, weight_decay=wd)
    tst = FusedSGD(ps_tst, lr=0.1, momentum=momentum, nesterov=nesterov, weight_decay=wd)
    for i in range(10):
        set_same_grads(ps_ref, ps_tst, seed=i)
        ref.step()
        tst.step()
    torch.cuda.synchronize()
    for a, b in zip(ps_ref, ps_tst):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_fused_adagrad_vs_torch_gpu():
    ps_ref, ps_tst = make_params()
    ref = torch.optim.Adagrad(ps_ref, lr=1e-2, eps=1e-10, weight_decay=1e-4)
    tst = FusedAdagrad(ps_tst, lr=1e-2, eps=1e-10, weight_decay=1e-4)
    for i in range(10):
        set_same_grads(ps_ref, ps_tst, seed=i)
        ref.step()
        tst.step()
    torch.cuda.synchronize()
    for a, b in zip(ps_ref, ps_tst):
        torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-5)


def test_fused_lamb_gpu_vs_cpu_reference():
    # GPU fused LAMB vs the same optimizer's CPU reference path
    ps_gpu, _ = make_params()
    ps_cpu = [p.detach().cpu().clone().requires_grad_(True) for p in ps_gpu]
    gpu = FusedLAMB(ps_gpu, lr=1e-3, weight_decay=0.01)
    cpu = FusedLAMB(ps_cpu, lr=1e-3, weight_decay=0.01)
    for i in range(5):
        torch.manual_seed(i)
        for a, b in zip(ps_gpu, ps_cpu):
            g = torch.randn_like(a)
            a.grad = g.clone()
            b.grad = g.cpu().clone()
        gpu.step()
        cpu.step()
    torch.cuda.synchronize()
    for a, b in zip(ps_gpu, ps_cpu):
        torch.testing.assert_close(a.cpu(), b, rtol=1e-4, atol=1e-5)


def test_fused_novograd_gpu_vs_cpu_reference():
    ps_gpu, _ = make_params()
    ps_cpu = [p.detach().cpu().clone().requires_grad_(True) for p in ps_gpu]
    gpu = FusedNovoGrad(ps_gpu, lr=1e-2, weight_decay=1e-4)
    cpu = FusedNovoGrad(ps_cpu, lr=1e-2, weight_decay=1e-4)
    for i in range(5):
        torch.manual_seed(i)
        for a, b in zip(ps_gpu, ps_cpu):
            g = torch.randn_like(a)
            a.grad = g.clone()
            b.grad = g.cpu().clone()
        gpu.step()
        cpu.step()
    torch.cuda.synchronize()
    for a, b in zip(ps_gpu, ps_cpu):
        torch.testing.assert_close(a.cpu(), b, rtol=1e-4, atol=1e-5)


def test_loss_scaler_gpu_dynamic():
    from apex_amd.amp.scaler import LossScaler

    s = LossScaler("dynamic", init_scale=2.0 ** 10, scale_window=2)
    g = [torch.ones(64, device="cuda") * 2048.0]
    out = [torch.empty_like(g[0])]
    assert not s.unscale_grads(g, out)
    torch.testing.assert_close(out[0], g[0] / 2.0 ** 10)
    # overflow path
    g_inf = [torch.full((8,), float("inf"), device="cuda")]
    o = [torch.empty_like(g_inf[0])]
    assert s.unscale_grads(g_inf, o)
    assert s.loss_scale() < 2.0 ** 10 or s._hysteresis_t is not None


def test_fused_lamb_capturable_graph_replay():
    """FusedLAMB(capturable=True): graph capture + replays track the eager
    FusedLAMB (device step advances per replay, in-kernel bias correction)."""
    from apex_amd.optimizers import FusedLAMB

    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(64, 64), torch.nn.Tanh(),
                                torch.nn.Linear(64, 32)).cuda()
    ref = torch.nn.Sequential(torch.nn.Linear(64, 64), torch.nn.Tanh(),
                              torch.nn.Linear(64, 32)).cuda()
    ref.load_state_dict(model.state_dict())
    opt = FusedLAMB(model.parameters(), lr=1e-3, weight_decay=0.01,
                    capturable=True, set_grad_none=False)
    ropt = FusedLAMB(ref.parameters(), lr=1e-3, weight_decay=0.01,
                     set_grad_none=False)
    xs = [torch.randn(8, 64, device="cuda") for _ in range(5)]

    def fwd_bwd(m, x):
        for p in m.parameters():
            if p.grad is not None:
                p.grad.zero_()
        m(x).pow(2).mean().backward()

    # eager warmup (momentum init, algo caches)
    fwd_bwd(model, xs[0]); opt.step()
    fwd_bwd(ref, xs[0]); ropt.step()

    fwd_bwd(model, xs[1])
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        opt.step()
    g.replay()
    fwd_bwd(ref, xs[1]); ropt.step()
    torch.cuda.synchronize()
    for p, rp in zip(model.parameters(), ref.parameters()):
        torch.testing.assert_close(p.detach(), rp.detach(), rtol=2e-4, atol=2e-5)

    for x in xs[2:]:
        fwd_bwd(model, x)
        g.replay()
        fwd_bwd(ref, x); ropt.step()
    torch.cuda.synchronize()
    for p, rp in zip(model.parameters(), ref.parameters()):
        torch.testing.assert_close(p.detach(), rp.detach(), rtol=5e-4, atol=5e-5)
    assert int(opt._step_t.item()) == 5


@pytest.mark.gpu
def test_fused_lamb_capturable_eager_lr_schedule():
    """Eager capturable steps must track group["lr"] changes (the device
    _lr_t is refreshed per group outside capture); two groups with
    different lrs each see their own lr."""
    torch.manual_seed(0)
    pa = torch.randn(64, device="cuda", requires_grad=True)
    pb = torch.randn(64, device="cuda", requires_grad=True)
    ref_a = pa.detach().clone().requires_grad_(True)
    ref_b = pb.detach().clone().requires_grad_(True)

    opt = FusedLAMB([{"params": [pa], "lr": 1e-2},
                     {"params": [pb], "lr": 1e-3}],
                    weight_decay=0.01, capturable=True)
    ref = FusedLAMB([{"params": [ref_a], "lr": 1e-2},
                     {"params": [ref_b], "lr": 1e-3}],
                    weight_decay=0.01, capturable=False)

    for i in range(3):
        if i == 2:  # lr schedule tick
            for o in (opt, ref):
                o.param_groups[0]["lr"] = 5e-3
                o.param_groups[1]["lr"] = 5e-4
        torch.manual_seed(10 + i)
        for p, r in ((pa, ref_a), (pb, ref_b)):
            g = torch.randn_like(p)
            p.grad, r.grad = g.clone(), g.clone()
        opt.step()
        ref.step()

    torch.testing.assert_close(pa, ref_a, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(pb, ref_b, rtol=1e-5, atol=1e-6)
    # fp32 compare (5e-4 is not exactly representable in float32)
    assert float(opt._lr_t) == pytest.approx(5e-4, rel=1e-6)
