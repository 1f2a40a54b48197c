"""Probe: per-tensor error of linear_gelu_linear fwd/bwd vs fp32 tanh-GELU
chain (bf16 path uses the split-epilogue fallback), and the O0/O2/O3 ResNet
first-loss comparison."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import apex_amd._fused_dense as fd


def probe_gelu(dtype):
    torch.manual_seed(8)
    m, nin, nh, nout = 32, 64, 256, 48
    x = torch.randn(m, nin, device="cuda", dtype=dtype)
    w1 = torch.randn(nh, nin, device="cuda", dtype=dtype) * 0.1
    b1 = torch.randn(nh, device="cuda", dtype=dtype)
    w2 = torch.randn(nout, nh, device="cuda", dtype=dtype) * 0.1
    b2 = torch.randn(nout, device="cuda", dtype=dtype)
    o1, o2, gi = fd.linear_gelu_linear_forward(x, w1, b1, w2, b2)

    xf = x.float().requires_grad_(True)
    w1f = w1.float().requires_grad_(True)
    b1f = b1.float().requires_grad_(True)
    w2f = w2.float().requires_grad_(True)
    b2f = b2.float().requires_grad_(True)
    z1 = torch.nn.functional.linear(xf, w1f, b1f)
    o1f = torch.nn.functional.gelu(z1, approximate="tanh")
    o2f = torch.nn.functional.linear(o1f, w2f, b2f)

    print(f"[{dtype}] fwd: |gi-z1| {(gi.float()-z1).abs().max():.4f} "
          f"|o1-ref| {(o1.float()-o1f).abs().max():.4f} |o2-ref| {(o2.float()-o2f).abs().max():.4f}")

    dy = torch.randn(m, nout, device="cuda", dtype=dtype)
    dx, dw1, db1, dw2, db2 = fd.linear_gelu_linear_backward(x, gi, o1, w1, w2, dy)
    o2f.backward(dy.float())
    for name, a, b in [("dx", dx, xf.grad), ("dw1", dw1, w1f.grad), ("db1", db1, b1f.grad),
                       ("dw2", dw2, w2f.grad), ("db2", db2, b2f.grad)]:
        err = (a.float() - b).abs().max().item()
        rel = err / (b.abs().max().item() + 1e-9)
        print(f"[{dtype}] bwd {name}: max abs {err:.5f} rel {rel:.4f}")


def probe_resnet_o2():
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from apex_amd.models import resnet50

    torch.manual_seed(7)
    model = resnet50(num_classes=100).cuda()
    gen = torch.Generator().manual_seed(7)
    x = torch.randn(16, 3, 96, 96, generator=gen).cuda()
    y = torch.randint(0, 100, (16,), generator=gen).cuda()

    with torch.no_grad():
        loss0 = torch.nn.functional.cross_entropy(model(x).float(), y)
        # O3-style: full bf16
        m3 = resnet50(num_classes=100)
        torch.manual_seed(7)
        m3 = resnet50(num_classes=100).cuda().to(torch.bfloat16)
        loss3 = torch.nn.functional.cross_entropy(m3(x.bfloat16()).float(), y)
        # O2-style: bf16 except BN fp32
        torch.manual_seed(7)
        m2 = resnet50(num_classes=100).cuda().to(torch.bfloat16)
        for mod in m2.modules():
            if isinstance(mod, torch.nn.modules.batchnorm._BatchNorm):
                mod.float()
        loss2 = torch.nn.functional.cross_entropy(m2(x.bfloat16()).float(), y)
    print(f"resnet first loss: O0 {loss0:.4f}  O3-bf16 {loss3:.4f}  O2-bf16+bnfp32 {loss2:.4f}")


def run_all():
    probe_gelu(torch.bfloat16)
    probe_gelu(torch.float16)
    probe_gelu(torch.float32)
    probe_resnet_o2()


def probe_train_o2():
    """Replicate the integration test's O2 _train first loss, bisecting amp."""
    from apex_amd import amp
    from apex_amd.amp._amp_state import _amp_state
    from apex_amd.models import resnet50
    from apex_amd.optimizers import FusedSGD

    def first_loss(use_amp, opt_level="O2"):
        _amp_state.reset()
        torch.manual_seed(7)
        torch.backends.cudnn.deterministic = True
        model = resnet50(num_classes=100).cuda()
        opt = FusedSGD(model.parameters(), lr=0.01, momentum=0.9)
        if use_amp:
            model, opt = amp.initialize(model, opt, opt_level=opt_level,
                                        cast_model_type=None if opt_level in ("O0","O1") else torch.bfloat16,
                                        loss_scale=128.0, verbosity=0)
        gen = torch.Generator().manual_seed(7)
        x = torch.randn(16, 3, 96, 96, generator=gen).cuda()
        y = torch.randint(0, 100, (16,), generator=gen).cuda()
        out = model(x)
        return float(torch.nn.functional.cross_entropy(out.float(), y))

    print("train-first-loss O0:", first_loss(True, "O0"))
    print("train-first-loss O2 amp:", first_loss(True, "O2"))
    print("train-first-loss O3 amp:", first_loss(True, "O3"))
    print("train-first-loss no-amp fp32:", first_loss(False))


if __name__ == "__main__":
    run_all()
    probe_train_o2()
