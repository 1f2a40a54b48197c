"""Probe: O2 (bf16 + fp32 masters + FusedSGD) 12-step trajectory vs a manual
master-weights reference implementing the same math."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def run_amp_o2(loss_scale=128.0, iters=12):
    from apex_amd import amp
    from apex_amd.amp._amp_state import _amp_state
    from apex_amd.models import resnet50
    from apex_amd.optimizers import FusedSGD
    _amp_state.reset()
    torch.manual_seed(7)
    model = resnet50(num_classes=100).cuda()
    opt = FusedSGD(model.parameters(), lr=0.01, momentum=0.9)
    model, opt = amp.initialize(model, opt, opt_level="O2",
                                cast_model_type=torch.bfloat16, loss_scale=loss_scale, verbosity=0)
    gen = torch.Generator().manual_seed(7)
    x = torch.randn(16, 3, 96, 96, generator=gen).cuda()
    y = torch.randint(0, 100, (16,), generator=gen).cuda()
    losses = []
    for i in range(iters):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x).float(), y)
        with amp.scale_loss(loss, opt) as scaled:
            scaled.backward()
        opt.step()
        losses.append(round(float(loss.detach()), 4))
        if i == 0:
            # master/model consistency after first step
            st = opt._amp_stash
            md = max((m.to(p.dtype) - p).abs().max().item()
                     for m, p in zip(st.all_fp32_from_fp16_params, st.all_fp16_params))
            print("  after step1 max |master-model|:", md)
    return losses


def run_manual_master(iters=12):
    from apex_amd.models import resnet50
    torch.manual_seed(7)
    model = resnet50(num_classes=100).cuda()
    # manual O2: bf16 model (BN fp32), fp32 masters, plain torch SGD on masters
    model = model.to(torch.bfloat16)
    for m in model.modules():
        if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
            m.float()
    params = [p for p in model.parameters()]
    masters = [p.detach().float().clone() for p in params]
    moms = [torch.zeros_like(m) for m in masters]
    gen = torch.Generator().manual_seed(7)
    x = torch.randn(16, 3, 96, 96, generator=gen).cuda().bfloat16()
    y = torch.randint(0, 100, (16,), generator=gen).cuda()
    losses = []
    for i in range(iters):
        for p in params:
            p.grad = None
        loss = torch.nn.functional.cross_entropy(model(x).float(), y)
        (loss * 128.0).backward()
        with torch.no_grad():
            for p, mstr, mom in zip(params, masters, moms):
                g = p.grad.float() / 128.0
                if i == 0:
                    mom.copy_(g)
                else:
                    mom.mul_(0.9).add_(g)
                mstr.sub_(0.01 * mom)
                p.copy_(mstr.to(p.dtype))
        losses.append(round(float(loss.detach()), 4))
    return losses


if __name__ == "__main__":
    print("amp O2 static128:", run_amp_o2(128.0))
    print("amp O2 scale1   :", run_amp_o2(1.0))
    print("manual master   :", run_manual_master())
