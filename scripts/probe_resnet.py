"""Probe ResNet-50 AMP step-time variants: SyncBN vs BN, NCHW vs channels_last."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def run(use_syncbn, channels_last, batch=128, iters=10):
    from apex_amd import amp
    from apex_amd.amp._amp_state import _amp_state
    from apex_amd.models import resnet50
    from apex_amd.optimizers import FusedSGD
    from apex_amd.parallel import convert_syncbn_model

    _amp_state.reset()
    torch.manual_seed(0)
    torch.backends.cudnn.benchmark = True
    model = resnet50()
    if use_syncbn:
        model = convert_syncbn_model(model)
    model = model.cuda()
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4)
    model, opt = amp.initialize(model, opt, opt_level="O1",
                                cast_model_type=torch.bfloat16, loss_scale=1.0, verbosity=0)
    x = torch.randn(batch, 3, 224, 224, device="cuda")
    if channels_last:
        x = x.to(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (batch,), device="cuda")
    crit = torch.nn.CrossEntropyLoss()

    def step():
        opt.zero_grad()
        loss = crit(model(x).float(), y)
        with amp.scale_loss(loss, opt) as scaled:
            scaled.backward()
        opt.step()

    for _ in range(5):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        step()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / iters * 1000
    print(f"syncbn={int(use_syncbn)} channels_last={int(channels_last)}: "
          f"{ms:8.2f} ms/step  {batch/ms*1000:8.0f} img/s")


if __name__ == "__main__":
    run(True, False)
    run(True, True)
    run(False, False)
    run(False, True)
