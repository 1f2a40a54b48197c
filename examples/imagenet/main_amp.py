"""ImageNet-style AMP training harness (reference: examples/imagenet/main_amp.py
and tests/L1/common/main_amp.py — the determinism/throughput harness).

Synthetic-data edition (this environment has no dataset access): fixed seed,
``--deterministic`` mode, per-iteration loss + img/s reporting, opt-level
sweep compatible with the L1 cross-product runner:

    python -m torch.distributed.run --nproc-per-node N examples/imagenet/main_amp.py \
        --opt-level O1 --loss-scale dynamic --epochs 1 --iters 100
"""

import argparse
import json
import os
import time

os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")  # cold-box guard (see bench.py)

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--arch", default="resnet50")
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--weight-decay", type=float, default=1e-4)
    p.add_argument("--opt-level", default="O1", choices=["O0", "O1", "O2", "O3"])
    p.add_argument("--loss-scale", default=None)
    p.add_argument("--keep-batchnorm-fp32", default=None)
    p.add_argument("--cast-dtype", default="bf16", choices=["fp16", "bf16"])
    p.add_argument("--sync-bn", action="store_true")
    p.add_argument("--deterministic", action="store_true")
    p.add_argument("--seed", type=int, default=1)
    p.add_argument("--print-freq", type=int, default=10)
    p.add_argument("--json-out", default=None)
    return p.parse_args()


def main():
    args = parse_args()
    from apex_amd import amp
    from apex_amd.models import resnet50
    from apex_amd.optimizers import FusedSGD
    from apex_amd.parallel import DistributedDataParallel as DDP
    from apex_amd.parallel import convert_syncbn_model

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1
    if distributed:
        dist.init_process_group(backend="nccl")
        torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank) if torch.cuda.is_available() else torch.device("cpu")

    torch.manual_seed(args.seed)
    if args.deterministic:
        torch.use_deterministic_algorithms(True, warn_only=True)
        torch.backends.cudnn.deterministic = True
        torch.backends.cudnn.benchmark = False

    model = resnet50()
    if args.sync_bn:
        model = convert_syncbn_model(model)
    model = model.to(device)
    optimizer = FusedSGD(model.parameters(), lr=args.lr, momentum=args.momentum,
                         weight_decay=args.weight_decay)

    loss_scale = args.loss_scale
    if loss_scale is not None and loss_scale != "dynamic":
        loss_scale = float(loss_scale)
    cast_dtype = torch.bfloat16 if args.cast_dtype == "bf16" else torch.float16
    model, optimizer = amp.initialize(
        model, optimizer, opt_level=args.opt_level,
        cast_model_type=None if args.opt_level == "O0" else cast_dtype,
        keep_batchnorm_fp32=args.keep_batchnorm_fp32,
        loss_scale=loss_scale, verbosity=1 if rank == 0 else 0,
    )
    if distributed:
        model = DDP(model, message_size=16_000_000)

    criterion = torch.nn.CrossEntropyLoss().to(device)
    # fixed synthetic dataset: one batch per iteration, deterministic
    gen = torch.Generator().manual_seed(args.seed + rank)
    images = torch.randn(args.batch_size, 3, args.image_size, args.image_size, generator=gen)
    targets = torch.randint(0, 1000, (args.batch_size,), generator=gen)
    images = images.to(device)
    targets = targets.to(device)

    model.train()
    records = []
    t_start = time.perf_counter()
    for it in range(args.iters):
        t0 = time.perf_counter()
        optimizer.zero_grad()
        output = model(images)
        loss = criterion(output.float(), targets)
        with amp.scale_loss(loss, optimizer) as scaled_loss:
            scaled_loss.backward()
        optimizer.step()
        if device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        records.append({"iter": it, "loss": float(loss.detach()),
                        "imgs_per_s": world * args.batch_size / dt})
        if rank == 0 and it % args.print_freq == 0:
            print(f"iter {it:4d}  loss {records[-1]['loss']:.6f}  "
                  f"speed {records[-1]['imgs_per_s']:.1f} img/s")

    if rank == 0:
        total = time.perf_counter() - t_start
        summary = {
            "opt_level": args.opt_level,
            "loss_scale": args.loss_scale,
            "iters": args.iters,
            "imgs_per_s_avg": world * args.batch_size * args.iters / total,
            "records": records,
        }
        print(json.dumps({k: v for k, v in summary.items() if k != "records"}))
        if args.json_out:
            with open(args.json_out, "w") as f:
                json.dump(summary, f)
    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
