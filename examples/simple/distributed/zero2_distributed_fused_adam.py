"""ZeRO-2 training with DistributedFusedAdam.

Run (one process per GPU, RCCL over xGMI):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 \
        examples/simple/distributed/zero2_distributed_fused_adam.py

Shows the memory-lean configuration: bf16 model, implicit fp32 masters
(``store_param_remainders``), fp16 scaled moments (``with_scaled_states``),
overlapped gradient reduce-scatter and lazy per-bucket param all-gathers.
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", "..", ".."))

import torch
import torch.distributed as dist


def main():
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0"))) \
        if torch.cuda.is_available() else torch.device("cpu")
    if device.type == "cuda":
        torch.cuda.set_device(device)

    from apex_amd.contrib.optimizers import DistributedFusedAdam
    from apex_amd.parallel import DistributedDataParallel as DDP

    torch.manual_seed(0)
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    model = torch.nn.Sequential(
        torch.nn.Linear(1024, 4096), torch.nn.GELU(),
        torch.nn.Linear(4096, 1024),
    ).to(device=device, dtype=dtype)

    opt = DistributedFusedAdam(
        model.parameters(), lr=1e-3, weight_decay=0.01,
        bucket_cap_mb=64,
        overlap_grad_sync=True,
        overlap_param_sync=True,
        store_param_remainders=(dtype == torch.bfloat16),
        with_scaled_states=False,  # flip on to halve moment memory
    )
    opt.register_model_for_param_sync(model)

    # NOTE: grads are reduced by the optimizer's own hooks (ZeRO-2), so the
    # model is NOT wrapped in DDP — DDP would all-reduce a second time.
    _ = DDP  # imported to show the alternative exists

    for step in range(10):
        x = torch.randn(32, 1024, device=device, dtype=dtype)
        loss = model(x).float().pow(2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        if rank == 0 and step % 2 == 0:
            print(f"step {step}  loss {float(loss):.6f}")

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
