"""Minimal apex_amd DDP example (reference: examples/simple/distributed).

    python -m torch.distributed.run --nproc-per-node 2 --master-addr 127.0.0.1 \
        examples/simple/distributed/distributed_data_parallel.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..", ".."))

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()
    dist.init_process_group(backend="nccl" if use_cuda else "gloo")
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    from apex_amd import amp
    from apex_amd.optimizers import FusedAdam
    from apex_amd.parallel import DistributedDataParallel as DDP

    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(64, 256), torch.nn.ReLU(),
                                torch.nn.Linear(256, 16)).to(device)
    opt = FusedAdam(model.parameters(), lr=1e-3)
    model, opt = amp.initialize(model, opt, opt_level="O1",
                                cast_model_type=torch.bfloat16, loss_scale=1.0, verbosity=0)
    model = DDP(model, message_size=1_000_000, num_allreduce_streams=2)

    torch.manual_seed(rank)
    x = torch.randn(32, 64, device=device)
    y = torch.randn(32, 16, device=device)
    for it in range(10):
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x).float(), y)
        with amp.scale_loss(loss, opt) as scaled:
            scaled.backward()
        opt.step()
        if rank == 0 and it % 3 == 0:
            print(f"iter {it} loss {float(loss.detach()):.5f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
