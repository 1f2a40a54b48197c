"""Shared test helpers."""

import os
import tempfile

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _dist_worker(rank, world_size, backend, file_name, fn, args):
    dist.init_process_group(
        backend=backend,
        init_method=f"file://{file_name}",
        world_size=world_size,
        rank=rank,
    )
    if backend == "nccl":
        torch.cuda.set_device(rank)
    try:
        fn(rank, world_size, *args)
    finally:
        dist.destroy_process_group()


def run_distributed(fn, world_size=2, backend="gloo", args=()):
    """Spawn `world_size` processes running fn(rank, world_size, *args)."""
    with tempfile.NamedTemporaryFile(delete=False) as f:
        file_name = f.name
    os.unlink(file_name)
    mp.spawn(
        _dist_worker,
        args=(world_size, backend, file_name, fn, args),
        nprocs=world_size,
        join=True,
    )
