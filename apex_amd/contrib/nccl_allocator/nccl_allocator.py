"""RCCL-registered memory pool (the reference's nccl_allocator over RCCL).

API parity with apex/contrib/nccl_allocator/nccl_allocator.py:18-75:
``init()``, ``create_nccl_mem_pool()``, and the ``nccl_mem`` context manager
that routes allocations inside it through ncclMemAlloc so RCCL can register
the buffers for zero-copy collectives. (NVLS itself has no xGMI analogue;
RCCL's user-buffer registration is the MI355X benefit.)
"""

import contextlib

import torch

from ..._ext import get_ext

_allocator = None


def init():
    global _allocator
    if _allocator is None:
        ext = get_ext("rccl_allocator")
        _allocator = ext.get_rccl_allocator()
    return _allocator


def create_nccl_mem_pool(symmetric=False):
    allocator = init()
    return torch.cuda.MemPool(allocator, symmetric=symmetric)


@contextlib.contextmanager
def nccl_mem(pool, enabled=True, device=None, group=None):
    if not enabled:
        yield
        return
    dev = device if device is not None else torch.cuda.current_device()
    with torch.cuda.use_mem_pool(pool, device=dev):
        yield
