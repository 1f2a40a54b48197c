"""Intra-node HIP-IPC peer memory pool.

API parity with the reference ``apex.contrib.peer_memory.PeerMemoryPool``
(apex/contrib/peer_memory/peer_memory.py:6-41): a static and a dynamic
region carved from one hipMalloc'd slab per rank; IPC handles exchanged via
``torch.distributed.all_gather``; ``allocate_peer_tensors`` returns one view
per peer over xGMI-reachable memory.
"""

import torch
import torch.distributed as dist

from ..._ext import get_ext


class PeerMemoryPool:
    def __init__(self, static_size, dynamic_size, peer_ranks=None):
        ext = get_ext("peer_memory")
        self.alignment = 256
        self.static_size = (static_size + self.alignment - 1) // self.alignment * self.alignment
        self.dynamic_size = (dynamic_size + self.alignment - 1) // self.alignment * self.alignment

        rank = dist.get_rank()
        world_size = dist.get_world_size()
        self.peer_ranks = peer_ranks if peer_ranks is not None else list(range(world_size))
        self.rank = rank

        total = self.static_size + self.dynamic_size
        self.raw = ext.allocate_raw(total)

        # exchange IPC handles
        handle = ext.get_raw_ipc_address(self.raw)  # cpu byte tensor
        handles = [torch.empty_like(handle) for _ in range(world_size)]
        dist.all_gather(handles, handle)
        ipc = torch.stack(handles)  # [world, bytes]
        self.raw_peers = ext.get_raw_peers(ipc, rank, self.raw)

        self.static_offset = 0
        self.dynamic_offset = 0

    def __del__(self):
        try:
            ext = get_ext("peer_memory")
            if ext is not None and self.raw:
                ext.free_raw(self.raw)
        except Exception:
            pass

    def reset(self):
        self.dynamic_offset = 0

    def allocate_peer_tensors(self, shape, dtype, channels_last, dynamic):
        ext = get_ext("peer_memory")
        nels = 1
        for s in shape:
            nels *= s
        viewers = {
            torch.float16: (ext.blob_view_half, 2),
            torch.float32: (ext.blob_view_float, 4),
            torch.bfloat16: (ext.blob_view_bfloat16, 2),
            torch.int32: (ext.blob_view_int, 4),
        }
        if dtype not in viewers:
            raise RuntimeError(f"PeerMemoryPool: unsupported dtype {dtype}")
        view_fn, esize = viewers[dtype]
        nbytes = (nels * esize + self.alignment - 1) // self.alignment * self.alignment
        if dynamic:
            start = self.static_size + self.dynamic_offset
            self.dynamic_offset += nbytes
            assert self.dynamic_offset <= self.dynamic_size, "peer memory dynamic pool exhausted"
        else:
            start = self.static_offset
            self.static_offset += nbytes
            assert self.static_offset <= self.static_size, "peer memory static pool exhausted"
        return [view_fn(self.raw_peers[r] + start, list(shape), channels_last)
                for r in self.peer_ranks]
