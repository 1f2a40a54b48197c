"""1-D halo exchange over peer (IPC) memory — the xGMI direct-store path.

Reference surface: apex/contrib/peer_memory/peer_memory.py
(PeerHaloExchanger1d): splits a conv activation along H across ranks and
exchanges ``half_halo`` edge rows with the two neighbours by storing straight
into their IPC-mapped buffers, then synchronizing the group.
"""

import torch
import torch.distributed as dist

from ..._ext import get_ext


class PeerHaloExchanger1d:
    def __init__(self, ranks, rank_in_group, peer_pool, half_halo):
        self.peer_group_size = len(ranks)
        self.ranks = ranks
        self.peer_rank = rank_in_group
        self.low_neighbor = (self.peer_rank + self.peer_group_size - 1) % self.peer_group_size
        self.high_neighbor = (self.peer_rank + 1) % self.peer_group_size
        self.low_zero = self.peer_rank == 0
        self.high_zero = self.peer_rank == self.peer_group_size - 1
        self.pool = peer_pool
        self.half_halo = half_halo

    def _allocate_buffers(self, halo_shape, dtype, channels_last):
        # recv buffers on every peer; index by this rank to find "mine"
        self.top_in = self.pool.allocate_peer_tensors(halo_shape, dtype, channels_last, True)
        self.btm_in = self.pool.allocate_peer_tensors(halo_shape, dtype, channels_last, True)

    def __call__(self, y, half_halo=None, explicit_nhwc=False, numSM=0, diagnostics=False):
        """y: [N, C, H, W] (or NHWC with explicit_nhwc); pad with half_halo
        rows on each side; fills the padding from the neighbours."""
        ext = get_ext("peer_memory")
        hh = half_halo if half_halo is not None else self.half_halo
        h_dim = 1 if explicit_nhwc else 2
        H = y.shape[h_dim]
        assert H > 2 * hh, "tensor too small for halo exchange"

        top_out = y.narrow(h_dim, hh, hh).contiguous()
        btm_out = y.narrow(h_dim, H - 2 * hh, hh).contiguous()
        halo_shape = list(top_out.shape)
        self.pool.reset()
        self._allocate_buffers(halo_shape, y.dtype, False)

        # store my edges into the neighbours' recv buffers over xGMI
        ext.push_pull_halos_1d(
            top_out, btm_out,
            self.top_in[self.low_neighbor].data_ptr(),
            self.btm_in[self.high_neighbor].data_ptr(),
        )
        torch.cuda.synchronize()
        dist.barrier()

        mine_top = self.btm_in[self.peer_rank]  # filled by my low neighbor's bottom
        mine_btm = self.top_in[self.peer_rank]  # filled by my high neighbor's top
        if not self.low_zero:
            y.narrow(h_dim, 0, hh).copy_(mine_top)
        if not self.high_zero:
            y.narrow(h_dim, H - hh, hh).copy_(mine_btm)
        return y
