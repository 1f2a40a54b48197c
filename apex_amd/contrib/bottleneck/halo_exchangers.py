"""Halo exchangers for spatially-parallel convolutions.

API parity with the reference apex/contrib/bottleneck/halo_exchangers.py:
10-165 — four implementations with the same
``left_right_halo_exchange(left_output_halo, right_output_halo)`` contract:

- HaloExchangerNoComm   : single-rank no-op (returns the peers' own halos)
- HaloExchangerAllGather: all_gather of the halo slabs over the group
- HaloExchangerSendRecv : point-to-point isend/irecv pairs (RCCL over xGMI)
- HaloExchangerPeer     : direct stores into IPC-mapped peer memory (the
                          xGMI-native path, via PeerMemoryPool)
"""

import torch
import torch.distributed as dist


class HaloExchanger:
    def __init__(self, ranks, rank_in_group):
        self.stream1 = torch.cuda.Stream() if torch.cuda.is_available() else None
        self.stream2 = torch.cuda.Stream() if torch.cuda.is_available() else None
        self.group_size = len(ranks)
        self.ranks = ranks
        self.rank_in_group = rank_in_group
        self.wrap_around_left_rank_in_group = (rank_in_group + self.group_size - 1) % self.group_size
        self.wrap_around_right_rank_in_group = (rank_in_group + 1) % self.group_size
        self.left_rank = ranks[rank_in_group - 1] if rank_in_group > 0 else -1
        self.left_zero = rank_in_group == 0
        self.right_rank = ranks[rank_in_group + 1] if rank_in_group < self.group_size - 1 else -1
        self.right_zero = rank_in_group == self.group_size - 1


class HaloExchangerNoComm(HaloExchanger):
    def __init__(self, ranks, rank_in_group):
        super().__init__(ranks, rank_in_group)

    def left_right_halo_exchange(self, left_output_halo, right_output_halo,
                                 left_input_halo=None, right_input_halo=None):
        if left_input_halo is None:
            return right_output_halo, left_output_halo
        left_input_halo.copy_(right_output_halo)
        right_input_halo.copy_(left_output_halo)


class HaloExchangerAllGather(HaloExchanger):
    def __init__(self, ranks, rank_in_group, comm=None):
        super().__init__(ranks, rank_in_group)
        self.comm = comm  # process group

    def left_right_halo_exchange(self, left_output_halo, right_output_halo,
                                 left_input_halo=None, right_input_halo=None):
        send = torch.cat([left_output_halo.flatten(), right_output_halo.flatten()])
        gathered = [torch.empty_like(send) for _ in range(self.group_size)]
        dist.all_gather(gathered, send, group=self.comm)
        n = left_output_halo.numel()
        wrap_l = gathered[self.wrap_around_left_rank_in_group][n:].view_as(right_output_halo)
        wrap_r = gathered[self.wrap_around_right_rank_in_group][:n].view_as(left_output_halo)
        if left_input_halo is None:
            return wrap_l, wrap_r
        left_input_halo.copy_(wrap_l)
        right_input_halo.copy_(wrap_r)


class HaloExchangerSendRecv(HaloExchanger):
    def __init__(self, ranks, rank_in_group, comm=None):
        super().__init__(ranks, rank_in_group)
        self.comm = comm

    def left_right_halo_exchange(self, left_output_halo, right_output_halo,
                                 left_input_halo=None, right_input_halo=None):
        ret = left_input_halo is None
        if ret:
            left_input_halo = torch.empty_like(right_output_halo)
            right_input_halo = torch.empty_like(left_output_halo)
        lg = self.ranks[self.wrap_around_left_rank_in_group]
        rg = self.ranks[self.wrap_around_right_rank_in_group]
        # left_input receives the LEFT peer's right edge; ordering matters
        # when both peers are the same rank (group size 2): sends and recvs
        # must pair up in matching program order on both sides.
        ops = [
            dist.P2POp(dist.isend, right_output_halo.contiguous(), rg, group=self.comm),
            dist.P2POp(dist.irecv, left_input_halo, lg, group=self.comm),
            dist.P2POp(dist.isend, left_output_halo.contiguous(), lg, group=self.comm),
            dist.P2POp(dist.irecv, right_input_halo, rg, group=self.comm),
        ]
        for w in dist.batch_isend_irecv(ops):
            w.wait()
        if ret:
            return left_input_halo, right_input_halo


class HaloExchangerPeer(HaloExchanger):
    def __init__(self, ranks, rank_in_group, peer_pool, explicit_nhwc=False, numSM=0):
        super().__init__(ranks, rank_in_group)
        self.peer_pool = peer_pool
        self.explicit_nhwc = explicit_nhwc
        self.numSM = numSM

    def left_right_halo_exchange(self, left_output_halo, right_output_halo,
                                 left_input_halo=None, right_input_halo=None):
        from ..._ext import get_ext

        ext = get_ext("peer_memory")
        ret = left_input_halo is None
        shape = list(left_output_halo.shape)
        self.peer_pool.reset()
        left_in = self.peer_pool.allocate_peer_tensors(shape, left_output_halo.dtype, False, True)
        right_in = self.peer_pool.allocate_peer_tensors(shape, left_output_halo.dtype, False, True)
        # write into the neighbours' buffers over xGMI
        ext.push_pull_halos_1d(
            left_output_halo.contiguous(), right_output_halo.contiguous(),
            right_in[self.wrap_around_left_rank_in_group].data_ptr(),
            left_in[self.wrap_around_right_rank_in_group].data_ptr(),
        )
        torch.cuda.synchronize()
        dist.barrier()
        mine_left = left_in[self.rank_in_group].clone()
        mine_right = right_in[self.rank_in_group].clone()
        if ret:
            return mine_left, mine_right
        left_input_halo.copy_(mine_left)
        right_input_halo.copy_(mine_right)
