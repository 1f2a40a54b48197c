"""Raw RCCL p2p communicator (the reference's contrib.nccl_p2p over RCCL)."""

from ..._ext import get_ext


def get_unique_nccl_id(n=1):
    return get_ext("rccl_p2p").get_unique_nccl_id(n)


def init_nccl_comm(unique_nccl_id, my_rank, num_ranks):
    return get_ext("rccl_p2p").init_nccl_comm(unique_nccl_id, my_rank, num_ranks)


def left_right_halo_exchange(handle, left_rank, right_rank, left_output_halo, right_output_halo):
    return get_ext("rccl_p2p").left_right_halo_exchange(
        handle, left_rank, right_rank, left_output_halo, right_output_halo
    )


def left_right_halo_exchange_inplace(handle, left_rank, right_rank, left_output_halo,
                                     right_output_halo, left_input_halo, right_input_halo):
    return get_ext("rccl_p2p").left_right_halo_exchange_inplace(
        handle, left_rank, right_rank, left_output_halo, right_output_halo,
        left_input_halo, right_input_halo,
    )


__all__ = [
    "get_unique_nccl_id",
    "init_nccl_comm",
    "left_right_halo_exchange",
    "left_right_halo_exchange_inplace",
]
