from .nccl_allocator import init, create_nccl_mem_pool, nccl_mem

__all__ = ["init", "create_nccl_mem_pool", "nccl_mem"]
