from .group_norm import GroupNorm, cuda_group_norm_nhwc_forward

__all__ = ["GroupNorm", "cuda_group_norm_nhwc_forward"]
