from .fused_dense import (
    FusedDense,
    FusedDenseFunc,
    FusedDenseGeluDense,
    FusedDenseGeluDenseFunc,
    fused_dense_function,
    fused_dense_gelu_dense_function,
)

__all__ = [
    "FusedDense",
    "FusedDenseFunc",
    "FusedDenseGeluDense",
    "FusedDenseGeluDenseFunc",
    "fused_dense_function",
    "fused_dense_gelu_dense_function",
]
