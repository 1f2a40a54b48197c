"""OpenFold acceleration kernels (HIP counterparts of the reference's
apex/contrib/openfold_triton — Triton is not part of the MI355X stack, so
the same capabilities are served by the library's HIP kernels):

- LayerNormSmallShapeOptImpl -> the wave64 fused_norm kernels
- AttnTri (MHA)              -> scaled_masked_softmax + GEMM composition
- FusedAdamSWA               -> fused Adam step + stochastic weight averaging
"""

from .fused_adam_swa import FusedAdamSWA
from .layer_norm import LayerNormSmallShapeOptImpl
from .mha import AttnBiasJIT, AttnNoBiasJIT, AttnTri, CanSchTriMHA

__all__ = ["FusedAdamSWA", "LayerNormSmallShapeOptImpl", "AttnTri",
           "AttnBiasJIT", "AttnNoBiasJIT", "CanSchTriMHA"]
