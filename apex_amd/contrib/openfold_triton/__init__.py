"""Alias for :mod:`apex_amd.contrib.openfold` under the reference's module
name (apex/contrib/openfold_triton/__init__.py:24-38) so reference import
paths work verbatim. Triton is not part of the MI355X stack — the same
capabilities run on the library's HIP kernels (see contrib/openfold)."""

from ..openfold import (
    AttnBiasJIT,
    AttnNoBiasJIT,
    AttnTri,
    CanSchTriMHA,
    FusedAdamSWA,
    LayerNormSmallShapeOptImpl,
)
from ..openfold.mha import disable, enable

__all__ = [
    "FusedAdamSWA",
    "LayerNormSmallShapeOptImpl",
    "AttnTri",
    "AttnBiasJIT",
    "AttnNoBiasJIT",
    "CanSchTriMHA",
    "enable",
    "disable",
]
