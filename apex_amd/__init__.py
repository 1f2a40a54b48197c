"""apex_amd — MI355X-native mixed-precision / fused-kernel training library.

A from-scratch CDNA4 (gfx950) implementation of the NVIDIA/apex feature set:
``amp`` (O0-O3 mixed precision with on-device dynamic loss scaling),
``optimizers`` (multi-tensor fused optimizers), ``normalization``
(FusedLayerNorm / FusedRMSNorm), ``parallel`` (bucketed-allreduce DDP and
Welford SyncBatchNorm over RCCL/xGMI), ``fused_dense`` / ``mlp`` (hipBLASLt
epilogue GEMMs), the Megatron softmax / RoPE / wgrad kernel family under
``transformer``, and ``contrib``.

Reference API surface: /root/reference (NVIDIA/apex), see SURVEY.md.
All device code is hand-written HIP for wave64 / MFMA / LDS on gfx950 —
no CUDA compatibility layer.
"""

import logging
import warnings

import torch

from . import optimizers  # noqa: F401
from . import normalization  # noqa: F401
from . import amp  # noqa: F401
from . import parallel  # noqa: F401
from . import fused_dense  # noqa: F401
from . import mlp  # noqa: F401
from . import multi_tensor_apply  # noqa: F401
from . import tracing  # noqa: F401
from . import transformer  # noqa: F401
from . import contrib  # noqa: F401

__all__ = [
    "amp",
    "optimizers",
    "normalization",
    "parallel",
    "fused_dense",
    "mlp",
    "multi_tensor_apply",
    "transformer",
    "contrib",
]

__version__ = "0.1.0"

logger = logging.getLogger(__name__)


def deprecated_warning(msg: str) -> None:
    """Rank-0-gated deprecation warning (reference: apex/__init__.py:37-43)."""
    if (
        torch.distributed.is_available()
        and torch.distributed.is_initialized()
        and torch.distributed.get_rank() != 0
    ):
        return
    warnings.warn(msg, FutureWarning, stacklevel=2)
