"""Alias for :mod:`apex_amd.contrib.gbn` under the reference's module name
(apex/contrib/cudnn_gbn/__init__.py:1) so reference import paths work
verbatim. There is no cuDNN on ROCm — the implementation runs on the
library's Welford/syncbn HIP kernels with RCCL stat exchange."""

from ..gbn import GroupBatchNorm2d

__all__ = ["GroupBatchNorm2d"]
