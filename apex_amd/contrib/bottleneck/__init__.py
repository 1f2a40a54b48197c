from .bottleneck import Bottleneck, SpatialBottleneck, FrozenBatchNorm2d
from .halo_exchangers import (
    HaloExchanger,
    HaloExchangerNoComm,
    HaloExchangerAllGather,
    HaloExchangerSendRecv,
    HaloExchangerPeer,
)

__all__ = [
    "Bottleneck",
    "SpatialBottleneck",
    "FrozenBatchNorm2d",
    "HaloExchanger",
    "HaloExchangerNoComm",
    "HaloExchangerAllGather",
    "HaloExchangerSendRecv",
    "HaloExchangerPeer",
]
