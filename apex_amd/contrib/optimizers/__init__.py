from .distributed_fused_adam import DistributedFusedAdam
from .distributed_fused_lamb import DistributedFusedLAMB
from .fp16_optimizer import FP16_Optimizer

__all__ = ["DistributedFusedAdam", "DistributedFusedLAMB", "FP16_Optimizer"]
