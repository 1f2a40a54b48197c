from .distributed_fused_adam import DistributedFusedAdam
from .distributed_fused_lamb import DistributedFusedLAMB
from .fp16_optimizer import FP16_Optimizer

# The reference keeps deprecated copies of the core fused optimizers under
# contrib.optimizers (apex/contrib/optimizers/fused_adam.py etc.); here they
# are the same gfx950 implementations.
from ...optimizers import FusedAdam, FusedLAMB, FusedSGD  # noqa: F401

__all__ = [
    "DistributedFusedAdam",
    "DistributedFusedLAMB",
    "FP16_Optimizer",
    "FusedAdam",
    "FusedLAMB",
    "FusedSGD",
]
