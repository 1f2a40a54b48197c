from .distributed import DistributedDataParallel, Reducer, flat_dist_call
from .sync_batchnorm import SyncBatchNorm, convert_syncbn_model, create_syncbn_process_group

__all__ = [
    "DistributedDataParallel",
    "Reducer",
    "SyncBatchNorm",
    "convert_syncbn_model",
    "create_syncbn_process_group",
    "flat_dist_call",
]
