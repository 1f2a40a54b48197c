"""Multi-process distributed test base (reference:
apex/distributed_testing/distributed_test_base.py:24-120).

Spawn-based MultiProcessTestCase: world = min(device_count, 4) (or 2 on CPU),
file-store init, per-rank ``dist.init_process_group``. ``NcclDistributedTestBase``
requires GPUs (RCCL); ``GlooDistributedTestBase`` runs on CPU CI.
"""


import torch
import torch.distributed as dist
from torch.testing._internal.common_distributed import MultiProcessTestCase


class DistributedTestBase(MultiProcessTestCase):
    BACKEND = None

    def setUp(self):
        super().setUp()
        self._spawn_processes()

    def tearDown(self):
        torch.cuda.empty_cache() if torch.cuda.is_available() else None
        super().tearDown()

    @property
    def world_size(self):
        if torch.cuda.is_available():
            return min(torch.cuda.device_count(), 4)
        return 2

    @property
    def init_method(self):
        return f"file://{self.file_name}"

    def _create_process_group(self):
        dist.init_process_group(
            backend=self.BACKEND,
            init_method=self.init_method,
            world_size=self.world_size,
            rank=self.rank,
        )
        if self.BACKEND == "nccl":
            torch.cuda.set_device(self.rank)


class NcclDistributedTestBase(DistributedTestBase):
    BACKEND = "nccl"


class GlooDistributedTestBase(DistributedTestBase):
    BACKEND = "gloo"
