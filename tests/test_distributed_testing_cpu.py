"""apex_amd.distributed_testing harness smoke: the GlooDistributedTestBase
actually spawns ranks, forms the process group, and propagates child
failures to the parent (reference:
apex/distributed_testing/distributed_test_base.py:24-120)."""

import unittest

import torch
import torch.distributed as dist

from apex_amd.distributed_testing import GlooDistributedTestBase


class _AllReduceCase(GlooDistributedTestBase):
    __test__ = False  # driven via _run_case, not collected directly

    def test_allreduce_sum(self):
        self._create_process_group()
        t = torch.tensor([float(self.rank + 1)])
        dist.all_reduce(t)
        expect = sum(range(1, self.world_size + 1))
        assert float(t) == expect, (float(t), expect)
        dist.destroy_process_group()


class _FailingCase(GlooDistributedTestBase):
    __test__ = False  # driven via _run_case, not collected directly

    def test_child_fails(self):
        self._create_process_group()
        dist.destroy_process_group()
        assert False, "deliberate child failure"


def _run_case(cls, name):
    result = unittest.TestResult()
    cls(name).run(result)
    return result


def test_gloo_base_runs_multiprocess_allreduce():
    result = _run_case(_AllReduceCase, "test_allreduce_sum")
    assert result.wasSuccessful(), (result.errors, result.failures)


def test_gloo_base_propagates_child_failure():
    result = _run_case(_FailingCase, "test_child_fails")
    assert not result.wasSuccessful()
