"""torchsched partitioner + multi-stream executor (CPU: partition logic and
sequential-fallback correctness; stream machinery is covered by the gpu
tier)."""

import torch
import torch.fx

from apex_amd.contrib.torchsched.scheduler import (
    MultiStreamGraphModule, max_parallel_width, partition_graph,
)


class ForkJoin(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.a = torch.nn.Linear(16, 16)
        self.b = torch.nn.Linear(16, 16)
        self.c = torch.nn.Linear(16, 16)

    def forward(self, x):
        y = torch.relu(self.a(x))     # shared producer
        p = torch.tanh(self.b(y))     # branch 1
        q = torch.sigmoid(self.c(y))  # branch 2
        return p + q                  # join


def test_partition_fork_join():
    gm = torch.fx.symbolic_trace(ForkJoin())
    parts, part_of = partition_graph(gm)
    # chain(a,relu) | chain(b,tanh) | chain(c,sigmoid) | join(add)
    assert len(parts) == 4
    assert max_parallel_width(parts) == 2  # the two branches
    join = parts[-1]
    assert len(join.deps) == 2


def test_partition_straight_chain_stays_single():
    class Chain(torch.nn.Module):
        def forward(self, x):
            return torch.relu(x).tanh().sigmoid()

    gm = torch.fx.symbolic_trace(Chain())
    parts, _ = partition_graph(gm)
    assert len(parts) == 1
    assert max_parallel_width(parts) == 1


def test_multistream_module_cpu_matches_eager():
    torch.manual_seed(0)
    m = ForkJoin()
    gm = torch.fx.symbolic_trace(m)
    ms = MultiStreamGraphModule(gm)
    x = torch.randn(4, 16)
    torch.testing.assert_close(ms(x), m(x))


def test_backend_registered():
    import torch._dynamo as dynamo

    from apex_amd.contrib import torchsched

    assert callable(torchsched.get_backend())
    assert "torchsched" in dynamo.list_backends()


def test_compiled_partitions_cpu_matches_eager():
    torch.manual_seed(1)
    m = ForkJoin()
    gm = torch.fx.symbolic_trace(m)
    ms = MultiStreamGraphModule(gm, compile_partitions=True)
    x = torch.randn(4, 16)
    torch.testing.assert_close(ms(x), m(x), rtol=1e-5, atol=1e-6)
    # second call reuses the compiled callables
    x2 = torch.randn(4, 16)
    torch.testing.assert_close(ms(x2), m(x2), rtol=1e-5, atol=1e-6)


def test_lifetime_freeing_preserves_shared_values():
    """A value consumed by several partitions AND returned as an output must
    survive until its last use; earlier frees must not corrupt results."""

    class Shared(torch.nn.Module):
        def forward(self, x):
            y = torch.relu(x)          # shared producer, also returned
            a = torch.tanh(y)
            b = torch.sigmoid(y)
            return a + b, y

    torch.manual_seed(2)
    m = Shared()
    gm = torch.fx.symbolic_trace(m)
    ms = MultiStreamGraphModule(gm)
    x = torch.randn(4, 8)
    out = ms(x)
    ref = m(x)
    torch.testing.assert_close(out[0], ref[0])
    torch.testing.assert_close(out[1], ref[1])
