"""torchsched multi-stream execution on a real GPU: fork/join graphs must
produce eager-identical results through the side-stream/event path."""

import torch
import torch.fx
import pytest

pytestmark = pytest.mark.gpu


def test_multistream_fork_join_gpu():
    from apex_amd.contrib.torchsched.scheduler import MultiStreamGraphModule

    class Wide(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.stem = torch.nn.Linear(256, 256)
            self.branches = torch.nn.ModuleList(
                [torch.nn.Linear(256, 256) for _ in range(4)])

        def forward(self, x):
            y = torch.relu(self.stem(x))
            outs = [torch.tanh(b(y)) for b in self.branches]
            return outs[0] + outs[1] + outs[2] + outs[3]

    torch.manual_seed(0)
    m = Wide().cuda()
    gm = torch.fx.symbolic_trace(m)
    ms = MultiStreamGraphModule(gm)
    x = torch.randn(64, 256, device="cuda")
    for _ in range(3):  # repeated calls re-drive stream assignment
        out = ms(x)
        torch.cuda.synchronize()
        torch.testing.assert_close(out, m(x), rtol=1e-5, atol=1e-5)


def test_torchsched_compile_backend_gpu():
    import apex_amd.contrib.torchsched  # registers the backend

    class Net(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.a = torch.nn.Linear(128, 128)
            self.b = torch.nn.Linear(128, 128)

        def forward(self, x):
            h = torch.relu(x)
            return torch.tanh(self.a(h)) + torch.sigmoid(self.b(h))

    m = Net().cuda()
    cm = torch.compile(m, backend="torchsched")
    x = torch.randn(32, 128, device="cuda")
    torch.testing.assert_close(cm(x), m(x), rtol=1e-5, atol=1e-5)


def test_compiled_partitions_gpu():
    from apex_amd.contrib.torchsched.scheduler import MultiStreamGraphModule

    class FJ(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.a = torch.nn.Linear(128, 128)
            self.b = torch.nn.Linear(128, 128)

        def forward(self, x):
            h = torch.relu(x)
            return torch.tanh(self.a(h)) + torch.sigmoid(self.b(h))

    torch.manual_seed(2)
    m = FJ().cuda()
    gm = torch.fx.symbolic_trace(m)
    ms = MultiStreamGraphModule(gm, compile_partitions=True)
    x = torch.randn(16, 128, device="cuda")
    out = ms(x)
    torch.cuda.synchronize()
    torch.testing.assert_close(out, m(x), rtol=1e-4, atol=1e-4)
