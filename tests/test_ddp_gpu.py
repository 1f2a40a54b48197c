"""DDP GPU-path test (world_size=1 over RCCL): exercises the side-stream
bucketed all-reduce machinery on device — hooks, flatten, comm streams,
copy-back — against a no-DDP reference. Multi-rank correctness is covered by
the gloo tier; 8-GPU scaling is driver-side."""

import os
import tempfile

import torch
import torch.distributed as dist
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture()
def nccl_world1():
    if dist.is_initialized():
        dist.destroy_process_group()
    with tempfile.NamedTemporaryFile(delete=False) as f:
        name = f.name
    os.unlink(name)
    dist.init_process_group(backend="nccl", init_method=f"file://{name}", world_size=1, rank=0)
    yield
    dist.destroy_process_group()


def test_ddp_gpu_streams_grads_match(nccl_world1):
    from apex_amd.parallel import DistributedDataParallel as DDP

    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(64, 128), torch.nn.ReLU(),
                                torch.nn.Linear(128, 32)).cuda()
    ref = torch.nn.Sequential(torch.nn.Linear(64, 128), torch.nn.ReLU(),
                              torch.nn.Linear(128, 32)).cuda()
    ref.load_state_dict(model.state_dict())
    ddp = DDP(model, message_size=1, num_allreduce_streams=3,
              gradient_predivide_factor=2.0)

    for it in range(4):
        torch.manual_seed(10 + it)
        x = torch.randn(8, 64, device="cuda")
        ddp.zero_grad()
        ddp(x).pow(2).mean().backward()
        for p in ref.parameters():
            p.grad = None
        ref(x).pow(2).mean().backward()
        torch.cuda.synchronize()
        for p, rp in zip(ddp.module.parameters(), ref.parameters()):
            torch.testing.assert_close(p.grad, rp.grad, rtol=1e-5, atol=1e-6)


def test_ddp_gpu_delay_allreduce(nccl_world1):
    from apex_amd.parallel import DistributedDataParallel as DDP

    torch.manual_seed(1)
    model = torch.nn.Linear(32, 32).cuda()
    ddp = DDP(model, delay_allreduce=True)
    x = torch.randn(4, 32, device="cuda")
    ddp(x).sum().backward()
    torch.cuda.synchronize()
    assert all(p.grad is not None for p in ddp.module.parameters())
