"""CPU tests for conv_bias_relu, bottleneck, halo exchangers (gloo), gbn."""

import torch
import torch.distributed as dist
import pytest

from utils import run_distributed


def test_conv_bias_relu_matches_torch():
    from apex_amd.contrib.conv_bias_relu import ConvBiasReLU, ConvBias

    torch.manual_seed(0)
    x = torch.randn(2, 8, 9, 9, requires_grad=True)
    w = torch.randn(16, 8, 3, 3, requires_grad=True)
    b = torch.randn(1, 16, 1, 1, requires_grad=True)
    y = ConvBiasReLU(x, w, b, 1, 1)
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    ref = torch.relu(torch.nn.functional.conv2d(xr, wr, br.reshape(-1), 1, 1))
    torch.testing.assert_close(y, ref)
    g = torch.randn_like(y)
    y.backward(g)
    ref.backward(g)
    torch.testing.assert_close(x.grad, xr.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(w.grad, wr.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(b.grad.reshape(-1), br.grad.reshape(-1), rtol=1e-4, atol=1e-5)

    y2 = ConvBias(x.detach(), w.detach(), b.detach(), 1, 1)
    ref2 = torch.nn.functional.conv2d(x.detach(), w.detach(), b.detach().reshape(-1), 1, 1)
    torch.testing.assert_close(y2, ref2)


def test_conv_frozen_scale_bias_relu():
    from apex_amd.contrib.conv_bias_relu import ConvFrozenScaleBiasReLU

    torch.manual_seed(1)
    x = torch.randn(2, 4, 7, 7, requires_grad=True)
    w = torch.randn(8, 4, 3, 3, requires_grad=True)
    s = torch.randn(1, 8, 1, 1).abs() + 0.1
    b = torch.randn(1, 8, 1, 1)
    y = ConvFrozenScaleBiasReLU(x, w, s, b, 1, 1)
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    ref = torch.relu(torch.nn.functional.conv2d(xr, wr, None, 1, 1) * s + b)
    torch.testing.assert_close(y, ref)
    g = torch.randn_like(y)
    y.backward(g)
    ref.backward(g)
    torch.testing.assert_close(x.grad, xr.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(w.grad, wr.grad, rtol=1e-4, atol=1e-5)


def test_bottleneck_forward_shapes():
    from apex_amd.contrib.bottleneck import Bottleneck

    torch.manual_seed(2)
    block = Bottleneck(64, 32, 128, stride=2)
    x = torch.randn(2, 64, 16, 16)
    y = block(x)
    assert y.shape == (2, 128, 8, 8)
    assert (y >= 0).all()  # final relu

    block2 = Bottleneck(64, 32, 64, stride=1)
    y2 = block2(torch.randn(2, 64, 8, 8))
    assert y2.shape == (2, 64, 8, 8)


def _halo_worker(rank, world_size, kind):
    from apex_amd.contrib.bottleneck import (
        HaloExchangerAllGather,
        HaloExchangerSendRecv,
    )

    cls = {"allgather": HaloExchangerAllGather, "sendrecv": HaloExchangerSendRecv}[kind]
    hx = cls(list(range(world_size)), rank)
    left_out = torch.full((1, 2, 3), float(rank * 10 + 1))
    right_out = torch.full((1, 2, 3), float(rank * 10 + 2))
    left_in, right_in = hx.left_right_halo_exchange(left_out, right_out)
    left_peer = (rank - 1) % world_size
    right_peer = (rank + 1) % world_size
    torch.testing.assert_close(left_in, torch.full_like(left_in, float(left_peer * 10 + 2)))
    torch.testing.assert_close(right_in, torch.full_like(right_in, float(right_peer * 10 + 1)))


@pytest.mark.parametrize("kind", ["allgather", "sendrecv"])
def test_halo_exchangers_gloo(kind):
    run_distributed(_halo_worker, world_size=2, args=(kind,))


def _spatial_worker(rank, world_size, method=1):
    from apex_amd.contrib.bottleneck import (
        Bottleneck, SpatialBottleneck, HaloExchangerAllGather,
    )

    torch.manual_seed(3)
    H = 16
    full = Bottleneck(8, 4, 8, stride=1)
    x_full = torch.randn(2, 8, H, 8)
    # broadcast so both ranks share weights and input
    for p in full.state_dict().values():
        dist.broadcast(p, 0)
    dist.broadcast(x_full, 0)
    y_full = full(x_full)

    hx = HaloExchangerAllGather(list(range(world_size)), rank)
    sp = SpatialBottleneck(8, 4, 8, stride=1,
                           spatial_parallel_args=(world_size, rank, None, hx, method, False))
    sp.load_state_dict(full.state_dict())
    h_local = H // world_size
    x_local = x_full[:, :, rank * h_local:(rank + 1) * h_local, :].contiguous()
    y_local = sp(x_local)
    expected = y_full[:, :, rank * h_local:(rank + 1) * h_local, :]
    torch.testing.assert_close(y_local, expected, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("method", [1, 2, 3])
def test_spatial_bottleneck_matches_full(method):
    run_distributed(_spatial_worker, world_size=2, args=(method,))


def test_gbn_single_rank_matches_bn():
    from apex_amd.contrib.gbn import GroupBatchNorm2d

    torch.manual_seed(4)
    gbn = GroupBatchNorm2d(8)
    bn = torch.nn.BatchNorm2d(8)
    gbn.train()
    bn.train()
    with torch.no_grad():
        bn.weight.copy_(gbn.weight)
        bn.bias.copy_(gbn.bias)
    x = torch.randn(4, 8, 5, 5)
    torch.testing.assert_close(gbn(x), bn(x), rtol=1e-5, atol=1e-6)


def _spatial_stride2_worker(rank, world_size, method):
    # stride 2: method 1 splices halos with explicit boundary rows; methods
    # 2/3 correct only the top edge row (the bottom field is fully local)
    from apex_amd.contrib.bottleneck import (
        Bottleneck, SpatialBottleneck, HaloExchangerAllGather,
    )

    torch.manual_seed(5)
    H = 16
    full = Bottleneck(8, 4, 16, stride=2)
    x_full = torch.randn(2, 8, H, 8)
    for p in full.state_dict().values():
        dist.broadcast(p, 0)
    dist.broadcast(x_full, 0)
    y_full = full(x_full)

    hx = HaloExchangerAllGather(list(range(world_size)), rank)
    sp = SpatialBottleneck(8, 4, 16, stride=2,
                           spatial_parallel_args=(world_size, rank, None, hx, method, False))
    sp.load_state_dict(full.state_dict())
    h_local = H // world_size
    x_local = x_full[:, :, rank * h_local:(rank + 1) * h_local, :].contiguous()
    y_local = sp(x_local)
    h_out = y_full.shape[2] // world_size
    expected = y_full[:, :, rank * h_out:(rank + 1) * h_out, :]
    torch.testing.assert_close(y_local, expected, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("method", [1, 2, 3])
def test_spatial_bottleneck_stride2(method):
    run_distributed(_spatial_stride2_worker, world_size=2, args=(method,))
