"""conv_bias_relu + Bottleneck on hardware (MIOpen convs with composed
epilogues — pure torch ops, numerics vs fp32 references)."""

import torch
import pytest

pytestmark = pytest.mark.gpu


def test_conv_bias_relu_gpu():
    from apex_amd.contrib.conv_bias_relu import ConvBiasReLU

    torch.manual_seed(0)
    x = torch.randn(4, 8, 16, 16, device="cuda", requires_grad=True)
    # w/b must be LEAF tensors or .grad stays None (round-1 bug: the `* 0.1`
    # made w a non-leaf and aborted the driver's --maxfail=1 GPU tier).
    w = (torch.randn(16, 8, 3, 3, device="cuda") * 0.1).requires_grad_()
    b = torch.randn(1, 16, 1, 1, device="cuda", requires_grad=True)
    y = ConvBiasReLU(x, w, b, 1, 1)
    ref = torch.relu(torch.nn.functional.conv2d(
        x.detach(), w.detach(), stride=1, padding=1) + b.detach())
    torch.testing.assert_close(y, ref, rtol=1e-4, atol=1e-4)
    y.sum().backward()
    assert x.grad is not None and w.grad is not None and b.grad is not None


def test_bottleneck_block_gpu():
    from apex_amd.contrib.bottleneck import Bottleneck

    torch.manual_seed(1)
    blk = Bottleneck(16, 8, 32, stride=2).cuda()
    x = torch.randn(2, 16, 16, 16, device="cuda", requires_grad=True)
    y = blk(x)
    assert y.shape == (2, 32, 8, 8)
    y.sum().backward()
    assert x.grad is not None
    assert torch.isfinite(y).all()


def test_frozen_bn_scale_bias_gpu():
    from apex_amd.contrib.bottleneck import FrozenBatchNorm2d

    bn = FrozenBatchNorm2d(8).cuda()
    bn.running_var.uniform_(0.5, 2.0)
    bn.running_mean.normal_()
    x = torch.randn(2, 8, 4, 4, device="cuda")
    ref = (x - bn.running_mean.view(1, -1, 1, 1)) / bn.running_var.view(1, -1, 1, 1).sqrt()
    torch.testing.assert_close(bn(x), ref, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("stride", [1, 2])
def test_conv1x1_bias_relu_fused_gpu(stride):
    """The 1x1 path runs as ONE hipBLASLt RELU_BIAS GEMM; numerics + grads
    must match the eager composition (incl. the stride-2 subsample)."""
    from apex_amd.contrib.conv_bias_relu import ConvBiasReLU

    torch.manual_seed(2)
    x = torch.randn(4, 16, 14, 14, device="cuda", requires_grad=True)
    w = (torch.randn(32, 16, 1, 1, device="cuda") * 0.1).requires_grad_()
    b = torch.randn(1, 32, 1, 1, device="cuda", requires_grad=True)
    y = ConvBiasReLU(x, w, b, 0, stride)
    ref = torch.relu(torch.nn.functional.conv2d(
        x.detach(), w.detach(), stride=stride) + b.detach())
    torch.testing.assert_close(y, ref, rtol=1e-4, atol=1e-4)
    g = torch.randn_like(y)
    y.backward(g)
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    yr = torch.relu(torch.nn.functional.conv2d(xr, wr, stride=stride) + br)
    yr.backward(g)
    torch.testing.assert_close(x.grad, xr.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(w.grad, wr.grad, rtol=1e-4, atol=1e-3)
    torch.testing.assert_close(b.grad, br.grad, rtol=1e-4, atol=1e-3)


def test_conv1x1_frozen_scale_bias_relu_fused_gpu():
    from apex_amd.contrib.bottleneck import FrozenBatchNorm2d
    from apex_amd.contrib.conv_bias_relu import ConvFrozenScaleBiasReLU

    torch.manual_seed(3)
    bn = FrozenBatchNorm2d(32).cuda()
    bn.running_var.uniform_(0.5, 2.0)
    bn.running_mean.normal_()
    s, b = bn.get_scale_bias()
    x = torch.randn(2, 16, 8, 8, device="cuda", requires_grad=True)
    w = (torch.randn(32, 16, 1, 1, device="cuda") * 0.1).requires_grad_()
    y = ConvFrozenScaleBiasReLU(x, w, s, b, 0, 1)
    ref = torch.relu(torch.nn.functional.conv2d(x.detach(), w.detach()) * s + b)
    torch.testing.assert_close(y, ref, rtol=1e-4, atol=1e-4)
    g = torch.randn_like(y)
    y.backward(g)
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    yr = torch.relu(torch.nn.functional.conv2d(xr, wr) * s + b)
    yr.backward(g)
    torch.testing.assert_close(x.grad, xr.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(w.grad, wr.grad, rtol=1e-4, atol=1e-3)
