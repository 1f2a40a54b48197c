"""GPU SyncBatchNorm kernel tests (single GPU: SBN == torch BN; raw kernel
calls vs torch reference math — pattern of reference
tests/distributed/synced_batchnorm/single_gpu_unit_test.py)."""

import torch
import pytest

pytestmark = pytest.mark.gpu


def test_welford_mean_var_kernel():
    import apex_amd._syncbn as syncbn

    torch.manual_seed(0)
    x = torch.randn(8, 16, 13, 17, device="cuda")
    mean, var = syncbn.welford_mean_var(x)
    ref_mean = x.mean(dim=(0, 2, 3))
    ref_var = x.var(dim=(0, 2, 3), unbiased=False)
    torch.testing.assert_close(mean, ref_mean, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(var, ref_var, rtol=1e-5, atol=1e-5)


def test_welford_mean_var_c_last_kernel():
    import apex_amd._syncbn as syncbn

    torch.manual_seed(0)
    x = torch.randn(8, 13, 17, 16, device="cuda")  # NHWC
    mean, var = syncbn.welford_mean_var_c_last(x)
    ref_mean = x.mean(dim=(0, 1, 2))
    ref_var = x.var(dim=(0, 1, 2), unbiased=False)
    torch.testing.assert_close(mean, ref_mean, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(var, ref_var, rtol=1e-5, atol=1e-5)


def test_welford_parallel_kernel():
    import apex_amd._syncbn as syncbn

    torch.manual_seed(1)
    xs = [torch.randn(100 + i * 10, 7, device="cuda") for i in range(3)]
    means = torch.stack([x.mean(0) for x in xs])
    vars_ = torch.stack([x.var(0, unbiased=False) for x in xs])
    counts = torch.tensor([x.shape[0] for x in xs], dtype=torch.int32, device="cuda")
    mean, var = syncbn.welford_parallel(means, vars_, counts)
    allx = torch.cat(xs, 0)
    torch.testing.assert_close(mean, allx.mean(0), rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(var, allx.var(0, unbiased=False), rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16])
def test_syncbn_module_matches_bn_gpu(dtype):
    from apex_amd.parallel import SyncBatchNorm

    torch.manual_seed(2)
    C = 16
    x = torch.randn(8, C, 14, 14, device="cuda", dtype=dtype, requires_grad=True)
    x2 = x.detach().float().clone().requires_grad_(True)
    sbn = SyncBatchNorm(C).cuda().to(dtype if dtype != torch.float16 else torch.float32)
    bn = torch.nn.BatchNorm2d(C).cuda()
    sbn.train()
    bn.train()
    with torch.no_grad():
        bn.weight.copy_(sbn.weight)
        bn.bias.copy_(sbn.bias)

    y = sbn(x)
    y_ref = bn(x2)
    tol = dict(rtol=1e-4, atol=1e-4) if dtype == torch.float32 else dict(rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(y.float(), y_ref, **tol)

    g = torch.randn_like(y_ref)
    y.backward(g.to(y.dtype))
    y_ref.backward(g)
    torch.testing.assert_close(x.grad.float(), x2.grad, **tol)
    # weight/bias grads accumulate 1568 low-precision products per channel
    wtol = dict(rtol=1e-3, atol=1e-3) if dtype == torch.float32 else dict(rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(sbn.weight.grad, bn.weight.grad, **wtol)
    torch.testing.assert_close(sbn.bias.grad, bn.bias.grad, **wtol)
    torch.testing.assert_close(sbn.running_mean, bn.running_mean, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(sbn.running_var, bn.running_var, rtol=1e-4, atol=1e-5)


def test_syncbn_channel_last_gpu():
    from apex_amd.parallel import SyncBatchNorm

    torch.manual_seed(3)
    C = 32
    x = torch.randn(4, 9, 9, C, device="cuda", requires_grad=True)
    sbn = SyncBatchNorm(C, channel_last=True).cuda()
    sbn.train()
    y = sbn(x)
    # reference: permute to NCHW and use torch BN
    x2 = x.detach().permute(0, 3, 1, 2).contiguous().requires_grad_(True)
    bn = torch.nn.BatchNorm2d(C).cuda()
    bn.train()
    with torch.no_grad():
        bn.weight.copy_(sbn.weight)
        bn.bias.copy_(sbn.bias)
    y_ref = bn(x2).permute(0, 2, 3, 1)
    torch.testing.assert_close(y, y_ref, rtol=1e-4, atol=1e-4)
    g = torch.randn_like(y)
    y.backward(g)
    y_ref.backward(g)
    torch.testing.assert_close(x.grad, x2.grad.permute(0, 2, 3, 1), rtol=1e-4, atol=1e-4)


def test_syncbn_fuse_relu_backward_gpu():
    """fuse_relu backward must gate grads by the ReLU mask (kernel path)."""
    from apex_amd.parallel import SyncBatchNorm

    torch.manual_seed(7)
    C = 16
    x = torch.randn(4, C, 6, 6, device="cuda", requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    sbn = SyncBatchNorm(C, fuse_relu=True).cuda()
    bn = torch.nn.BatchNorm2d(C).cuda()
    sbn.train(); bn.train()
    with torch.no_grad():
        bn.weight.copy_(sbn.weight)
        bn.bias.copy_(sbn.bias)
    y1 = sbn(x)
    y2 = torch.relu(bn(x2))
    torch.testing.assert_close(y1, y2, rtol=1e-4, atol=1e-5)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    torch.testing.assert_close(x.grad, x2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(sbn.weight.grad, bn.weight.grad, rtol=1e-4, atol=1e-4)


def test_groupbn_add_relu_gpu():
    """relu(bn(x) + z) NHWC path on the welford kernels."""
    from apex_amd.contrib.groupbn import BatchNorm2d_NHWC

    torch.manual_seed(8)
    C = 16
    x = torch.randn(3, 6, 6, C, device="cuda", requires_grad=True)
    z = torch.randn(3, 6, 6, C, device="cuda", requires_grad=True)
    m = BatchNorm2d_NHWC(C, fuse_relu=True).cuda()
    m.train()
    x2 = x.detach().clone().requires_grad_(True)
    z2 = z.detach().clone().requires_grad_(True)
    bn = torch.nn.BatchNorm2d(C).cuda()
    bn.train()
    with torch.no_grad():
        bn.weight.copy_(m.weight)
        bn.bias.copy_(m.bias)
    y1 = m(x, z)
    y2 = torch.relu(bn(x2.permute(0, 3, 1, 2)).permute(0, 2, 3, 1) + z2)
    torch.testing.assert_close(y1, y2, rtol=1e-4, atol=1e-5)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    torch.testing.assert_close(x.grad, x2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(z.grad, z2.grad, rtol=1e-4, atol=1e-5)
