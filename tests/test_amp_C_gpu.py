"""GPU numerics tests: every _amp_C multi-tensor HIP kernel vs plain
PyTorch fp32 reference math (run with -m gpu on an MI355X)."""

import torch
import pytest

pytestmark = pytest.mark.gpu

CHUNK = 2048 * 32


def ext():
    import apex_amd._amp_C as m

    return m


def noop():
    return torch.zeros(1, dtype=torch.int32, device="cuda")


def make_list(shapes=((1024,), (3333,), (64, 129), (7,)), dtype=torch.float32, seed=0):
    torch.manual_seed(seed)
    return [torch.randn(*s, device="cuda").to(dtype) for s in shapes]


# ---------- scale ----------
@pytest.mark.parametrize("in_dtype,out_dtype", [
    (torch.float32, torch.float32),
    (torch.float16, torch.float32),
    (torch.bfloat16, torch.float32),
    (torch.float32, torch.bfloat16),
])
def test_multi_tensor_scale(in_dtype, out_dtype):
    xs = make_list(dtype=in_dtype)
    outs = [torch.empty_like(x, dtype=out_dtype) for x in xs]
    flag = noop()
    ext().multi_tensor_scale(CHUNK, flag, [xs, outs], 0.125)
    torch.cuda.synchronize()
    assert flag.item() == 0
    for x, o in zip(xs, outs):
        torch.testing.assert_close(o, (x.float() * 0.125).to(out_dtype), rtol=1e-3, atol=1e-3)


def test_multi_tensor_scale_overflow_flag():
    xs = make_list(dtype=torch.float32)
    xs[1].view(-1)[7] = float("inf")
    outs = [torch.empty_like(x) for x in xs]
    flag = noop()
    ext().multi_tensor_scale(CHUNK, flag, [xs, outs], 1.0)
    torch.cuda.synchronize()
    assert flag.item() == 1


def test_multi_tensor_scale_misaligned():
    # odd slice offsets force the non-vectorized path
    base = torch.randn(5001, device="cuda")
    xs = [base[1:4002]]
    outs = [torch.empty_like(xs[0])]
    flag = noop()
    ext().multi_tensor_scale(CHUNK, flag, [xs, outs], 2.0)
    torch.cuda.synchronize()
    torch.testing.assert_close(outs[0], xs[0] * 2.0)


def test_multi_tensor_scale_many_tensors():
    # >max_tensors per launch → multi-launch path
    xs = [torch.randn(17 + i, device="cuda") for i in range(300)]
    outs = [torch.empty_like(x) for x in xs]
    flag = noop()
    ext().multi_tensor_scale(CHUNK, flag, [xs, outs], -1.5)
    torch.cuda.synchronize()
    for x, o in zip(xs, outs):
        torch.testing.assert_close(o, x * -1.5)


def test_multi_tensor_scale_large_tensor():
    # > one chunk per tensor (many workgroups)
    x = torch.randn(3 * CHUNK + 12345, device="cuda")
    out = torch.empty_like(x)
    flag = noop()
    ext().multi_tensor_scale(CHUNK, flag, [[x], [out]], 3.0)
    torch.cuda.synchronize()
    torch.testing.assert_close(out, x * 3.0)


# ---------- axpby ----------
def test_multi_tensor_axpby():
    xs = make_list()
    ys = make_list(seed=1)
    outs = [torch.empty_like(x) for x in xs]
    flag = noop()
    ext().multi_tensor_axpby(CHUNK, flag, [xs, ys, outs], 2.0, -0.5, -1)
    torch.cuda.synchronize()
    assert flag.item() == 0
    for x, y, o in zip(xs, ys, outs):
        torch.testing.assert_close(o, 2.0 * x - 0.5 * y, rtol=1e-5, atol=1e-6)


def test_multi_tensor_axpby_arg_to_check():
    xs = make_list()
    ys = make_list(seed=1)
    ys[0].view(-1)[0] = float("nan")
    outs = [torch.empty_like(x) for x in xs]
    flag = noop()
    ext().multi_tensor_axpby(CHUNK, flag, [xs, ys, outs], 1.0, 1.0, 0)  # check x only
    torch.cuda.synchronize()
    assert flag.item() == 0
    flag = noop()
    ext().multi_tensor_axpby(CHUNK, flag, [xs, ys, outs], 1.0, 1.0, 1)  # check y
    torch.cuda.synchronize()
    assert flag.item() == 1


# ---------- norms ----------
def test_multi_tensor_l2norm():
    xs = make_list()
    norm, per_tensor = ext().multi_tensor_l2norm(CHUNK, noop(), [xs], True)
    torch.cuda.synchronize()
    expected = torch.norm(torch.cat([x.reshape(-1) for x in xs]))
    torch.testing.assert_close(norm.squeeze(), expected, rtol=1e-5, atol=1e-6)
    for i, x in enumerate(xs):
        torch.testing.assert_close(per_tensor[i], x.norm(), rtol=1e-5, atol=1e-6)


def test_multi_tensor_l2norm_deterministic():
    xs = make_list(shapes=((511, 513), (1029,), (2048, 65)))
    a = ext().multi_tensor_l2norm(CHUNK, noop(), [xs], False)[0]
    b = ext().multi_tensor_l2norm(CHUNK, noop(), [xs], False)[0]
    torch.cuda.synchronize()
    assert a.item() == b.item()  # bitwise-stable reduction order


def test_multi_tensor_l2norm_fp16():
    xs = make_list(dtype=torch.float16)
    norm, _ = ext().multi_tensor_l2norm(CHUNK, noop(), [xs], False)
    expected = torch.norm(torch.cat([x.reshape(-1).float() for x in xs]))
    torch.testing.assert_close(norm.squeeze(), expected, rtol=1e-3, atol=1e-3)


def test_multi_tensor_unscale_l2norm():
    xs = make_list()
    inv_scale = torch.tensor([0.25], device="cuda")
    norm, _ = ext().multi_tensor_unscale_l2norm(CHUNK, noop(), [xs], inv_scale, False)
    expected = torch.norm(torch.cat([x.reshape(-1) * 0.25 for x in xs]))
    torch.testing.assert_close(norm.squeeze(), expected, rtol=1e-5, atol=1e-6)
    # inputs unmodified
    torch.testing.assert_close(xs[0], make_list()[0])


def test_multi_tensor_l2norm_scale():
    xs = make_list()
    outs = [torch.empty_like(x) for x in xs]
    norm, _ = ext().multi_tensor_l2norm_scale(CHUNK, noop(), [xs, outs], 0.5, False)
    expected = torch.norm(torch.cat([x.reshape(-1) * 0.5 for x in xs]))
    torch.testing.assert_close(norm.squeeze(), expected, rtol=1e-5, atol=1e-6)
    for x, o in zip(xs, outs):
        torch.testing.assert_close(o, x * 0.5)


def test_multi_tensor_maxnorm():
    xs = make_list()
    mx = ext().multi_tensor_maxnorm(CHUNK, noop(), [xs])
    for i, x in enumerate(xs):
        torch.testing.assert_close(mx[i], x.abs().max(), rtol=1e-6, atol=0)


# ---------- update_scale_hysteresis ----------
def test_update_scale_hysteresis():
    scale = torch.tensor([1024.0], device="cuda")
    growth = torch.zeros(1, dtype=torch.int32, device="cuda")
    hyst = torch.tensor([2], dtype=torch.int32, device="cuda")
    found_inf = torch.zeros(1, dtype=torch.int32, device="cuda")

    # growth after interval clean steps
    for _ in range(3):
        ext().update_scale_hysteresis(scale, growth, hyst, found_inf, 2.0, 0.5, 3, 2)
    torch.cuda.synchronize()
    assert scale.item() == 2048.0

    # first inf: hysteresis consumes, no backoff yet
    found_inf.fill_(1)
    ext().update_scale_hysteresis(scale, growth, hyst, found_inf, 2.0, 0.5, 3, 2)
    torch.cuda.synchronize()
    assert scale.item() == 2048.0
    # second inf: backoff
    ext().update_scale_hysteresis(scale, growth, hyst, found_inf, 2.0, 0.5, 3, 2)
    torch.cuda.synchronize()
    assert scale.item() == 1024.0
