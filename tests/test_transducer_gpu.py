"""GPU numerics for the transducer joint and RNN-T loss vs fp32 references."""

import torch
import pytest

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("relu", [False, True])
def test_transducer_joint_gpu(relu):
    from apex_amd.contrib.transducer import TransducerJoint

    torch.manual_seed(0)
    B, T, U, H = 3, 7, 5, 32
    f = torch.randn(B, T, H, device="cuda", requires_grad=True)
    g = torch.randn(B, U, H, device="cuda", requires_grad=True)
    f_len = torch.tensor([7, 5, 6], dtype=torch.int32, device="cuda")
    g_len = torch.tensor([5, 3, 4], dtype=torch.int32, device="cuda")
    joint = TransducerJoint(relu=relu)
    out = joint(f, g, f_len, g_len)

    fr = f.detach().clone().requires_grad_(True)
    gr = g.detach().clone().requires_grad_(True)
    ref = fr.unsqueeze(2) + gr.unsqueeze(1)
    mask_t = torch.arange(T, device="cuda")[None, :, None, None] < f_len[:, None, None, None]
    mask_u = torch.arange(U, device="cuda")[None, None, :, None] < g_len[:, None, None, None]
    ref = ref * (mask_t & mask_u)
    if relu:
        ref = torch.relu(ref)
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)

    go = torch.randn_like(out)
    out.backward(go)
    ref.backward(go)
    torch.testing.assert_close(f.grad, fr.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(g.grad, gr.grad, rtol=1e-4, atol=1e-5)


def test_transducer_loss_gpu():
    from apex_amd.contrib.transducer import TransducerLoss
    from apex_amd.contrib.transducer.transducer import _ref_rnnt_loss

    torch.manual_seed(1)
    B, T, Umax, V = 2, 6, 4, 10  # Umax = max labels + 1
    blank = 0
    y_len = torch.tensor([3, 2], dtype=torch.int32, device="cuda")
    f_len = torch.tensor([6, 4], dtype=torch.int32, device="cuda")
    label = torch.randint(1, V, (B, Umax - 1), dtype=torch.int32, device="cuda")
    logits = torch.randn(B, T, Umax, V, device="cuda", requires_grad=True)
    x = torch.log_softmax(logits, dim=-1)

    loss_mod = TransducerLoss()
    losses = loss_mod(x, label, f_len, y_len, blank_idx=blank)
    ref = _ref_rnnt_loss(x.detach().cpu(), label.cpu().long(), f_len.cpu(), y_len.cpu(), blank)
    torch.testing.assert_close(losses.cpu(), ref, rtol=1e-4, atol=1e-4)

    # grads vs autograd through the python DP
    losses.sum().backward()
    logits2 = logits.detach().cpu().requires_grad_(True)
    x2 = torch.log_softmax(logits2, dim=-1)
    _ref_rnnt_loss(x2, label.cpu().long(), f_len.cpu(), y_len.cpu(), blank).sum().backward()
    torch.testing.assert_close(logits.grad.cpu(), logits2.grad, rtol=1e-3, atol=1e-4)


def test_transducer_loss_decreases():
    """End-to-end sanity: a tiny joint+loss setup trains."""
    from apex_amd.contrib.transducer import TransducerJoint, TransducerLoss

    torch.manual_seed(2)
    B, T, Umax, H, V = 2, 5, 3, 16, 8
    f = torch.randn(B, T, H, device="cuda", requires_grad=True)
    g = torch.randn(B, Umax, H, device="cuda", requires_grad=True)
    proj = torch.nn.Linear(H, V).cuda()
    f_len = torch.full((B,), T, dtype=torch.int32, device="cuda")
    g_len = torch.full((B,), Umax - 1, dtype=torch.int32, device="cuda")
    y_len = g_len.clone()
    label = torch.randint(1, V, (B, Umax - 1), dtype=torch.int32, device="cuda")
    joint = TransducerJoint()
    loss_mod = TransducerLoss()
    opt = torch.optim.Adam([f, g] + list(proj.parameters()), lr=5e-2)
    losses = []
    for _ in range(25):
        opt.zero_grad()
        h = joint(f, g, f_len, g_len + 1)
        x = torch.log_softmax(proj(h), dim=-1)
        loss = loss_mod(x, label, f_len, y_len).mean()
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.5


def test_transducer_joint_packed_matches_dense():
    from apex_amd.contrib.transducer import TransducerJoint

    torch.manual_seed(4)
    B, T, U, H = 3, 6, 4, 16
    f = torch.randn(B, T, H, device="cuda", requires_grad=True)
    g = torch.randn(B, U, H, device="cuda", requires_grad=True)
    f_len = torch.tensor([6, 4, 5], dtype=torch.int32, device="cuda")
    g_len = torch.tensor([4, 2, 3], dtype=torch.int32, device="cuda")
    batch_offset = torch.cumsum((f_len * g_len).long(), 0)
    packed_batch = int(batch_offset[-1])

    jp = TransducerJoint(pack_output=True)
    out_p = jp(f, g, f_len, g_len, batch_offset=batch_offset, packed_batch=packed_batch)
    assert out_p.shape == (packed_batch, H)

    f2 = f.detach().clone().requires_grad_(True)
    g2 = g.detach().clone().requires_grad_(True)
    jd = TransducerJoint()
    out_d = jd(f2, g2, f_len, g_len)
    # gather the dense valid region into packed order for comparison
    rows = []
    for b in range(B):
        rows.append(out_d[b, :int(f_len[b]), :int(g_len[b]), :].reshape(-1, H))
    ref = torch.cat(rows, 0)
    torch.testing.assert_close(out_p, ref, rtol=1e-5, atol=1e-6)

    go = torch.randn_like(out_p)
    out_p.backward(go)
    # dense backward with the same grads scattered in
    go_dense = torch.zeros_like(out_d)
    off = 0
    for b in range(B):
        n = int(f_len[b] * g_len[b])
        go_dense[b, :int(f_len[b]), :int(g_len[b]), :] = go[off:off + n].reshape(
            int(f_len[b]), int(g_len[b]), H)
        off += n
    out_d.backward(go_dense)
    torch.testing.assert_close(f.grad, f2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(g.grad, g2.grad, rtol=1e-4, atol=1e-5)


def test_transducer_loss_packed_matches_dense():
    from apex_amd.contrib.transducer import TransducerLoss

    torch.manual_seed(5)
    B, T, Umax, V = 2, 5, 4, 8
    f_len = torch.tensor([5, 3], dtype=torch.int32, device="cuda")
    y_len = torch.tensor([3, 2], dtype=torch.int32, device="cuda")
    label = torch.randint(1, V, (B, Umax - 1), dtype=torch.int32, device="cuda")
    x = torch.log_softmax(torch.randn(B, T, Umax, V, device="cuda"), dim=-1)
    x_dense = x.detach().clone().requires_grad_(True)

    # build packed input
    rows = []
    for b in range(B):
        rows.append(x[b, :int(f_len[b]), :int(y_len[b]) + 1, :].reshape(-1, V))
    x_packed = torch.cat(rows, 0).detach().clone().requires_grad_(True)
    batch_offset = torch.cumsum((f_len * (y_len + 1)).long(), 0)

    dense = TransducerLoss()
    packed = TransducerLoss(packed_input=True)
    l_d = dense(x_dense, label, f_len, y_len)
    l_p = packed(x_packed, label, f_len, y_len, batch_offset=batch_offset, max_f_len=T)
    torch.testing.assert_close(l_p, l_d, rtol=1e-5, atol=1e-5)

    l_p.sum().backward()
    l_d.sum().backward()
    # compare packed grads against the dense grads' valid region
    off = 0
    for b in range(B):
        n = int(f_len[b] * (y_len[b] + 1))
        ref = x_dense.grad[b, :int(f_len[b]), :int(y_len[b]) + 1, :].reshape(-1, V)
        torch.testing.assert_close(x_packed.grad[off:off + n], ref, rtol=1e-4, atol=1e-5)
        off += n


def test_joint_fused_dropout():
    """Fused philox dropout: scaled kept values, ~p drop rate, deterministic
    under torch.manual_seed, and the backward mask matches the forward mask
    (grads flow only through kept elements)."""
    from apex_amd.contrib.transducer import TransducerJoint

    torch.manual_seed(42)
    B, T, U, H = 4, 24, 12, 64
    f = torch.randn(B, T, H, device="cuda", requires_grad=True)
    g = torch.randn(B, U, H, device="cuda", requires_grad=True)
    f_len = torch.full((B,), T, dtype=torch.int32, device="cuda")
    g_len = torch.full((B,), U, dtype=torch.int32, device="cuda")
    p = 0.3

    joint = TransducerJoint(dropout=True, dropout_prob=p).train()
    torch.manual_seed(7)
    out = joint(f, g, f_len, g_len)

    base = f.detach().unsqueeze(2) + g.detach().unsqueeze(1)
    kept = out != 0
    # kept values are base / (1-p)
    torch.testing.assert_close(out[kept], (base / (1 - p))[kept], rtol=1e-5, atol=1e-5)
    rate = 1.0 - kept.float().mean().item()
    assert abs(rate - p) < 0.02, f"drop rate {rate} vs p {p}"

    # determinism: same torch seed -> same mask
    torch.manual_seed(7)
    out2 = joint(f, g, f_len, g_len)
    assert torch.equal(out.detach(), out2.detach())

    # backward: grads flow exactly through the kept elements with 1/(1-p)
    dout = torch.randn_like(out)
    out.backward(dout)
    fr = f.detach().clone().requires_grad_(True)
    gr = g.detach().clone().requires_grad_(True)
    ref = (fr.unsqueeze(2) + gr.unsqueeze(1)) * kept.float() / (1 - p)
    ref.backward(dout)
    torch.testing.assert_close(f.grad, fr.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(g.grad, gr.grad, rtol=1e-4, atol=1e-4)


def test_joint_fused_dropout_packed():
    from apex_amd.contrib.transducer import TransducerJoint

    torch.manual_seed(1)
    B, T, U, H = 3, 10, 6, 32
    f = torch.randn(B, T, H, device="cuda", requires_grad=True)
    g = torch.randn(B, U, H, device="cuda", requires_grad=True)
    f_len = torch.tensor([10, 7, 5], dtype=torch.int32, device="cuda")
    g_len = torch.tensor([6, 4, 3], dtype=torch.int32, device="cuda")
    batch_offset = torch.cumsum(f_len.long() * g_len.long(), 0)
    packed = int(batch_offset[-1])
    p = 0.25
    joint = TransducerJoint(pack_output=True, dropout=True, dropout_prob=p).train()
    out = joint(f, g, f_len, g_len, batch_offset=batch_offset, packed_batch=packed)
    assert out.shape == (packed, H)
    kept = out != 0
    rate = 1.0 - kept.float().mean().item()
    assert abs(rate - p) < 0.05
    out.sum().backward()
    assert torch.isfinite(f.grad).all() and torch.isfinite(g.grad).all()
