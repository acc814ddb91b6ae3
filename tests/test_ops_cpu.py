"""CPU reference-op tests: gather_mean + fused LSTM cell numerics.

The same reference functions are the ground truth the HIP kernels are
compared against in tests/test_ops_gpu.py.
"""
import numpy as np
import pytest
import torch

from nerrf_amd.ops import gather_mean, lstm_cell
from nerrf_amd.ops.reference import (
    gather_mean_bwd_ref,
    gather_mean_ref,
    lstm_pointwise_fwd_ref,
)

torch.manual_seed(0)


def test_gather_mean_matches_loop():
    n, k, d = 37, 5, 19
    h = torch.randn(n, d)
    idx = torch.randint(0, n, (n, k))
    w = torch.rand(n, k) + 0.1
    out = gather_mean(h, idx, w)
    for i in range(0, n, 7):
        acc = torch.zeros(d)
        for j in range(k):
            acc += w[i, j] * h[idx[i, j]]
        acc /= w[i].sum()
        assert torch.allclose(out[i], acc, atol=1e-5)


def test_gather_mean_backward_matches_autograd():
    n, k, d = 23, 4, 11
    h = torch.randn(n, d, requires_grad=True)
    idx = torch.randint(0, n, (n, k))
    w = torch.rand(n, k) + 0.1
    out = gather_mean(h, idx, w)
    g = torch.randn_like(out)
    out.backward(g)
    manual = h.grad.clone()

    h2 = h.detach().clone().requires_grad_(True)
    out2 = gather_mean_ref(h2, idx, w)
    out2.backward(g)
    assert torch.allclose(manual, h2.grad, atol=1e-5)
    # and the explicit bwd reference
    explicit = gather_mean_bwd_ref(g, idx, w, n)
    assert torch.allclose(manual, explicit, atol=1e-5)


def test_lstm_cell_matches_torch_lstmcell():
    """Unmasked fused cell == torch.nn.LSTMCell (same gate convention)."""
    b, e, hd = 9, 7, 13
    cell = torch.nn.LSTMCell(e, hd)
    x = torch.randn(b, e)
    h0 = torch.randn(b, hd)
    c0 = torch.randn(b, hd)
    h_ref, c_ref = cell(x, (h0, c0))

    # our op takes xg = x W_ih^T + b_ih precomputed, w_hh, b = b_hh
    xg = x @ cell.weight_ih.t() + cell.bias_ih
    h_new, c_new = lstm_cell(xg, h0, c0, cell.weight_hh, cell.bias_hh)
    assert torch.allclose(h_new, h_ref, atol=1e-5)
    assert torch.allclose(c_new, c_ref, atol=1e-5)


def test_lstm_cell_mask_passthrough():
    b, hd = 6, 8
    xg = torch.randn(b, 4 * hd)
    h0 = torch.randn(b, hd)
    c0 = torch.randn(b, hd)
    w_hh = torch.randn(4 * hd, hd) * 0.1
    bias = torch.randn(4 * hd)
    mask = torch.tensor([1.0, 0.0, 1.0, 0.0, 1.0, 0.0])
    h1, c1 = lstm_cell(xg, h0, c0, w_hh, bias, mask)
    assert torch.allclose(h1[1], h0[1])
    assert torch.allclose(c1[3], c0[3])
    assert not torch.allclose(h1[0], h0[0])


def test_lstm_cell_backward_matches_autograd():
    """Manual backward (the kernel math) vs torch autograd through the ref."""
    b, hd = 5, 6
    xg = torch.randn(b, 4 * hd, requires_grad=True)
    h0 = torch.randn(b, hd, requires_grad=True)
    c0 = torch.randn(b, hd, requires_grad=True)
    w_hh = torch.randn(4 * hd, hd, requires_grad=True)
    bias = torch.randn(4 * hd, requires_grad=True)
    mask = torch.tensor([1.0, 1.0, 0.0, 1.0, 0.0])

    h1, c1 = lstm_cell(xg, h0, c0, w_hh, bias, mask)
    loss = (h1 * torch.arange(hd).float()).sum() + (c1 * 0.3).sum()
    loss.backward()
    grads_manual = [t.grad.clone() for t in (xg, h0, c0, w_hh, bias)]

    # pure autograd graph
    for t in (xg, h0, c0, w_hh, bias):
        t.grad = None
    gates_pre = torch.addmm(bias, h0, w_hh.t()) + xg
    h_ref, c_ref, _ = lstm_pointwise_fwd_ref(gates_pre, c0, h0, mask)
    loss2 = (h_ref * torch.arange(hd).float()).sum() + (c_ref * 0.3).sum()
    loss2.backward()
    grads_auto = [t.grad.clone() for t in (xg, h0, c0, w_hh, bias)]
    for gm, ga, name in zip(grads_manual, grads_auto, ["xg", "h0", "c0", "w_hh", "b"]):
        assert torch.allclose(gm, ga, atol=1e-5), f"grad mismatch: {name}"


def test_lstm_cell_multi_step_gradcheck():
    """Small double-precision gradcheck through two chained fused steps."""
    torch.manual_seed(1)
    b, hd = 3, 4
    xg1 = torch.randn(b, 4 * hd, dtype=torch.double, requires_grad=True)
    xg2 = torch.randn(b, 4 * hd, dtype=torch.double, requires_grad=True)
    h0 = torch.randn(b, hd, dtype=torch.double, requires_grad=True)
    c0 = torch.randn(b, hd, dtype=torch.double, requires_grad=True)
    w_hh = (torch.randn(4 * hd, hd, dtype=torch.double) * 0.2).requires_grad_(True)
    bias = torch.randn(4 * hd, dtype=torch.double, requires_grad=True)
    mask = torch.tensor([1.0, 0.0, 1.0], dtype=torch.double)

    def fn(xg1, xg2, h0, c0, w_hh, bias):
        h1, c1 = lstm_cell(xg1, h0, c0, w_hh, bias, mask)
        h2, c2 = lstm_cell(xg2, h1, c1, w_hh, bias, mask)
        return (h2 * 1.7).sum() + (c2 * 0.9).sum()

    assert torch.autograd.gradcheck(fn, (xg1, xg2, h0, c0, w_hh, bias), eps=1e-6, atol=1e-4)


def test_gather_mean_rejects_nothing_on_cpu():
    # CPU path never requires the native extension
    h = torch.randn(4, 8)
    idx = torch.zeros(4, 2, dtype=torch.int64)
    w = torch.ones(4, 2)
    out = gather_mean(h, idx, w)
    assert torch.allclose(out, h[0].expand(4, 8))


def test_lstm_sequence_matches_torch_lstm():
    """Full-length lstm_sequence (fwd + reverse) == torch.nn.LSTM bidirectional."""
    from nerrf_amd.ops import lstm_sequence

    torch.manual_seed(2)
    b, t, e, hd = 4, 11, 5, 8
    ref_lstm = torch.nn.LSTM(e, hd, num_layers=1, bidirectional=True, batch_first=False)
    x = torch.randn(t, b, e)
    out_ref, _ = ref_lstm(x)

    mask = torch.ones(t, b)
    outs = []
    for di, rev in enumerate([False, True]):
        w_ih = getattr(ref_lstm, f"weight_ih_l0{'_reverse' if rev else ''}")
        w_hh = getattr(ref_lstm, f"weight_hh_l0{'_reverse' if rev else ''}")
        b_ih = getattr(ref_lstm, f"bias_ih_l0{'_reverse' if rev else ''}")
        b_hh = getattr(ref_lstm, f"bias_hh_l0{'_reverse' if rev else ''}")
        xg = torch.matmul(x.reshape(t * b, e), w_ih.t()).reshape(t, b, 4 * hd) + b_ih
        h0 = torch.zeros(b, hd)
        c0 = torch.zeros(b, hd)
        outs.append(lstm_sequence(xg, h0, c0, w_hh, b_hh, mask, reverse=rev))
    out = torch.cat(outs, dim=-1)
    assert torch.allclose(out, out_ref, atol=1e-5)


def test_lstm_sequence_backward_matches_autograd():
    from nerrf_amd.ops import lstm_sequence
    from nerrf_amd.ops.reference import lstm_pointwise_fwd_ref

    torch.manual_seed(3)
    b, t, hd = 3, 6, 4
    xg = torch.randn(t, b, 4 * hd, requires_grad=True)
    h0 = torch.randn(b, hd, requires_grad=True)
    c0 = torch.randn(b, hd, requires_grad=True)
    w_hh = (torch.randn(4 * hd, hd) * 0.3).requires_grad_(True)
    bias = torch.randn(4 * hd, requires_grad=True)
    mask = (torch.rand(t, b) > 0.25).float()

    out = lstm_sequence(xg, h0, c0, w_hh, bias, mask, reverse=False)
    g = torch.randn_like(out)
    out.backward(g)
    manual = [p.grad.clone() for p in (xg, h0, c0, w_hh, bias)]

    for p in (xg, h0, c0, w_hh, bias):
        p.grad = None
    h, c = h0, c0
    outs = []
    for ti in range(t):
        gates_pre = torch.addmm(bias, h, w_hh.t()) + xg[ti]
        h, c, _ = lstm_pointwise_fwd_ref(gates_pre, c, h, mask[ti])
        outs.append(h)
    torch.stack(outs).backward(g)
    auto = [p.grad for p in (xg, h0, c0, w_hh, bias)]
    for gm, ga, name in zip(manual, auto, ["xg", "h0", "c0", "w_hh", "bias"]):
        assert torch.allclose(gm, ga, atol=1e-5), f"grad {name}"


def test_lstm_sequence_reverse_backward():
    from nerrf_amd.ops import lstm_sequence
    from nerrf_amd.ops.reference import lstm_pointwise_fwd_ref

    torch.manual_seed(4)
    b, t, hd = 2, 5, 3
    xg = torch.randn(t, b, 4 * hd, requires_grad=True)
    h0 = torch.zeros(b, hd)
    c0 = torch.zeros(b, hd)
    w_hh = (torch.randn(4 * hd, hd) * 0.3).requires_grad_(True)
    bias = torch.randn(4 * hd, requires_grad=True)
    mask = torch.ones(t, b)

    out = lstm_sequence(xg, h0, c0, w_hh, bias, mask, reverse=True)
    g = torch.randn_like(out)
    out.backward(g)
    manual = [p.grad.clone() for p in (xg, w_hh, bias)]

    for p in (xg, w_hh, bias):
        p.grad = None
    h, c = h0, c0
    outs = [None] * t
    for ti in range(t - 1, -1, -1):
        gates_pre = torch.addmm(bias, h, w_hh.t()) + xg[ti]
        h, c, _ = lstm_pointwise_fwd_ref(gates_pre, c, h, mask[ti])
        outs[ti] = h
    torch.stack(outs).backward(g)
    auto = [p.grad for p in (xg, w_hh, bias)]
    for gm, ga, name in zip(manual, auto, ["xg", "w_hh", "bias"]):
        assert torch.allclose(gm, ga, atol=1e-5), f"grad {name}"


def test_reverse_index_consistent_with_bwd_ref():
    from nerrf_amd.graph.sampling import reverse_index

    torch.manual_seed(5)
    n, k, d = 29, 6, 9
    idx = torch.randint(0, n, (n, k))
    w = torch.rand(n, k) + 0.05
    g = torch.randn(n, d)
    expected = gather_mean_bwd_ref(g, idx, w, n)
    rd, rs, rw = reverse_index(idx.numpy(), w.numpy())
    assert (np.diff(rd) >= 0).all()  # sorted by destination
    out = torch.zeros(n, d)
    for e in range(len(rd)):
        out[int(rd[e])] += float(rw[e]) * g[int(rs[e])]
    assert torch.allclose(out, expected, atol=1e-5)


def test_lstm_bilayer_matches_two_directions():
    """lstm_bilayer == cat(lstm_sequence fwd, lstm_sequence bwd) — outputs
    and every gradient (xg, weights, biases)."""
    from nerrf_amd.ops.lstm_seq import lstm_bilayer, lstm_sequence

    torch.manual_seed(11)
    t, b, h = 9, 7, 16
    mk = (torch.rand(t, b) > 0.25).float()

    def inputs():
        torch.manual_seed(12)
        xf = (torch.randn(t, b, 4 * h) * 0.3).requires_grad_(True)
        xb = (torch.randn(t, b, 4 * h) * 0.3).requires_grad_(True)
        wf = (torch.randn(4 * h, h) * 0.2).requires_grad_(True)
        wb = (torch.randn(4 * h, h) * 0.2).requires_grad_(True)
        bf = torch.randn(4 * h).requires_grad_(True)
        bb = torch.randn(4 * h).requires_grad_(True)
        return xf, xb, wf, wb, bf, bb

    h0 = torch.zeros(b, h)
    c0 = torch.zeros(b, h)

    xf1, xb1, wf1, wb1, bf1, bb1 = inputs()
    out_ref = torch.cat(
        [
            lstm_sequence(xf1, h0, c0, wf1, bf1, mk, reverse=False),
            lstm_sequence(xb1, h0, c0, wb1, bb1, mk, reverse=True),
        ],
        dim=-1,
    )
    gseed = torch.Generator().manual_seed(3)
    g = torch.randn(out_ref.shape, generator=gseed)
    out_ref.backward(g)

    xf2, xb2, wf2, wb2, bf2, bb2 = inputs()
    out = lstm_bilayer(xf2, xb2, h0, c0, wf2, bf2, wb2, bb2, mk)
    torch.testing.assert_close(out, out_ref)
    out.backward(g)
    for a, bT, name in [
        (xf1, xf2, "xg_f"), (xb1, xb2, "xg_b"), (wf1, wf2, "w_f"),
        (wb1, wb2, "w_b"), (bf1, bf2, "b_f"), (bb1, bb2, "b_b"),
    ]:
        torch.testing.assert_close(a.grad, bT.grad, msg=name)
