"""Data layer tests: synth generator, loaders, labels, sequences."""
import numpy as np

from nerrf_amd.data.labels import event_labels, malicious_pids
from nerrf_amd.data.sequences import build_sequences
from nerrf_amd.data.synth import SynthConfig, generate
from nerrf_amd.data.trace import SYSCALL_IDS, load_csv, load_trace, write_csv


def test_synth_deterministic():
    a1, w1 = generate(SynthConfig(seed=7, duration_s=30))
    a2, w2 = generate(SynthConfig(seed=7, duration_s=30))
    assert len(a1) == len(a2)
    assert np.array_equal(a1.ts, a2.ts)
    assert np.array_equal(a1.syscall, a2.syscall)
    assert w1.t_start == w2.t_start


def test_synth_time_sorted():
    arr, _ = generate(SynthConfig(seed=3, duration_s=40))
    assert np.all(np.diff(arr.ts) >= 0)


def test_synth_attack_pattern():
    cfg = SynthConfig(seed=1, duration_s=60, n_victim_files=10)
    arr, win = generate(cfg)
    assert win is not None
    y = event_labels(arr, win)
    assert y.sum() > 0
    # every victim file sees rename + unlink inside the window
    n_ren = int(((arr.syscall == SYSCALL_IDS["rename"]) & (y > 0.5)).sum())
    n_unl = int(((arr.syscall == SYSCALL_IDS["unlink"]) & (y > 0.5)).sum())
    assert n_ren == cfg.n_victim_files
    assert n_unl == cfg.n_victim_files
    assert set(malicious_pids(arr, y).tolist()) == {6666}


def test_no_attack_no_labels():
    arr, win = generate(SynthConfig(seed=2, attack=False, duration_s=30))
    assert win is None
    assert event_labels(arr, win).sum() == 0


def test_csv_roundtrip(tmp_path):
    arr, _ = generate(SynthConfig(seed=5, duration_s=20, benign_rate_hz=50))
    p = tmp_path / "t.csv"
    write_csv(p, arr)
    back = load_csv(p)
    assert len(back) == len(arr)
    assert np.allclose(back.ts, arr.ts, atol=1e-5)
    assert np.array_equal(back.syscall, arr.syscall)
    assert np.array_equal(back.nbytes, arr.nbytes)


def test_load_trace_dispatch(tmp_path):
    arr, _ = generate(SynthConfig(seed=5, duration_s=10, benign_rate_hz=20))
    p = tmp_path / "t.csv"
    write_csv(p, arr)
    assert len(load_trace(p)) == len(arr)


def test_jsonl_loader(tmp_path):
    p = tmp_path / "t.jsonl"
    p.write_text(
        '{"timestamp": 1.5, "event": "write", "path": "/a/b.dat", "size": 100, "pid": 3}\n'
        '{"timestamp": 2.5, "event": "rename", "path": "/a/b.dat", "new_path": "/a/b.lockbit3", "pid": 3}\n'
        '{"timestamp": 0.5, "event": "open", "path": "/a/b.dat", "pid": 3}\n'
    )
    arr = load_trace(p)
    assert len(arr) == 3
    # sorted by time; "open" normalised to openat
    assert arr.syscall[0] == SYSCALL_IDS["openat"]
    assert arr.syscall[1] == SYSCALL_IDS["write"]
    assert arr.syscall[2] == SYSCALL_IDS["rename"]
    assert arr.new_path_id[2] >= 0


def test_sequences_shapes_and_labels():
    arr, win = generate(SynthConfig(seed=9, duration_s=60, n_victim_files=8))
    y = event_labels(arr, win)
    seqs = build_sequences(arr, y, seq_len=100)
    assert seqs.feats.shape[1] == 100
    assert seqs.feats.shape[2] == 16
    assert (seqs.lengths >= 2).all()
    assert (seqs.lengths <= 100).all()
    assert seqs.labels is not None and seqs.labels.sum() >= 8  # victim files flagged
    # one-hot rows sum to 1 for valid steps
    for bi in range(min(4, len(seqs.lengths))):
        t = seqs.lengths[bi]
        assert np.allclose(seqs.feats[bi, :t, :10].sum(axis=1), 1.0)
        assert np.allclose(seqs.feats[bi, t:], 0.0)


def test_toy_trace_exists_and_loads():
    arr = load_trace("datasets/traces/toy_trace.csv")
    assert len(arr) > 100


def test_event_array_time_window_and_slice():
    arr, _ = generate(SynthConfig(seed=6, duration_s=40, benign_rate_hz=50))
    w = arr.time_window(10.0, 20.0)
    assert (w.ts >= 10.0).all() and (w.ts < 20.0).all()
    s = arr.slice(5, 15)
    assert len(s) == 10
    assert s.paths is arr.paths  # tables shared, not copied


def test_string_table_roundtrip():
    from nerrf_amd.data.trace import StringTable

    t = StringTable()
    a = t.intern("/x/y")
    b = t.intern("/x/z")
    assert t.intern("/x/y") == a  # stable ids
    assert t.lookup(b) == "/x/z"
    assert t.get("/nope") is None
    assert len(t) == 2


def test_concat_event_arrays():
    from nerrf_amd.data.trace import concat

    a1, _ = generate(SynthConfig(seed=1, duration_s=10, benign_rate_hz=20, attack=False))
    a2, _ = generate(SynthConfig(seed=2, duration_s=10, benign_rate_hz=20, attack=False))
    c = concat([a1, a2])
    assert len(c) == len(a1) + len(a2)
    assert (c.ts[:-1] <= c.ts[1:]).all()


def test_perf_noop_on_cpu():
    import torch

    from nerrf_amd.perf import enable_tuned_gemms

    if not torch.cuda.is_available():
        assert enable_tuned_gemms() is False


def test_build_sequences_torch_matches_numpy():
    """The torch (GPU-path) sequence builder reproduces the numpy build
    bit-for-bit on CPU tensors: same grouping, lengths, ids and features."""
    import numpy as np
    import torch

    from nerrf_amd.data.sequences import build_sequences, build_sequences_torch
    from nerrf_amd.data.synth import SynthConfig, generate

    arr, _ = generate(SynthConfig(duration_s=6.0, benign_rate_hz=2000.0,
                                  n_benign_files=150, seed=11))
    ref = build_sequences(arr, None)
    feats, lengths, fids = build_sequences_torch(arr, device="cpu")
    assert feats.shape == ref.feats.shape
    assert np.array_equal(lengths.numpy(), ref.lengths)
    assert np.array_equal(fids.numpy(), ref.file_path_id)
    torch.testing.assert_close(feats, torch.from_numpy(ref.feats), rtol=1e-6, atol=1e-6)


def test_build_sequences_torch_empty():
    import torch

    from nerrf_amd.data.sequences import build_sequences_torch
    from nerrf_amd.data.trace import EventArrayBuilder, StringTable

    ev = EventArrayBuilder(StringTable(), StringTable()).build()
    feats, lengths, fids = build_sequences_torch(ev, device="cpu")
    assert feats.shape[0] == 0 and lengths.numel() == 0


def test_load_jsonl_tolerates_corrupt_lines(tmp_path):
    p = tmp_path / "t.jsonl"
    p.write_text(
        '{"timestamp": 1.0, "event": "write", "path": "/a", "size": 4, "pid": 9}\n'
        "{this is not json\n"
        '{"timestamp": 2.0, "event": "read", "path": "/a", "size": 2, "pid": 9}\n'
    )
    from nerrf_amd.data.trace import load_trace

    arr = load_trace(p)
    assert len(arr) == 2


def test_write_csv_load_csv_roundtrip(tmp_path):
    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.data.trace import load_trace, write_csv

    arr, _ = generate(SynthConfig(duration_s=4.0, benign_rate_hz=200.0,
                                  n_benign_files=30, attack=False, seed=6))
    p = tmp_path / "rt.csv"
    write_csv(p, arr)
    back = load_trace(p)
    assert len(back) == len(arr)
    assert np.allclose(back.ts, arr.ts, atol=1e-5)
    assert np.array_equal(back.syscall, arr.syscall)
    assert np.array_equal(back.nbytes, arr.nbytes)
    # path strings survive (ids may be renumbered)
    a = [arr.paths.lookup(int(i)) if i >= 0 else "" for i in arr.path_id[:200]]
    b = [back.paths.lookup(int(i)) if i >= 0 else "" for i in back.path_id[:200]]
    assert a == b


def test_export_dataset_parquet_roundtrip(tmp_path):
    """tools/export_dataset.py: labelled per-event parquet + manifest,
    deterministic, attack rows labelled inside their ground-truth window."""
    import json
    import sys

    sys.path.insert(0, "tools")
    from export_dataset import export

    meta = export(str(tmp_path), n_scenarios=3, duration_s=30.0,
                  benign_rate_hz=100.0, base_seed=5)
    assert meta["rows"] > 0
    assert len(meta["scenarios"]) == 3
    import pyarrow.parquet as pq

    t = pq.read_table(tmp_path / "events.parquet")
    assert t.num_rows == meta["rows"]
    df = t.to_pydict()
    assert set(df["kind"]) == {"lockbit", "supply_chain", "supply_chain_net"}
    # lockbit scenario has labelled attack events, all inside the window
    man = json.loads((tmp_path / "manifest.json").read_text())
    lb = next(s for s in man["scenarios"] if s["kind"] == "lockbit")
    import numpy as np

    kinds = np.array(df["kind"])
    labels = np.array(df["label"])
    ts = np.array(df["ts"])
    sel = (kinds == "lockbit") & (labels == 1)
    assert sel.sum() > 0
    w = lb["attack_window"]
    assert ts[sel].min() >= w["t_start"] - 1e-6
    assert ts[sel].max() <= w["t_end"] + 1e-6
    # determinism
    meta2 = export(str(tmp_path / "again"), n_scenarios=3, duration_s=30.0,
                   benign_rate_hz=100.0, base_seed=5)
    assert meta2["rows"] == meta["rows"]
