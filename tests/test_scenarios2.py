"""Supply-chain scenario + multi-shard detection merge."""
import os

import numpy as np
import pytest
import torch.distributed as dist
import torch.multiprocessing as mp

from nerrf_amd.harness.supply_chain import (
    recover_supply_chain,
    run_supply_chain_attack,
    seed_app,
)


def test_supply_chain_attack_and_recovery(tmp_path):
    manifest = seed_app(tmp_path, n_deps=10, n_data=5, seed=3)
    rep = run_supply_chain_attack(tmp_path, trace_path=tmp_path / "t.jsonl")
    assert len(rep.backdoored) == 10
    assert rep.bytes_staged == 5 * 24 * 1024
    # backdoor present before recovery
    first = open(rep.backdoored[0], "rb").read()
    assert first.startswith(b"// postinstall payload")
    out = recover_supply_chain(tmp_path, rep, manifest)
    assert out["recovered_ok"]
    assert out["restored_deps"] == 10
    assert out["blob_quarantined"]
    assert out["data_loss_mb"] == 0.0


def test_supply_chain_graph_signature(tmp_path):
    """No LockBit indicators fire, but the graph exposes the attacker:
    one process with maximal out-degree touching deps + data + blob."""
    from nerrf_amd.data.trace import load_trace
    from nerrf_amd.graph.constructor import build_graph

    seed_app(tmp_path, n_deps=8, n_data=4, seed=1)
    rep = run_supply_chain_attack(tmp_path, trace_path=tmp_path / "t.jsonl", pid=7777)
    arr = load_trace(tmp_path / "t.jsonl")
    g = build_graph(arr)
    # no suspicious-extension / ransom-note flags anywhere
    assert g.x[:, 13].max() == 0.0
    assert g.x[:, 20].max() == 0.0
    # the attacker process node has the window's max out-degree
    proc_nodes = np.nonzero(g.node_kind == 0)[0]
    attacker = [n for n in proc_nodes if g.node_key[n] == 7777]
    assert attacker
    assert g.x[attacker[0], 3] == g.x[proc_nodes, 3].max()


def _worker_merge(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from nerrf_amd.data.synth import SynthConfig, generate
        from nerrf_amd.models.graphsage import SageConfig
        from nerrf_amd.models.joint import JointConfig, NerrfJointModel
        from nerrf_amd.models.lstm import LSTMConfig
        from nerrf_amd.serve.engine import StreamingEngine

        model = NerrfJointModel(JointConfig(sage=SageConfig(layers=2, hidden=24), lstm=LSTMConfig(hidden=16)))
        engine = StreamingEngine(model=model, device="cpu")
        engine.store.window_s = 1e9
        # only rank 1's shard sees the attack
        arr, _ = generate(SynthConfig(seed=40 + rank, duration_s=25, benign_rate_hz=40, attack=(rank == 1)))
        engine.ingest_events(arr)
        det = engine.score_window()
        merged = engine.merge_detections(det)
        results[rank] = {
            "local_alarm": det.alarm,
            "merged_alarm": merged.alarm,
            "merged_events": merged.window_events,
            "merged_enc": len(merged.encrypted_paths),
        }
    finally:
        dist.destroy_process_group()


def test_sharded_detection_merge():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_merge, args=(2, port, results), nprocs=2, join=True)
    # rank 0 saw a clean shard locally but the merged view raises the alarm
    assert results[0]["local_alarm"] is False
    assert results[0]["merged_alarm"] is True
    assert results[1]["merged_alarm"] is True
    assert results[0]["merged_events"] == results[1]["merged_events"]
    assert results[0]["merged_enc"] > 0


def test_fp_undo_zero_on_hard_negatives(tmp_path):
    """FP-undo target: on benign lookalikes (log rotation, backup daemon,
    compiler walking a dependency tree)
    the response loop takes ZERO destructive actions — even when the model
    raises scores, there is nothing the executor will touch and the sandbox
    gate refuses an empty restore."""
    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.serve.engine import StreamingEngine, load_model_from_checkpoint

    model = load_model_from_checkpoint("checkpoints/pretrained")
    for kind in ("benign_rotate", "benign_backup", "benign_build"):
        engine = StreamingEngine(model=model, device="cpu")
        engine.store.window_s = 1e9
        arr, _ = generate(SynthConfig(seed=91, duration_s=40, benign_rate_hz=50, kind=kind))
        engine.ingest_events(arr)
        det = engine.score_window()
        # rule indicators must stay silent on these
        assert det.indicators["suspicious_ext_count"] == 0
        assert det.indicators["ransom_note"] == 0
        # a victim dir with only clean files: respond() must not alter it
        victim = tmp_path / kind
        victim.mkdir()
        (victim / "a.dat").write_bytes(b"x" * 64)
        plan = engine.plan(det, n_sims=256)
        res = engine.respond(det, plan, str(victim))
        assert res.files_restored == 0
        assert (victim / "a.dat").read_bytes() == b"x" * 64  # untouched


def test_socket_nodes_and_exfil_destination_alarm():
    """supply_chain_net: the exfil socket becomes a kind-2 node, the
    destination-allowlist indicator fires, and the engine alarms even
    without a trained model (defence in depth)."""
    import numpy as np

    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.graph.constructor import build_graph
    from nerrf_amd.serve.engine import StreamingEngine

    arr, w = generate(SynthConfig(kind="supply_chain_net", duration_s=20.0,
                                  benign_rate_hz=120.0, seed=5))
    g = build_graph(arr, window=w)
    sock = np.nonzero(g.node_kind == 2)[0]
    assert len(sock) == 1
    assert g.x[sock, 13] == 1.0  # unlisted destination -> suspicious channel
    assert g.x[sock, 0] == 0.0 and g.x[sock, 1] == 0.0  # one-hot: neither

    eng = StreamingEngine(device="cpu")
    eng.store.window_s = 1e9
    eng.ingest_events(arr)
    det = eng.score_window()
    assert det.indicators["exfil_dest_count"] == 1.0
    assert det.exfil_destinations == ["tcp://203.0.113.37:443"]
    assert det.alarm


def test_allowlisted_destination_not_flagged():
    """Egress to an allowlisted destination (e.g. the backup server) does
    not set the suspicious channel or the exfil indicator."""
    import numpy as np

    from nerrf_amd.data.trace import EventArrayBuilder, StringTable
    from nerrf_amd.graph.constructor import build_graph
    from nerrf_amd.serve.engine import StreamingEngine

    b = EventArrayBuilder(StringTable(), StringTable())
    t = 0.0
    for i in range(40):
        b.add(ts=t, pid=42, syscall="read", path=f"/srv/data/f{i}.db", nbytes=4096)
        t += 0.01
        b.add(ts=t, pid=42, syscall="sendto", path="tcp://10.0.0.9:873", nbytes=4096)
        t += 0.01
    arr = b.build()
    g = build_graph(arr)
    sock = np.nonzero(g.node_kind == 2)[0]
    assert len(sock) == 1
    assert g.x[sock, 13] == 0.0  # allowlisted -> clean

    eng = StreamingEngine(device="cpu")
    eng.store.window_s = 1e9
    eng.ingest_events(arr)
    det = eng.score_window()
    assert det.indicators["exfil_dest_count"] == 0.0
    assert det.exfil_destinations == []


def test_sharded_detection_merge_world4():
    """Config-5 shard merge at a deeper fan-out (4 shards, gloo) — the
    8-GPU serving layout differs from this only in backend."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_merge, args=(4, port, results), nprocs=4, join=True)
    assert results[0]["merged_alarm"] is True
    assert all(results[r]["merged_alarm"] for r in range(4))
    assert len({results[r]["merged_events"] for r in range(4)}) == 1
