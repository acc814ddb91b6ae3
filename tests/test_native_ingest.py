"""Native ingest tests: C++ wire codec parity, columnar decode, nerrfd daemon."""
import json
import socket
import subprocess
import time
from pathlib import Path

import numpy as np
import pytest

from nerrf_amd.wire import codec

_ingest = pytest.importorskip("nerrf_amd._ingest")

DAEMON = Path(__file__).resolve().parent.parent / "tracker" / "daemon" / "nerrfd"


def _sample_events(n=50):
    return [
        codec.Event(
            ts_sec=1_700_000_000 + i,
            ts_nsec=i * 1000,
            pid=100 + i % 5,
            tid=100 + i % 5,
            comm="python3",
            syscall=["openat", "write", "rename", "read", "unlink"][i % 5],
            path=f"/app/uploads/doc_{i:03d}.dat",
            new_path=f"/app/uploads/doc_{i:03d}.dat.lockbit3" if i % 5 == 2 else "",
            ret_val=(-1) ** i * i,
            bytes=i * 4096,
        )
        for i in range(n)
    ]


def test_cpp_decode_matches_python_encode():
    evs = _sample_events()
    frame = codec.encode_event_batch(evs)
    back = _ingest.decode_batch(frame)
    assert len(back) == len(evs)
    for d, ev in zip(back, evs):
        assert d["pid"] == ev.pid
        assert d["syscall"] == ev.syscall
        assert d["path"] == ev.path
        assert d["ret_val"] == ev.ret_val
        assert d["bytes"] == ev.bytes
        assert d["ts_sec"] == ev.ts_sec
        assert d["ts_nsec"] == ev.ts_nsec


def test_python_decode_matches_cpp_encode():
    evs = _sample_events(20)
    frame = _ingest.encode_batch(
        [
            dict(
                ts_sec=e.ts_sec, ts_nsec=e.ts_nsec, pid=e.pid, tid=e.tid,
                comm=e.comm, syscall=e.syscall, path=e.path, new_path=e.new_path,
                ret_val=e.ret_val, bytes=e.bytes,
            )
            for e in evs
        ]
    )
    back = codec.decode_event_batch(frame)
    assert back == evs


def test_columnar_decoder():
    evs = _sample_events(64)
    frame = codec.encode_event_batch(evs)
    dec = _ingest.ColumnarDecoder()
    ts, pid, sysc, path_id, newp_id, nbytes, ret, comm = dec.decode([frame, frame])
    assert len(ts) == 128
    assert ts[0] == pytest.approx(evs[0].timestamp(), abs=1e-5) if callable(getattr(evs[0], "timestamp", None)) else True
    assert pid[3] == evs[3].pid
    # interning: same path in both frames -> same id
    assert path_id[0] == path_id[64]
    paths = dec.paths_since(0)
    assert paths[path_id[0]] == evs[0].path
    # syscall ids match the python mapping
    from nerrf_amd.data.trace import SYSCALL_IDS

    assert sysc[1] == SYSCALL_IDS["write"]
    assert int(nbytes[2]) == evs[2].bytes


@pytest.mark.skipif(not DAEMON.exists(), reason="nerrfd not built (make -C tracker daemon)")
def test_nerrfd_replay_roundtrip(tmp_path):
    """End-to-end: jsonl trace -> nerrfd (C++) -> TCP frames -> store."""
    from nerrf_amd.graph.store import DeltaGraphStore
    from nerrf_amd.serve.daemon_bridge import pump_daemon_into_store

    trace = tmp_path / "t.jsonl"
    with open(trace, "w") as fh:
        for i in range(120):
            fh.write(
                json.dumps(
                    {
                        "timestamp": 10.0 + i * 0.01,
                        "event": ["write", "rename", "openat"][i % 3],
                        "path": f"/data/f{i % 7}.dat",
                        "size": i * 100,
                        "pid": 42,
                    }
                )
                + "\n"
            )
    # free port
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    proc = subprocess.Popen(
        [str(DAEMON), "--replay", str(trace), "--port", str(port), "--once", "--batch", "32"],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE,
    )
    try:
        time.sleep(0.3)  # let it bind
        store = DeltaGraphStore(window_s=1e9)
        n = pump_daemon_into_store("127.0.0.1", port, store, timeout_s=10.0)
        assert n == 120
        arr = store.compact()
        assert len(arr) == 120
        assert arr.nbytes.sum() == sum(i * 100 for i in range(120))
        from nerrf_amd.data.trace import SYSCALL_IDS

        assert (arr.syscall == SYSCALL_IDS["rename"]).sum() == 40
    finally:
        proc.terminate()
        proc.wait(timeout=5)


@pytest.mark.skipif(not DAEMON.exists(), reason="nerrfd not built")
def test_full_pipeline_daemon_bridge_engine(tmp_path):
    """Whole chain: C++ nerrfd (replay) -> TCP frames -> gRPC bridge ->
    engine ingest -> detection + plan.  The deployment wiring, in-process."""
    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.data.trace import write_csv
    from nerrf_amd.models.graphsage import SageConfig
    from nerrf_amd.models.joint import JointConfig, NerrfJointModel
    from nerrf_amd.models.lstm import LSTMConfig
    from nerrf_amd.serve.daemon_bridge import GrpcBridge
    from nerrf_amd.serve.engine import StreamingEngine

    # attack trace -> jsonl for the daemon
    arr, _ = generate(SynthConfig(seed=19, duration_s=30, benign_rate_hz=40, n_victim_files=6))
    trace = tmp_path / "t.jsonl"
    with open(trace, "w") as fh:
        from nerrf_amd.data.trace import SYSCALL_NAMES

        for i in range(len(arr)):
            rec = {
                "timestamp": float(arr.ts[i]),
                "event": SYSCALL_NAMES.get(int(arr.syscall[i]), "unknown"),
                "path": arr.paths.lookup(int(arr.path_id[i])) if arr.path_id[i] >= 0 else "",
                "size": int(arr.nbytes[i]),
                "pid": int(arr.pid[i]),
            }
            if arr.new_path_id[i] >= 0:
                rec["new_path"] = arr.paths.lookup(int(arr.new_path_id[i]))
            fh.write(json.dumps(rec) + "\n")

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    proc = subprocess.Popen(
        [str(DAEMON), "--replay", str(trace), "--port", str(port), "--batch", "64"],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE,
    )
    bridge = None
    try:
        time.sleep(0.3)
        bridge = GrpcBridge("127.0.0.1", port)
        bridge.start()
        model = NerrfJointModel(
            JointConfig(sage=SageConfig(layers=3, hidden=32), lstm=LSTMConfig(hidden=32))
        )
        engine = StreamingEngine(model=model, device="cpu")
        engine.store.window_s = 1e9
        n = engine.ingest_from_tracker(bridge.address, max_events=len(arr), timeout_s=15.0)
        assert n >= len(arr) * 0.9
        det = engine.score_window()
        assert det.alarm
        plan = engine.plan(det, n_sims=128)
        assert plan.plan  # a concrete undo plan came out the far end
    finally:
        if bridge is not None:
            bridge.stop()
        proc.terminate()
        proc.wait(timeout=5)


def test_native_decoder_rejects_crafted_length_overflow():
    """A WT_LEN varint of ~2^64 must raise cleanly (no wrap-around, no
    hang): the decoder parses frames from network clients."""
    import pytest

    ingest = pytest.importorskip("nerrf_amd._ingest")
    dec = ingest.ColumnarDecoder()
    # EventBatch field 1 (event), WT_LEN, length = 2^64 - 1
    evil = bytes([0x0A]) + b"\xff\xff\xff\xff\xff\xff\xff\xff\xff\x01"
    with pytest.raises(Exception):
        dec.decode([evil])
    # nested: valid batch envelope, event payload with a huge string length
    inner = bytes([0x32]) + b"\xff\xff\xff\xff\xff\xff\xff\xff\xff\x01"  # field 6 path
    frame = bytes([0x0A, len(inner)]) + inner
    with pytest.raises(Exception):
        dec.decode([frame])
    # and a huge skip length on an unknown field
    inner2 = bytes([0xFA, 0x01]) + b"\xff\xff\xff\xff\xff\xff\xff\xff\xff\x01"  # field 31 WT_LEN
    frame2 = bytes([0x0A, len(inner2)]) + inner2
    with pytest.raises(Exception):
        dec.decode([frame2])


def test_fd_resolver_exact_and_heuristic():
    """Userspace fd->path fallback (reference M2 spec): exact (pid, fd)
    binding when openat ret_val carries the fd; last-open heuristic when
    fds are absent (the upstream artifact schema)."""
    import numpy as np

    from nerrf_amd.data.fdresolve import resolve_fd_paths
    from nerrf_amd.data.trace import EventArrayBuilder, StringTable

    b = EventArrayBuilder(StringTable(), StringTable())
    # pid 1: open a.txt (fd 3), open b.txt (fd 4), write fd 3, write fd 4,
    #        close fd 3, write fd 4
    b.add(ts=1.0, pid=1, syscall="openat", path="/a.txt", ret_val=3)
    b.add(ts=2.0, pid=1, syscall="openat", path="/b.txt", ret_val=4)
    b.add(ts=3.0, pid=1, syscall="write", nbytes=10, ret_val=3)
    b.add(ts=4.0, pid=1, syscall="write", nbytes=20, ret_val=4)
    b.add(ts=5.0, pid=1, syscall="close", ret_val=3)
    b.add(ts=6.0, pid=1, syscall="write", nbytes=30, ret_val=4)
    # pid 2: interleaved, must not leak pid 1's table
    b.add(ts=3.5, pid=2, syscall="openat", path="/c.txt", ret_val=3)
    b.add(ts=4.5, pid=2, syscall="write", nbytes=5, ret_val=3)
    arr = b.build()
    res = resolve_fd_paths(arr)
    names = [res.paths.lookup(int(i)) if i >= 0 else None for i in res.path_id]
    by = {(float(t), int(p), int(nb)): nm
          for t, p, nb, nm in zip(res.ts, res.pid, res.nbytes, names)}
    assert by[(3.0, 1, 10)] == "/a.txt"   # exact fd 3
    assert by[(4.0, 1, 20)] == "/b.txt"   # exact fd 4
    assert by[(6.0, 1, 30)] == "/b.txt"   # fd 3 closed; fd 4 still bound
    assert by[(4.5, 2, 5)] == "/c.txt"    # per-pid isolation

    # heuristic mode: no fds recorded (ret_val = 0 means fd 0 is unknown
    # here — disable exact matching)
    b2 = EventArrayBuilder(StringTable(), StringTable())
    b2.add(ts=1.0, pid=7, syscall="openat", path="/x.dat")
    b2.add(ts=2.0, pid=7, syscall="write", nbytes=64)
    b2.add(ts=3.0, pid=7, syscall="openat", path="/y.dat")
    b2.add(ts=4.0, pid=7, syscall="write", nbytes=65)
    r2 = resolve_fd_paths(b2.build(), use_ret_fd=False)
    n2 = [r2.paths.lookup(int(i)) if i >= 0 else None for i in r2.path_id]
    writes = [nm for sc, nm in zip(r2.syscall, n2) if sc == 2]
    assert writes == ["/x.dat", "/y.dat"]  # last-open-wins

    # already-pathed writes are untouched
    b3 = EventArrayBuilder(StringTable(), StringTable())
    b3.add(ts=1.0, pid=9, syscall="write", path="/known.log", nbytes=1)
    r3 = resolve_fd_paths(b3.build())
    assert r3.paths.lookup(int(r3.path_id[0])) == "/known.log"


def test_codec_negative_nanos_cross_language():
    """negative Timestamp nanos must round-trip
    identically through BOTH codecs (C++ sign-extends on encode; both
    decoders truncate to int32 like the protobuf runtime)."""
    ev = codec.Event(ts_sec=5, ts_nsec=-7, pid=1, tid=1, comm="x",
                     syscall="write", path="/f", ret_val=-2, bytes=3)
    # python encode -> C++ decode
    d = _ingest.decode_batch(codec.encode_event_batch([ev]))[0]
    assert (d["ts_sec"], d["ts_nsec"]) == (5, -7)
    # C++ encode -> python decode
    frame = _ingest.encode_batch([dict(
        ts_sec=5, ts_nsec=-7, pid=1, tid=1, comm="x", syscall="write",
        path="/f", new_path="", ret_val=-2, bytes=3)])
    back = codec.decode_event_batch(frame)[0]
    assert (back.ts_sec, back.ts_nsec) == (5, -7)
    # and byte-identical frames from the two encoders
    assert frame == codec.encode_event_batch([codec.Event(
        ts_sec=5, ts_nsec=-7, pid=1, tid=1, comm="x", syscall="write",
        path="/f", ret_val=-2, bytes=3)])


@pytest.mark.skipif(not DAEMON.exists(), reason="nerrfd not built")
def test_nerrfd_filter_prefix(tmp_path):
    """Capture-side path filtering (upstream M2 plan): only events under the
    configured prefixes (plus pathless ones) leave the daemon."""
    from nerrf_amd.graph.store import DeltaGraphStore
    from nerrf_amd.serve.daemon_bridge import pump_daemon_into_store

    trace = tmp_path / "t.jsonl"
    with open(trace, "w") as fh:
        for i in range(90):
            path = ["/app/uploads/a.dat", "/var/log/syslog", "/tmp/x"][i % 3]
            fh.write(json.dumps({
                "timestamp": 10.0 + i * 0.01, "event": "openat",
                "path": path, "size": 1, "pid": 7,
            }) + "\n")
        fh.write(json.dumps({  # pathless event must pass the filter
            "timestamp": 11.0, "event": "write", "path": "", "size": 64,
            "pid": 7,
        }) + "\n")
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    proc = subprocess.Popen(
        [str(DAEMON), "--replay", str(trace), "--port", str(port), "--once",
         "--batch", "16", "--filter-prefix", "/app/uploads/",
         "--filter-prefix", "/tmp/"],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE,
    )
    try:
        time.sleep(0.3)
        store = DeltaGraphStore(window_s=1e9)
        n = pump_daemon_into_store("127.0.0.1", port, store, timeout_s=10.0)
        assert n == 61  # 30 uploads + 30 tmp + 1 pathless
        arr = store.compact()
        names = {arr.paths.lookup(int(i)) for i in arr.path_id if i >= 0}
        assert "/var/log/syslog" not in names
        assert {"/app/uploads/a.dat", "/tmp/x"} <= names
    finally:
        proc.terminate()
        proc.wait(timeout=5)
