"""Graph constructor + sampling tests."""
import numpy as np
import pytest

from nerrf_amd.data.labels import event_labels
from nerrf_amd.data.synth import SynthConfig, generate
from nerrf_amd.data.trace import EventArrayBuilder
from nerrf_amd.graph.constructor import NUM_NODE_FEATURES, build_graph, sliding_windows
from nerrf_amd.graph.sampling import sample_fanout, to_csr


def _mini_trace():
    b = EventArrayBuilder()
    b.add(ts=1.0, pid=10, syscall="openat", path="/d/a.dat")
    b.add(ts=2.0, pid=10, syscall="write", path="/d/a.dat", nbytes=100)
    b.add(ts=3.0, pid=10, syscall="rename", path="/d/a.dat", new_path="/d/a.dat.lockbit3")
    b.add(ts=4.0, pid=11, syscall="read", path="/d/b.dat", nbytes=50)
    return b.build()


def test_rename_union_merges_file_nodes():
    arr = _mini_trace()
    g = build_graph(arr)
    # /d/a.dat and /d/a.dat.lockbit3 are ONE node; /d/b.dat another; 2 procs
    n_files = int((g.node_kind == 1).sum())
    n_procs = int((g.node_kind == 0).sum())
    assert n_files == 2
    assert n_procs == 2


def test_feature_matrix_shape_and_indicators():
    arr = _mini_trace()
    g = build_graph(arr)
    assert g.x.shape == (g.num_nodes, NUM_NODE_FEATURES)
    assert np.isfinite(g.x).all()
    # the merged a.dat node carries the suspicious-extension flag (col 13)
    file_nodes = np.nonzero(g.node_kind == 1)[0]
    assert g.x[file_nodes, 13].max() == 1.0


def test_edge_directions():
    arr = _mini_trace()
    g = build_graph(arr)
    # writes: proc->file; reads: file->proc => both directions present
    src_kind = g.node_kind[g.edge_index[0]]
    dst_kind = g.node_kind[g.edge_index[1]]
    assert ((src_kind == 0) & (dst_kind == 1)).any()  # proc -> file
    assert ((src_kind == 1) & (dst_kind == 0)).any()  # file -> proc
    assert (g.edge_weight > 0).all() and (g.edge_weight <= 1).all()


def test_labels_propagate_to_nodes_and_edges():
    arr, win = generate(SynthConfig(seed=11, duration_s=60, n_victim_files=6))
    y = event_labels(arr, win)
    g = build_graph(arr, win, y)
    assert g.y_node is not None and g.y_node.sum() >= 6
    assert g.y_edge is not None and g.y_edge.sum() > 0
    # malicious edges connect two malicious endpoints
    bad = g.y_edge > 0.5
    assert (g.y_node[g.edge_index[0][bad]] > 0.5).all()


def test_sliding_windows_cover_trace():
    arr, _ = generate(SynthConfig(seed=4, duration_s=90, benign_rate_hz=100))
    wins = list(sliding_windows(arr, window_s=30, stride_s=15))
    assert len(wins) >= 5
    total = sum(len(w) for _, w in wins)
    assert total >= len(arr)  # overlap counts events twice


def test_csr_and_fanout():
    arr, _ = generate(SynthConfig(seed=6, duration_s=40))
    g = build_graph(arr)
    csr = to_csr(g.edge_index, g.num_nodes, g.edge_weight)
    assert csr.indptr[-1] == len(csr.indices)
    idx, w = sample_fanout(csr, 16, seed=1)
    assert idx.shape == (g.num_nodes, 16)
    assert w.shape == (g.num_nodes, 16)
    assert (idx >= 0).all() and (idx < g.num_nodes).all()
    assert (w >= 0).all()
    # determinism
    idx2, w2 = sample_fanout(csr, 16, seed=1)
    assert np.array_equal(idx, idx2) and np.array_equal(w, w2)
    # sampled ids are actual in-neighbors (or self for isolated)
    deg = np.diff(csr.indptr)
    for n in range(0, g.num_nodes, max(1, g.num_nodes // 17)):
        if deg[n] == 0:
            assert (idx[n] == n).all()
        else:
            nbrs = set(csr.indices[csr.indptr[n] : csr.indptr[n + 1]].tolist())
            assert set(idx[n].tolist()) <= nbrs


def test_empty_trace():
    b = EventArrayBuilder()
    arr = b.build()
    g = build_graph(arr)
    assert g.num_nodes == 0
    assert g.num_edges == 0


def test_rename_components_match_serial_union_find():
    """The scipy connected-components root assignment reproduces the serial
    min-parent union-find (_UnionFind) on a randomized rename graph."""
    import numpy as np

    from nerrf_amd.data.trace import EventArrayBuilder, StringTable, SYSCALL_IDS
    from nerrf_amd.graph.constructor import _UnionFind, build_graph_parts

    rng = np.random.default_rng(7)
    n_paths = 300
    paths = StringTable()
    names = [f"/d/f{i}" for i in range(n_paths)]
    b = EventArrayBuilder(paths, StringTable())
    pairs = []
    t = 0.0
    for _ in range(180):
        i, j = rng.integers(0, n_paths, size=2)
        pairs.append((int(i), int(j)))
        b.add(ts=t, pid=10, syscall="rename", path=names[i], new_path=names[j])
        t += 0.01
    # a few touches so every path id exists in the table
    for i in range(n_paths):
        b.add(ts=t, pid=10, syscall="write", path=names[i], nbytes=1)
        t += 0.001
    ev = b.build()
    parts = build_graph_parts(ev)

    uf = _UnionFind(len(ev.paths))
    ren = (ev.syscall == SYSCALL_IDS["rename"]) & (ev.new_path_id >= 0)
    for a, c in zip(ev.path_id[ren], ev.new_path_id[ren]):
        uf.union(int(a), int(c))
    expect = np.array([uf.find(i) for i in range(len(ev.paths))])
    assert np.array_equal(parts["path_root"], expect)


def test_edge_dedup_radix_fallback_matches_bincount():
    """Large key spaces take the LSD-radix branch; results must match the
    bincount branch (and the original np.unique semantics)."""
    import numpy as np

    from nerrf_amd.data.trace import EventArrayBuilder, StringTable
    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts

    rng = np.random.default_rng(3)
    n_procs_target, n_files_target = 3000, 3000  # 2*3000*3000 = 18M > 2**24
    b = EventArrayBuilder(StringTable(), StringTable())
    t = 0.0
    for _ in range(20000):
        p = int(rng.integers(0, n_procs_target))
        f = int(rng.integers(0, n_files_target))
        b.add(ts=t, pid=1000 + p, syscall="write" if rng.random() < 0.7 else "read",
              path=f"/data/f{f}", nbytes=64)
        t += 1e-4
    ev = b.build()
    parts = build_graph_parts(ev)
    assert 2 * parts["n_procs"] * parts["n_files"] >= (1 << 24)
    ed = build_edges_and_flags(parts)

    # independent reference: np.unique on the same compact key
    sc = ev.syscall
    from nerrf_amd.data.trace import SYSCALL_IDS

    read_like = np.isin(sc, [SYSCALL_IDS["read"], SYSCALL_IDS["exec"]]).astype(np.int64)
    n_files = parts["n_files"]
    valid = (parts["ev_file"] >= 0) & (parts["ev_proc"] >= 0)
    key = (((parts["ev_proc"] - n_files) * n_files + parts["ev_file"]) * 2 + read_like)[valid]
    uk, inv = np.unique(key, return_inverse=True)
    rec = np.exp(-(parts["t1"] - ev.ts[valid]) / 10.0)
    conf_ref = np.bincount(inv, weights=rec)
    assert ed["edge_index"].shape[1] == len(uk)
    w_ref = (1.0 - np.exp(-conf_ref)).astype(np.float32)
    assert np.allclose(np.sort(ed["edge_weight"]), np.sort(w_ref), atol=1e-6)


def test_store_compact_unsorted_stream_filter():
    """Out-of-order events still filter correctly (boolean-mask branch)."""
    from nerrf_amd.graph.store import DeltaGraphStore

    st = DeltaGraphStore(window_s=10.0, delta_s=2.0)
    # deliberately jittered timestamps
    for i, ts in enumerate([100.0, 99.5, 101.0, 104.9, 103.0, 112.0, 111.5]):
        st.append(ts=ts, pid=5, syscall="write", path=f"/f{i}", nbytes=1)
    ev = st.compact(now=112.0)
    kept = sorted(float(t) for t in ev.ts)
    assert kept == [103.0, 104.9, 111.5, 112.0]  # window [102, 112]


def test_incremental_merge_matches_full_rebuild():
    """Per-delta summary merge reproduces build_graph_parts +
    build_edges_and_flags on the same window (ids/order exact, weights fp)."""
    import numpy as np

    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts
    from nerrf_amd.graph.incremental import IncrementalWindowState, merge_window
    from nerrf_amd.serve.engine import StreamingEngine

    arr, _ = generate(SynthConfig(duration_s=20.0, benign_rate_hz=3000.0,
                                  n_benign_files=400, seed=13))
    eng = StreamingEngine(device="cpu")
    eng.ingest_events(arr)
    ev = eng.store.compact(None)
    ref_parts = build_graph_parts(ev)
    ref_ed = build_edges_and_flags(ref_parts)

    with eng.store._lock:
        deltas = list(eng.store._deltas)
    state = IncrementalWindowState()
    sums = state.summaries(deltas)
    parts, ed = merge_window(ev, sums)

    assert parts["n_files"] == ref_parts["n_files"]
    assert parts["n_procs"] == ref_parts["n_procs"]
    assert np.array_equal(parts["ev_file"], ref_parts["ev_file"])
    assert np.array_equal(parts["ev_proc"], ref_parts["ev_proc"])
    assert np.array_equal(parts["touched_roots"], ref_parts["touched_roots"])
    assert np.array_equal(parts["path_root"], ref_parts["path_root"])
    assert np.array_equal(ed["edge_index"], ref_ed["edge_index"])
    assert np.allclose(ed["edge_weight"], ref_ed["edge_weight"], atol=1e-5)
    assert np.allclose(ed["edge_ts"], ref_ed["edge_ts"], atol=1e-6)
    for k in ("in_deg", "out_deg", "peer", "suspicious", "note", "recon", "double_ext"):
        assert np.array_equal(ed[k], ref_ed[k]), k

    # summary cache: second call reuses every entry (no recompute)
    sums2 = state.summaries(deltas)
    assert all(a is b for a, b in zip(sums, sums2))


def test_incremental_rename_chain_across_deltas():
    """A rename chain a->b (delta 1), b->c (delta 2) must union {a,b,c} in
    the merged window exactly like the full rebuild."""
    import numpy as np

    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts
    from nerrf_amd.graph.incremental import IncrementalWindowState, merge_window
    from nerrf_amd.graph.store import DeltaGraphStore

    st = DeltaGraphStore(window_s=100.0, delta_s=2.0)
    st.append(ts=0.1, pid=7, syscall="write", path="/d/a", nbytes=8)
    st.append(ts=0.2, pid=7, syscall="rename", path="/d/a", new_path="/d/b")
    # force a new delta, then continue the chain
    st.append(ts=3.0, pid=7, syscall="write", path="/d/b", nbytes=8)
    st.append(ts=3.1, pid=7, syscall="rename", path="/d/b", new_path="/d/c")
    st.append(ts=6.0, pid=9, syscall="read", path="/d/c", nbytes=4)
    ev, deltas = st.compact_with_deltas(None)
    assert len(deltas) >= 2

    ref_parts = build_graph_parts(ev)
    ref_ed = build_edges_and_flags(ref_parts)
    parts, ed = merge_window(ev, IncrementalWindowState().summaries(deltas))
    # one merged file node for {a, b, c}
    assert parts["n_files"] == ref_parts["n_files"] == 1
    assert np.array_equal(parts["path_root"], ref_parts["path_root"])
    assert np.array_equal(ed["edge_index"], ref_ed["edge_index"])
    assert np.allclose(ed["edge_weight"], ref_ed["edge_weight"], atol=1e-6)


def test_incremental_random_streams_match_full():
    """Property check: random multi-delta streams, incremental == full."""
    import numpy as np

    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts
    from nerrf_amd.graph.incremental import IncrementalWindowState, merge_window
    from nerrf_amd.graph.store import DeltaGraphStore

    rng = np.random.default_rng(99)
    names = [f"/x/f{i}" for i in range(40)]
    calls = ["write", "read", "openat", "unlink", "chmod", "exec"]
    for trial in range(3):
        st = DeltaGraphStore(window_s=1000.0, delta_s=1.0)
        t = 0.0
        for _ in range(800):
            t += float(rng.random() * 0.02)
            if rng.random() < 0.1:
                i, j = rng.integers(0, len(names), 2)
                st.append(ts=t, pid=int(10 + rng.integers(0, 5)), syscall="rename",
                          path=names[i], new_path=names[j])
            else:
                st.append(ts=t, pid=int(10 + rng.integers(0, 5)),
                          syscall=str(rng.choice(calls)), path=str(rng.choice(names)),
                          nbytes=int(rng.integers(0, 4096)))
        ev, deltas = st.compact_with_deltas(None)
        ref_parts = build_graph_parts(ev)
        ref_ed = build_edges_and_flags(ref_parts)
        parts, ed = merge_window(ev, IncrementalWindowState().summaries(deltas))
        assert np.array_equal(parts["path_root"], ref_parts["path_root"]), trial
        assert np.array_equal(parts["ev_file"], ref_parts["ev_file"]), trial
        assert np.array_equal(parts["ev_proc"], ref_parts["ev_proc"]), trial
        assert np.array_equal(ed["edge_index"], ref_ed["edge_index"]), trial
        assert np.allclose(ed["edge_weight"], ref_ed["edge_weight"], atol=1e-6), trial
        assert np.array_equal(ed["suspicious"], ref_ed["suspicious"]), trial


def test_edge_cache_random_streams_match_full():
    """Property check: random streams merged TICK BY TICK with a persistent
    IncrementalWindowState (edge cache live, renames invalidating mid-stream)
    == full rebuild at every tick."""
    import numpy as np

    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts
    from nerrf_amd.graph.incremental import IncrementalWindowState, merge_window
    from nerrf_amd.graph.store import DeltaGraphStore

    rng = np.random.default_rng(123)
    names = [f"/x/f{i}" for i in range(40)]
    calls = ["write", "read", "openat", "unlink", "chmod", "exec"]
    for trial in range(2):
        st = DeltaGraphStore(window_s=1000.0, delta_s=1.0)
        state = IncrementalWindowState()
        t = 0.0
        for tick in range(8):
            for _ in range(120):
                t += float(rng.random() * 0.02)
                if rng.random() < 0.08:
                    i, j = rng.integers(0, len(names), 2)
                    st.append(ts=t, pid=int(10 + rng.integers(0, 5)),
                              syscall="rename", path=names[i], new_path=names[j])
                else:
                    st.append(ts=t, pid=int(10 + rng.integers(0, 5)),
                              syscall=str(rng.choice(calls)),
                              path=str(rng.choice(names)),
                              nbytes=int(rng.integers(0, 4096)))
            ev, deltas = st.compact_with_deltas(None)
            ref_ed = build_edges_and_flags(build_graph_parts(ev))
            _, ed = merge_window(ev, state.summaries(deltas), state=state)
            assert np.array_equal(ed["edge_index"], ref_ed["edge_index"]), (trial, tick)
            assert np.allclose(ed["edge_weight"], ref_ed["edge_weight"], atol=1e-6), (trial, tick)
            assert np.allclose(ed["edge_ts"], ref_ed["edge_ts"], atol=1e-6), (trial, tick)


def test_bulk_append_array_matches_scalar_append():
    """store.append_array == per-event append: same delta boundaries, same
    compacted window, same string interning."""
    import numpy as np

    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.data.trace import SYSCALL_NAMES
    from nerrf_amd.graph.store import DeltaGraphStore

    arr, _ = generate(SynthConfig(duration_s=12.0, benign_rate_hz=600.0,
                                  n_benign_files=60, seed=4))
    a = DeltaGraphStore(window_s=1000.0, delta_s=2.0)
    for i in range(len(arr)):
        a.append(
            ts=float(arr.ts[i]), pid=int(arr.pid[i]),
            syscall=SYSCALL_NAMES.get(int(arr.syscall[i]), "unknown"),
            path=arr.paths.lookup(int(arr.path_id[i])) if arr.path_id[i] >= 0 else "",
            new_path=arr.paths.lookup(int(arr.new_path_id[i])) if arr.new_path_id[i] >= 0 else "",
            nbytes=int(arr.nbytes[i]),
            comm=arr.comms.lookup(int(arr.comm_id[i])) if arr.comm_id[i] >= 0 else "",
        )
    b = DeltaGraphStore(window_s=1000.0, delta_s=2.0)
    b.append_array(arr)

    ea, da = a.compact_with_deltas(None)
    eb, db = b.compact_with_deltas(None)
    assert [len(d) for d in da] == [len(d) for d in db]
    assert np.array_equal(ea.ts, eb.ts)
    assert np.array_equal(ea.syscall, eb.syscall)
    assert np.array_equal(ea.nbytes, eb.nbytes)
    # same strings behind the (possibly differently-ordered) ids
    pa = [ea.paths.lookup(int(i)) if i >= 0 else "" for i in ea.path_id[:500]]
    pb = [eb.paths.lookup(int(i)) if i >= 0 else "" for i in eb.path_id[:500]]
    assert pa == pb
    ca = [ea.comms.lookup(int(i)) if i >= 0 else "" for i in ea.comm_id[:500]]
    cb = [eb.comms.lookup(int(i)) if i >= 0 else "" for i in eb.comm_id[:500]]
    assert ca == cb


def test_append_array_unsorted_and_eviction():
    """Bulk ingest sorts jittered timestamps and evicts out-of-window deltas
    like the scalar path."""
    import numpy as np

    from nerrf_amd.data.trace import EventArrayBuilder, StringTable
    from nerrf_amd.graph.store import DeltaGraphStore

    b = EventArrayBuilder(StringTable(), StringTable())
    ts = [5.0, 1.0, 3.0, 2.0, 4.0, 50.0, 49.0, 51.0]
    for i, t in enumerate(ts):
        b.add(ts=t, pid=3, syscall="write", path=f"/u/f{i}", nbytes=10)
    st = DeltaGraphStore(window_s=10.0, delta_s=2.0)
    st.append_array(b.build(sort=False))
    ev = st.compact(None)
    kept = [float(x) for x in ev.ts]
    # everything before 51-10=41 was evicted at ingest time
    assert kept == [49.0, 50.0, 51.0]
    assert st.evicted_events == 5
    # string identity survived the remap
    assert all(ev.paths.lookup(int(i)).startswith("/u/f") for i in ev.path_id)


@pytest.mark.parametrize("kind", ["supply_chain", "supply_chain_net", "benign_rotate", "benign_backup", "benign_build"])
def test_incremental_merge_all_scenario_kinds(kind):
    """Incremental == full across every synthetic scenario family."""
    import numpy as np

    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts
    from nerrf_amd.graph.incremental import IncrementalWindowState, merge_window
    from nerrf_amd.serve.engine import StreamingEngine

    arr, _ = generate(SynthConfig(kind=kind, duration_s=18.0, benign_rate_hz=800.0,
                                  n_benign_files=120, seed=31))
    eng = StreamingEngine(device="cpu")
    eng.store.window_s = 1e9
    eng.ingest_events(arr)
    ev, deltas = eng.store.compact_with_deltas(None)
    ref_parts = build_graph_parts(ev)
    ref_ed = build_edges_and_flags(ref_parts)
    parts, ed = merge_window(ev, IncrementalWindowState().summaries(deltas))
    assert np.array_equal(parts["ev_file"], ref_parts["ev_file"]), kind
    assert np.array_equal(parts["ev_proc"], ref_parts["ev_proc"]), kind
    assert np.array_equal(ed["edge_index"], ref_ed["edge_index"]), kind
    assert np.allclose(ed["edge_weight"], ref_ed["edge_weight"], atol=1e-5), kind
    for k in ("suspicious", "note", "recon", "double_ext", "in_deg", "out_deg"):
        assert np.array_equal(ed[k], ref_ed[k]), (kind, k)


def test_incremental_merge_radix_guard_wide_window():
    """merge_window on a window whose key space exceeds 1<<24 must take the
    guarded radix path (no kspace-dense allocations on the
    production tick) and still match the full rebuild."""
    import numpy as np

    from nerrf_amd.data.trace import EventArrayBuilder, StringTable
    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts
    from nerrf_amd.graph.incremental import IncrementalWindowState, merge_window

    rng = np.random.default_rng(17)
    paths_tab, comm_tab = StringTable(), StringTable()
    deltas = []
    t = 0.0
    for _ in range(4):  # 4 sealed deltas sharing the global string tables
        b = EventArrayBuilder(paths_tab, comm_tab)
        for _ in range(6000):
            p = int(rng.integers(0, 3000))
            f = int(rng.integers(0, 3000))
            b.add(ts=t, pid=1000 + p,
                  syscall="write" if rng.random() < 0.7 else "read",
                  path=f"/data/f{f}", nbytes=64)
            t += 1e-4
        deltas.append(b.build())

    # window = concatenation of the deltas
    from nerrf_amd.data.trace import concat

    ev = concat(deltas)
    parts_ref = build_graph_parts(ev)
    assert 2 * parts_ref["n_procs"] * parts_ref["n_files"] >= (1 << 24)
    ed_ref = build_edges_and_flags(parts_ref)

    parts, ed = merge_window(ev, IncrementalWindowState().summaries(deltas))
    assert parts["n_files"] == parts_ref["n_files"]
    assert parts["n_procs"] == parts_ref["n_procs"]
    assert np.array_equal(ed["edge_index"], ed_ref["edge_index"])
    assert np.allclose(ed["edge_weight"], ed_ref["edge_weight"], atol=1e-5)
    assert np.allclose(ed["edge_ts"], ed_ref["edge_ts"], atol=1e-6)


def test_bulk_append_small_batches_coalesce_deltas():
    """Many small append_array calls must not fragment the delta ring: the
    trailing delta keeps absorbing events until its delta_s span closes
    (same boundary rule as scalar append)."""
    import numpy as np

    from nerrf_amd.data.trace import EventArrayBuilder
    from nerrf_amd.graph.store import DeltaGraphStore

    st = DeltaGraphStore(window_s=1000.0, delta_s=5.0)
    t = 0.0
    for batch in range(60):  # 60 batches of 10 events over 30 s
        b = EventArrayBuilder(st.paths, st.comms)
        for _ in range(10):
            b.add(ts=t, pid=1, syscall="write", path=f"/f{int(t) % 5}", nbytes=1)
            t += 0.05
        st.append_array(b.build())
    # 30 s of events at delta_s=5 -> ~6 deltas, not 60
    assert len(st._deltas) <= 8
    arr = st.compact()
    assert len(arr) == 600
    assert bool(np.all(np.diff(arr.ts) >= 0))
    # boundary rule: every delta spans < delta_s from its first event
    for d in st._deltas:
        assert float(d.ts[-1]) - float(d.ts[0]) < 5.0


# ---------------------------------------------------------------------------
# process-identity channel (x[:, 27], NERRF_PROC_IDENTITY gate)
# ---------------------------------------------------------------------------

def _comm_events():
    """pid 10: all-allowlisted comm; pid 11: execs into an off-list payload;
    pid 12: no comm recorded."""
    b = EventArrayBuilder()
    b.add(ts=1.0, pid=10, syscall="write", path="/d/a.dat", nbytes=10, comm="nginx")
    b.add(ts=2.0, pid=10, syscall="read", path="/d/a.dat", nbytes=5, comm="nginx")
    b.add(ts=3.0, pid=11, syscall="write", path="/d/b.dat", nbytes=10, comm="python3")
    b.add(ts=4.0, pid=11, syscall="write", path="/d/b.dat", nbytes=10, comm="evilbin")
    b.add(ts=5.0, pid=12, syscall="read", path="/d/c.dat", nbytes=1)
    return b.build()


def test_trusted_proc_flags_semantics():
    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts

    arr = _comm_events()
    parts = build_graph_parts(arr)
    ed = build_edges_and_flags(parts)
    n_files = parts["n_files"]
    trusted = ed["trusted_proc"]
    assert trusted.shape == (parts["n_nodes"],)
    assert trusted[:n_files].sum() == 0  # file nodes never trusted
    # upids sorted -> proc nodes are pids 10, 11, 12 in order
    assert trusted[n_files + 0] == 1.0  # every event allowlisted
    assert trusted[n_files + 1] == 0.0  # one off-list comm clears the flag
    assert trusted[n_files + 2] == 0.0  # unknown comm is not trusted


def test_proc_identity_channel_default_off():
    from nerrf_amd.graph.constructor import build_graph

    g = build_graph(_comm_events())
    assert np.all(g.x[:, 27] == 0.0)


def test_proc_identity_channel_enabled(monkeypatch):
    from nerrf_amd.graph.constructor import build_graph, build_graph_parts

    monkeypatch.setenv("NERRF_PROC_IDENTITY", "1")
    arr = _comm_events()
    g = build_graph(arr)
    n_files = build_graph_parts(arr)["n_files"]
    assert g.x[:, 27].sum() == 1.0
    assert g.x[n_files + 0, 27] == 1.0


def test_proc_identity_incremental_matches_full(monkeypatch):
    """merge_window's trusted_proc == build_edges_and_flags' on the same
    window, including comms arriving across delta boundaries."""
    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts
    from nerrf_amd.graph.incremental import IncrementalWindowState, merge_window
    from nerrf_amd.graph.store import DeltaGraphStore

    monkeypatch.setenv("NERRF_PROC_IDENTITY", "1")
    st = DeltaGraphStore(window_s=100.0, delta_s=2.0)
    st.append(ts=0.1, pid=7, syscall="write", path="/d/a", nbytes=8, comm="postgres")
    st.append(ts=3.0, pid=7, syscall="write", path="/d/a", nbytes=8, comm="lockbit")
    st.append(ts=3.1, pid=8, syscall="read", path="/d/a", nbytes=4, comm="sshd")
    st.append(ts=6.0, pid=9, syscall="read", path="/d/a", nbytes=4)
    ev, deltas = st.compact_with_deltas(None)
    ref_ed = build_edges_and_flags(build_graph_parts(ev))
    _, ed = merge_window(ev, IncrementalWindowState().summaries(deltas))
    assert np.array_equal(ed["trusted_proc"], ref_ed["trusted_proc"])
    # pid 7 crossed deltas with one off-list comm -> 0; pid 8 trusted
    assert ed["trusted_proc"].sum() == 1.0


def test_edge_cache_parity_across_ticks():
    """Stable-prefix edge cache: repeated merges with a growing open delta
    (cache hits), new renames (sig invalidation), and a seal (token
    invalidation) all reproduce the full rebuild exactly."""
    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts
    from nerrf_amd.graph.incremental import IncrementalWindowState, merge_window
    from nerrf_amd.graph.store import DeltaGraphStore

    arr, _ = generate(SynthConfig(duration_s=12.0, benign_rate_hz=800.0,
                                  n_benign_files=300, seed=31))
    st = DeltaGraphStore(window_s=1e9, delta_s=2.0)
    state = IncrementalWindowState()
    # replay in 9 tick-sized chunks; each tick merges and checks parity
    bounds = np.linspace(0, len(arr), 10).astype(int)
    hits = 0
    for i in range(9):
        st.append_array(arr.slice(bounds[i], bounds[i + 1]))
        if i == 4:  # new rename mid-stream -> sig invalidation path
            t = float(arr.ts[bounds[i + 1] - 1]) + 0.001
            st.append(ts=t, pid=999, syscall="rename", path="/x/q.dat",
                      new_path="/x/q.dat.enc")
        ev, deltas = st.compact_with_deltas(None)
        sums = state.summaries(deltas)
        before = state._edge_cache
        parts, ed = merge_window(ev, sums, state=state)
        if before is not None and state._edge_cache is before:
            hits += 1
        ref_ed = build_edges_and_flags(build_graph_parts(ev))
        assert np.array_equal(ed["edge_index"], ref_ed["edge_index"]), f"tick {i}"
        assert np.allclose(ed["edge_weight"], ref_ed["edge_weight"], atol=1e-5), f"tick {i}"
        assert np.allclose(ed["edge_ts"], ref_ed["edge_ts"], atol=1e-6), f"tick {i}"
        for k in ("in_deg", "out_deg", "peer"):
            assert np.array_equal(ed[k], ref_ed[k]), f"tick {i} {k}"
    # same-delta-set re-merge must be a pure cache hit with identical output
    ev, deltas = st.compact_with_deltas(None)
    sums = state.summaries(deltas)
    a = merge_window(ev, sums, state=state)[1]
    cache_obj = state._edge_cache
    b = merge_window(ev, sums, state=state)[1]
    assert state._edge_cache is cache_obj  # hit, not rebuilt
    assert np.array_equal(a["edge_index"], b["edge_index"])
    assert np.array_equal(a["edge_weight"], b["edge_weight"])


def test_edge_cache_survives_window_expiry():
    """Delta expiry (window trim drops sealed deltas from the front)
    invalidates the stable-prefix cache token; every post-expiry merge
    still equals the full rebuild."""
    from nerrf_amd.graph.constructor import build_edges_and_flags, build_graph_parts
    from nerrf_amd.graph.incremental import IncrementalWindowState, merge_window
    from nerrf_amd.graph.store import DeltaGraphStore

    st = DeltaGraphStore(window_s=6.0, delta_s=1.0)
    state = IncrementalWindowState()
    rng = np.random.default_rng(17)
    t = 0.0
    n_checked = 0
    for tick in range(12):
        for _ in range(60):
            t += float(rng.random() * 0.05)
            st.append(ts=t, pid=int(20 + rng.integers(0, 3)),
                      syscall=["write", "read", "rename"][int(rng.integers(0, 3))],
                      path=f"/w/f{int(rng.integers(0, 25))}",
                      new_path=f"/w/g{int(rng.integers(0, 25))}"
                      if rng.random() < 0.15 else "",
                      nbytes=int(rng.integers(0, 2048)))
        ev, deltas = st.compact_with_deltas(None)
        if not len(ev):
            continue
        ref_ed = build_edges_and_flags(build_graph_parts(ev))
        _, ed = merge_window(ev, state.summaries(deltas), state=state)
        assert np.array_equal(ed["edge_index"], ref_ed["edge_index"]), tick
        assert np.allclose(ed["edge_weight"], ref_ed["edge_weight"], atol=1e-6), tick
        n_checked += 1
    assert n_checked >= 10


def test_aggregate_sparse_keys_branches_agree():
    """Property: the bincount branch and the radix branch of
    aggregate_sparse_keys produce identical (keys, sums, maxes) — checked
    by forcing both on the same data via the key_space guard."""
    from nerrf_amd.graph.constructor import aggregate_sparse_keys

    rng = np.random.default_rng(77)
    for trial in range(5):
        n = int(rng.integers(1, 5000))
        space = int(rng.integers(10, 1 << 20))
        keys = rng.integers(0, space, size=n).astype(np.int64)
        w = rng.random(n)
        ts = rng.random(n) * 100
        # small key_space -> bincount branch
        uk_a, s_a, m_a = aggregate_sparse_keys(keys, w, ts, space)
        # huge declared key_space -> radix branch (same keys)
        uk_b, s_b, m_b = aggregate_sparse_keys(keys, w, ts, 1 << 40)
        assert np.array_equal(uk_a, uk_b), trial
        assert np.allclose(s_a, s_b, atol=1e-9), trial
        assert np.array_equal(m_a, m_b), trial
        # against np.unique reference
        uk_r, inv = np.unique(keys, return_inverse=True)
        assert np.array_equal(uk_a, uk_r)
        assert np.allclose(s_a, np.bincount(inv, weights=w), atol=1e-9)


def test_sample_fanout_torch_validity_cpu():
    """Property: the device-side sampler (runs on CPU tensors too) draws
    only true in-neighbors, self-fills isolated nodes, and is
    deterministic per seed — same contract as the numpy sampler."""
    import torch

    from nerrf_amd.graph.sampling import sample_fanout_torch, to_csr

    arr, _ = generate(SynthConfig(seed=8, duration_s=40, benign_rate_hz=150))
    g = build_graph(arr)
    ei = torch.from_numpy(g.edge_index)
    ew = torch.from_numpy(g.edge_weight)
    idx, w = sample_fanout_torch(ei, ew, g.num_nodes, 16, seed=3)
    idx2, _ = sample_fanout_torch(ei, ew, g.num_nodes, 16, seed=3)
    assert torch.equal(idx, idx2)  # deterministic per seed
    csr = to_csr(g.edge_index, g.num_nodes, g.edge_weight)
    deg = np.diff(csr.indptr)
    idx_np = idx.numpy()
    for n in range(0, g.num_nodes, max(1, g.num_nodes // 23)):
        if deg[n] == 0:
            assert (idx_np[n] == n).all()
        else:
            nbrs = set(csr.indices[csr.indptr[n]:csr.indptr[n + 1]].tolist())
            assert set(idx_np[n].tolist()) <= nbrs
    assert (w >= 0).all()
