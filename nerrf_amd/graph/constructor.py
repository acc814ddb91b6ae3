"""Temporal dependency-graph construction from syscall event windows.

Behavioral spec (reference docs architecture.mdx:36-43, threat-model.mdx:179-188):
  * sliding window of 30-60 s over the event stream,
  * nodes = processes, files and socket destinations (files deduped across
    renames — the upstream spec dedups by inode; traces without inodes get
    rename-union instead, which is the same equivalence for the attack
    pattern; sockets enter the path domain as "tcp://host:port"),
  * edge weight = causality confidence (recency-decayed interaction count),
  * node features: in/out degree, temporal deltas, byte-count ratios,
    extension-pattern confidence, read/write/rename counters, plus the
    documented detection indicators (write-to-rename ratio, ransom-note name,
    recon burst) as explicit feature channels.

Everything is vectorised numpy; `TemporalGraph.to_torch()` produces the
contiguous tensors the GPU path stages into HBM.
"""
from __future__ import annotations

import re
from dataclasses import dataclass
from typing import Dict, Optional, Tuple

import numpy as np

from ..data.synth import AttackWindow
from ..data.trace import SYSCALL_IDS, EventArray

NUM_NODE_FEATURES = 32
# syscalls that imply process->file data flow vs file->process
_WRITE_LIKE = (SYSCALL_IDS["write"], SYSCALL_IDS["openat"], SYSCALL_IDS["chmod"])
_READ_LIKE = (SYSCALL_IDS["read"], SYSCALL_IDS["exec"])

_SUSPICIOUS_EXT = re.compile(r"\.(lockbit\w*|encrypted|locked|crypt\w*)$", re.IGNORECASE)
_RANSOM_NOTE = re.compile(r"(^|/)(README|HOW_TO|RESTORE)[-_]", re.IGNORECASE)
_RECON_BIN = re.compile(r"^/(usr/)?s?bin/")
_DOUBLE_EXT = re.compile(r"\.\w{1,5}\.\w{1,9}$")

# socket-destination policy: egress to a destination NOT matching the
# allowlist marks the socket node suspicious (the "destination allowlist"
# feature channel — threat-model.md hard-negative discussion).  Deployment
# overrides this module constant from config.
_SOCKET_PREFIXES = ("tcp://", "udp://")
DEST_ALLOWLIST = ("tcp://10.", "tcp://192.168.", "tcp://backup.", "udp://10.")

# process-identity channel (x[:, 27], reserved in round 1): comms a deployment
# considers "known-good daemons".  A process node is trusted only when EVERY
# event it emitted in the window carries an allowlisted comm — a single
# unknown or off-list comm (execve into a payload, masquerading child)
# clears the flag for the whole window.  Deployment overrides this module
# constant from config, like DEST_ALLOWLIST above.
TRUSTED_COMMS = frozenset({
    # system daemons
    "systemd", "systemd-journal", "journald", "dbus-daemon", "sshd",
    "cron", "crond", "rsyslogd", "agetty", "chronyd",
    # container / orchestration
    "containerd", "dockerd", "kubelet", "containerd-shim",
    # common benign workload comms (matches data/synth.py's benign mix)
    "nginx", "postgres", "node", "python3", "redis-server",
})


def proc_identity_enabled() -> bool:
    """Gate for the x[:, 27] trusted-process channel.

    Default OFF: the vendored checkpoint and the calibrated alarm were
    trained/swept with the channel all-zero, and its input weight column is
    untrained (zero input -> zero gradient), so enabling it without a
    retrain injects random-init noise into benign process nodes.  Flip
    NERRF_PROC_IDENTITY=1 together with a retrain (tools/train_mixed.py)
    and a re-calibration (tools/calibrate_alarm.py).
    """
    import os

    return os.environ.get("NERRF_PROC_IDENTITY", "0") == "1"


def trusted_proc_flags(
    events: EventArray,
    ev_proc: np.ndarray,
    n_files: int,
    n_procs: int,
    n_nodes: int,
) -> np.ndarray:
    """[n_nodes] float32: 1.0 on process nodes whose every window event has
    an allowlisted comm (TRUSTED_COMMS); 0.0 on file nodes, on processes
    with any off-list/unknown comm, and on processes with no comm'd events.
    Host string-domain work, like the path flags."""
    out = np.zeros(n_nodes, dtype=np.float32)
    if n_procs == 0:
        return out
    n_comms = len(events.comms)
    allowed = np.zeros(n_comms + 1, dtype=bool)
    for c in TRUSTED_COMMS:
        i = events.comms.get(c)
        if i is not None:
            allowed[i] = True
    sel = ev_proc >= 0
    if not sel.any():
        return out
    pidx = (ev_proc[sel] - n_files).astype(np.int64)
    cid = events.comm_id[sel]
    ok = (cid >= 0) & allowed[np.clip(cid, 0, n_comms)]
    n_ok = np.bincount(pidx, weights=ok.astype(np.float64), minlength=n_procs)
    n_all = np.bincount(pidx, minlength=n_procs)
    out[n_files:] = ((n_ok == n_all) & (n_all > 0)).astype(np.float32)
    return out


def rename_path_roots(ra: np.ndarray, rb: np.ndarray, n_paths: int) -> np.ndarray:
    """path_root from rename pairs: connected components restricted to the
    paths that actually appear in a rename (the full-path-space CC walked
    every interned string — ~12 ms/tick at 600k paths for a few hundred
    rename endpoints).  Root = minimum original path id of the component."""
    if ra.size == 0:
        return np.arange(n_paths, dtype=np.int64)
    from scipy.sparse import coo_matrix
    from scipy.sparse.csgraph import connected_components

    nodes = np.unique(np.concatenate([ra, rb]))
    m = len(nodes)
    la = np.searchsorted(nodes, ra)
    lb = np.searchsorted(nodes, rb)
    g = coo_matrix((np.ones(ra.size, dtype=np.int8), (la, lb)), shape=(m, m))
    n_comp, label = connected_components(g, directed=False)
    min_id = np.full(n_comp, np.iinfo(np.int64).max, dtype=np.int64)
    np.minimum.at(min_id, label, nodes)
    path_root = np.arange(n_paths, dtype=np.int64)
    path_root[nodes] = min_id[label]
    return path_root


def aggregate_sparse_keys(
    key: np.ndarray, wsum: np.ndarray, ts_v: np.ndarray, key_space: int
) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Group-by over integer keys -> (unique_keys, sum(wsum), max(ts_v)).

    Dense bincount when the key space is small (three O(n) passes); LSD
    radix sort otherwise so a window with many procs*files never allocates
    key_space-sized arrays (multi-GB at ~2k procs x ~4M paths).
    """
    if not key.size:
        e = np.empty(0, np.int64)
        return e, np.empty(0, np.float64), np.empty(0, np.float64)
    if key_space < (1 << 24):
        cnt = np.bincount(key, minlength=key_space)
        uk = np.nonzero(cnt)[0]
        agg_w = np.bincount(key, weights=wsum, minlength=key_space)[uk]
        last_all = np.full(key_space, -np.inf)
        np.maximum.at(last_all, key, ts_v)
        return uk, agg_w, last_all[uk]
    # LSD radix by 16-bit digits: numpy's stable argsort is radix only for
    # <= 16-bit dtypes
    order = np.arange(len(key), dtype=np.int64)
    shift = 0
    kmax = int(key.max())
    while kmax >> shift:
        digit = ((key >> shift) & 0xFFFF).astype(np.uint16)
        order = order[np.argsort(digit[order], kind="stable")]
        shift += 16
    ks = key[order]
    new_grp = np.empty(len(ks), dtype=bool)
    new_grp[0] = True
    np.not_equal(ks[1:], ks[:-1], out=new_grp[1:])
    starts = np.nonzero(new_grp)[0]
    uk = ks[starts]
    agg_w = np.add.reduceat(wsum[order], starts)
    agg_t = np.maximum.reduceat(ts_v[order], starts)
    return uk, agg_w, agg_t


class _UnionFind:
    def __init__(self, n: int) -> None:
        self.parent = np.arange(n, dtype=np.int64)

    def find(self, x: int) -> int:
        root = x
        while self.parent[root] != root:
            root = self.parent[root]
        while self.parent[x] != root:
            self.parent[x], x = root, self.parent[x]
        return root

    def union(self, a: int, b: int) -> None:
        ra, rb = self.find(a), self.find(b)
        if ra != rb:
            self.parent[max(ra, rb)] = min(ra, rb)


@dataclass
class TemporalGraph:
    """One 30-60 s window's dependency graph, columnar."""

    x: np.ndarray  # [N, F] float32 node features
    edge_index: np.ndarray  # [2, E] int64 (src, dst) in node ids
    edge_weight: np.ndarray  # [E] float32 causality confidence
    edge_ts: np.ndarray  # [E] float32 last-interaction time (window-relative)
    node_kind: np.ndarray  # [N] int8: 0 = process, 1 = file, 2 = socket
    node_key: np.ndarray  # [N] int64: pid for processes, path root id for files
    y_node: Optional[np.ndarray] = None  # [N] float32 anomaly ground truth
    y_edge: Optional[np.ndarray] = None  # [E] float32
    t0: float = 0.0
    t1: float = 0.0

    @property
    def num_nodes(self) -> int:
        return int(self.x.shape[0])

    @property
    def num_edges(self) -> int:
        return int(self.edge_index.shape[1])

    def to_torch(self, device=None, dtype=None):
        import torch

        dev = device if device is not None else "cpu"
        x = torch.from_numpy(self.x)
        if dtype is not None:
            x = x.to(dtype)
        return {
            "x": x.to(dev),
            "edge_index": torch.from_numpy(self.edge_index).to(dev),
            "edge_weight": torch.from_numpy(self.edge_weight).to(dev),
            "edge_ts": torch.from_numpy(self.edge_ts).to(dev),
            "node_kind": torch.from_numpy(self.node_kind).to(dev),
            "y_node": None if self.y_node is None else torch.from_numpy(self.y_node).to(dev),
            "y_edge": None if self.y_edge is None else torch.from_numpy(self.y_edge).to(dev),
        }


def _string_flag_bits(paths) -> np.ndarray:
    """Per-path regex indicator bits, cached on the StringTable.

    Tables only grow (the streaming store reuses one table across windows),
    so only strings added since the last call are regex-scanned."""
    cached = getattr(paths, "_nerrf_flag_cache", None)
    n = len(paths)
    if cached is not None and len(cached) >= n:
        return cached[:n]
    bits = np.zeros(n, dtype=np.uint8)
    start = 0
    if cached is not None:
        bits[: len(cached)] = cached
        start = len(cached)
    for path_idx in range(start, n):
        s = paths.strings[path_idx]
        b = 0
        if _SUSPICIOUS_EXT.search(s):
            b |= 1
        if _RANSOM_NOTE.search(s):
            b |= 2
        if _RECON_BIN.search(s):
            b |= 4
        if _DOUBLE_EXT.search(s):
            b |= 8
        if s.startswith(_SOCKET_PREFIXES):
            b |= 16  # socket node
            if not s.startswith(DEST_ALLOWLIST):
                b |= 32  # egress to an unlisted destination
        bits[path_idx] = b
    paths._nerrf_flag_cache = bits
    return bits


def _path_flags(
    paths, path_root: np.ndarray, root_to_file: np.ndarray, n_files: int
) -> Tuple[np.ndarray, ...]:
    """Per file-node string-pattern indicator features.

    Every alias of a file (pre- and post-rename paths) contributes its flags
    to the rename-union node, so a `.lockbit3` rename marks the merged node.
    """
    bits = _string_flag_bits(paths)
    node_of_path = root_to_file[path_root]  # [n_paths] file node or -1
    valid = node_of_path >= 0
    nodes = node_of_path[valid]
    b = bits[: len(node_of_path)][valid]
    suspicious = np.zeros(n_files, dtype=np.float32)
    note = np.zeros(n_files, dtype=np.float32)
    recon = np.zeros(n_files, dtype=np.float32)
    double_ext = np.zeros(n_files, dtype=np.float32)
    # unlisted socket destinations share the suspicious channel (no GPU
    # kernel signature change; the flag is host-computed either way)
    np.maximum.at(suspicious, nodes, (((b & 1) | ((b >> 5) & 1)) != 0).astype(np.float32))
    np.maximum.at(note, nodes, ((b >> 1) & 1).astype(np.float32))
    np.maximum.at(recon, nodes, ((b >> 2) & 1).astype(np.float32))
    np.maximum.at(double_ext, nodes, ((b >> 3) & 1).astype(np.float32))
    return suspicious, note, recon, double_ext


def file_node_kinds(paths, path_root: np.ndarray, touched_roots: np.ndarray) -> np.ndarray:
    """Kind per path-domain node: 1 = file, 2 = socket destination.

    Spec: nodes = processes/files/sockets (reference architecture.mdx:36-43).
    Socket destinations live in the path string domain ("tcp://host:port"),
    so they ride the same interning/rename/feature machinery; the kind
    one-hot (is_process, is_file) naturally encodes sockets as (0, 0) in
    both the CPU and GPU feature paths.
    """
    bits = _string_flag_bits(paths)
    is_sock = (bits[touched_roots] & 16) != 0 if len(touched_roots) else np.zeros(0, bool)
    return np.where(is_sock, np.int8(2), np.int8(1)).astype(np.int8)


def build_graph_parts(
    events: EventArray,
    causality_tau_s: float = 10.0,
) -> dict:
    """Identity + edge domain of graph construction (string/set work, host).

    Returns the intermediates shared by the CPU feature path (build_graph)
    and the GPU delta-compaction path (graph.gpu_store): per-event node ids,
    node tables, aggregated edges, degrees, and path-pattern flags.
    """
    n_ev = len(events)
    t0 = float(events.ts[0]) if n_ev else 0.0
    t1 = float(events.ts[-1]) if n_ev else 0.0
    span = max(t1 - t0, 1e-6)

    n_paths = len(events.paths)

    # ---- file identity: union path <-> new_path over renames --------------
    # connected components over the rename graph, canonical root = min path
    # id per component (the exact semantics of the serial union-find in
    # _UnionFind, which parents max->min; scipy's C traversal replaces a
    # per-rename Python loop that cost ~40 ms on rename-heavy windows)
    ren_mask = (events.syscall == SYSCALL_IDS["rename"]) & (events.new_path_id >= 0)
    ra = events.path_id[ren_mask]
    rb = events.new_path_id[ren_mask]
    m = (ra >= 0) & (rb >= 0)
    ra, rb = ra[m], rb[m]
    path_root = rename_path_roots(ra, rb, n_paths)

    # file nodes = distinct roots actually touched (presence mask instead of
    # a concatenate + sort-based unique over ~2x the event count)
    pi, npi = events.path_id, events.new_path_id
    seen = np.zeros(n_paths, dtype=bool)
    if n_ev:
        seen[path_root[pi[pi >= 0]]] = True
        seen[path_root[npi[npi >= 0]]] = True
    touched_roots = np.nonzero(seen)[0].astype(np.int64)
    n_files = len(touched_roots)

    # process nodes; dense pid LUT when the pid range is small (the usual
    # case: kernel pid_max), sorted-unique + searchsorted otherwise
    if n_ev:
        pmax = int(events.pid.max())
        if 0 <= events.pid.min() and pmax < (1 << 22):
            lut = np.full(pmax + 1, -1, dtype=np.int64)
            lut[events.pid] = 0
            upids = np.nonzero(lut >= 0)[0].astype(np.int64)
            lut[upids] = np.arange(len(upids), dtype=np.int64)
        else:
            upids = np.unique(events.pid)
            lut = None
    else:
        upids = np.empty(0, dtype=np.int64)
        lut = None
    n_procs = len(upids)
    n_nodes = n_files + n_procs

    # per-event node ids (vectorised via lookup tables)
    root_to_file = np.full(n_paths, -1, dtype=np.int64)
    if n_files:
        root_to_file[touched_roots] = np.arange(n_files, dtype=np.int64)
    ev_file = np.where(pi >= 0, root_to_file[path_root[np.clip(pi, 0, None)]], -1)
    if not n_ev:
        ev_proc = np.empty(0, dtype=np.int64)
    elif lut is not None:
        ev_proc = n_files + lut[events.pid]
    else:
        ev_proc = n_files + np.searchsorted(upids, events.pid)

    # degrees and peers are edge-domain: computed below with edges
    return {
        "events": events,
        "n_ev": n_ev,
        "t0": t0,
        "t1": t1,
        "span": span,
        "ev_file": ev_file,
        "ev_proc": ev_proc,
        "n_files": n_files,
        "n_procs": n_procs,
        "n_nodes": n_nodes,
        "path_root": path_root,
        "root_to_file": root_to_file,
        "touched_roots": touched_roots,
        "upids": upids,
    }


def build_edges_and_flags(parts: dict, causality_tau_s: float = 10.0) -> dict:
    """Edge aggregation, degrees/peers and path-pattern flags from parts."""
    events: EventArray = parts["events"]
    ev_file, ev_proc = parts["ev_file"], parts["ev_proc"]
    n_nodes, n_files, n_procs = parts["n_nodes"], parts["n_files"], parts["n_procs"]
    t0, t1, span = parts["t0"], parts["t1"], parts["span"]
    sc = events.syscall

    valid = (ev_file >= 0) & (ev_proc >= 0)
    read_like = np.zeros(int(sc.max()) + 1 if len(sc) else 1, dtype=bool)
    for s in _READ_LIKE:
        if s < len(read_like):
            read_like[s] = True
    direction = read_like[sc].astype(np.int64)  # 0: proc->file, 1: file->proc
    # compact (proc, file, direction) key: procs are contiguous above
    # n_files, so the key space is 2 * n_procs * n_files
    key = ((ev_proc - n_files).astype(np.int64) * n_files + ev_file.astype(np.int64)) * 2 + direction
    key = key[valid]
    key_space = 2 * n_procs * n_files
    if key.size:
        ts_v = events.ts[valid]
        # causality confidence: recency-decayed count, saturating
        rec = np.exp(-(t1 - ts_v) / causality_tau_s)
        # sort-free dedup + aggregation when the key space is small (three
        # O(n) bincount-class passes; np.unique's int64 sort cost ~40 ms per
        # 600k-event window), radix fallback otherwise
        uk, e_conf, e_last = aggregate_sparse_keys(key, rec, ts_v, key_space)
        dirs = uk % 2
        pf = uk // 2
        e_proc = n_files + (pf // n_files).astype(np.int64)
        e_file = (pf % n_files).astype(np.int64)
        src = np.where(dirs == 0, e_proc, e_file)
        dst = np.where(dirs == 0, e_file, e_proc)
        edge_index = np.stack([src, dst]).astype(np.int64)
        edge_weight = (1.0 - np.exp(-e_conf)).astype(np.float32)  # saturate to (0,1)
        edge_ts = ((e_last - t0) / span).astype(np.float32)
    else:
        edge_index = np.zeros((2, 0), dtype=np.int64)
        edge_weight = np.zeros(0, dtype=np.float32)
        edge_ts = np.zeros(0, dtype=np.float32)

    peer = np.zeros(n_nodes, dtype=np.float64)
    in_deg = np.zeros(n_nodes, dtype=np.float64)
    out_deg = np.zeros(n_nodes, dtype=np.float64)
    if edge_index.shape[1]:
        np.add.at(peer, edge_index[0], 1.0)
        np.add.at(peer, edge_index[1], 1.0)
        np.add.at(out_deg, edge_index[0], 1.0)
        np.add.at(in_deg, edge_index[1], 1.0)

    suspicious, note, recon, double_ext = _path_flags(
        events.paths, parts["path_root"], parts["root_to_file"], n_files
    )
    pad = np.zeros(n_procs, dtype=np.float32)
    return {
        "edge_index": edge_index,
        "edge_weight": edge_weight,
        "edge_ts": edge_ts,
        "in_deg": in_deg,
        "out_deg": out_deg,
        "peer": peer,
        "suspicious": np.concatenate([suspicious, pad]),
        "note": np.concatenate([note, pad]),
        "recon": np.concatenate([recon, pad]),
        "double_ext": np.concatenate([double_ext, pad]),
        "trusted_proc": trusted_proc_flags(events, ev_proc, n_files, n_procs, n_nodes),
    }


def build_graph(
    events: EventArray,
    window: Optional[AttackWindow] = None,
    y_event: Optional[np.ndarray] = None,
    causality_tau_s: float = 10.0,
    parts: Optional[dict] = None,
) -> TemporalGraph:
    """Build the dependency graph for one event window."""
    if parts is None:
        parts = build_graph_parts(events, causality_tau_s)
    n_ev = parts["n_ev"]
    t0, t1, span = parts["t0"], parts["t1"], parts["span"]
    ev_file, ev_proc = parts["ev_file"], parts["ev_proc"]
    n_files, n_procs, n_nodes = parts["n_files"], parts["n_procs"], parts["n_nodes"]
    path_root, root_to_file = parts["path_root"], parts["root_to_file"]
    touched_roots, upids = parts["touched_roots"], parts["upids"]

    # ---- per-node counters (bincount: ~5x np.add.at) ----------------------
    def _count(mask: np.ndarray, ids: np.ndarray) -> np.ndarray:
        sel = mask & (ids >= 0)
        if not sel.any():
            return np.zeros(n_nodes, dtype=np.float64)
        return np.bincount(ids[sel], minlength=n_nodes).astype(np.float64)

    def _sum(mask: np.ndarray, ids: np.ndarray, vals: np.ndarray) -> np.ndarray:
        sel = mask & (ids >= 0)
        if not sel.any():
            return np.zeros(n_nodes, dtype=np.float64)
        return np.bincount(ids[sel], weights=vals[sel].astype(np.float64), minlength=n_nodes)

    sc = events.syscall
    is_read = sc == SYSCALL_IDS["read"]
    is_write = sc == SYSCALL_IDS["write"]
    is_rename = sc == SYSCALL_IDS["rename"]
    is_unlink = sc == SYSCALL_IDS["unlink"]
    is_open = sc == SYSCALL_IDS["openat"]
    is_exec = sc == SYSCALL_IDS["exec"]

    # counts touch both the file node and the process node of each event
    cnt_read = _count(is_read, ev_file) + _count(is_read, ev_proc)
    cnt_write = _count(is_write, ev_file) + _count(is_write, ev_proc)
    cnt_rename = _count(is_rename, ev_file) + _count(is_rename, ev_proc)
    cnt_unlink = _count(is_unlink, ev_file) + _count(is_unlink, ev_proc)
    cnt_open = _count(is_open, ev_file) + _count(is_open, ev_proc)
    cnt_exec = _count(is_exec, ev_file) + _count(is_exec, ev_proc)
    cnt_total = _count(np.ones(n_ev, dtype=bool), ev_file) + _count(np.ones(n_ev, dtype=bool), ev_proc)

    bytes_read = _sum(is_read, ev_file, events.nbytes) + _sum(is_read, ev_proc, events.nbytes)
    bytes_write = _sum(is_write, ev_file, events.nbytes) + _sum(is_write, ev_proc, events.nbytes)

    # first/last event time per node
    t_first = np.full(n_nodes, np.inf)
    t_last = np.full(n_nodes, -np.inf)
    for ids in (ev_file, ev_proc):
        sel = ids >= 0
        if sel.any():
            np.minimum.at(t_first, ids[sel], events.ts[sel])
            np.maximum.at(t_last, ids[sel], events.ts[sel])
    t_first[~np.isfinite(t_first)] = t0
    t_last[~np.isfinite(t_last)] = t0

    # ---- edges + degrees + string flags (shared with the GPU path) --------
    ed = build_edges_and_flags(parts, causality_tau_s)
    edge_index, edge_weight, edge_ts = ed["edge_index"], ed["edge_weight"], ed["edge_ts"]
    in_deg, out_deg, peer = ed["in_deg"], ed["out_deg"], ed["peer"]
    suspicious, note, recon, double_ext = ed["suspicious"], ed["note"], ed["recon"], ed["double_ext"]

    # ---- assemble feature matrix ------------------------------------------
    x = np.zeros((n_nodes, NUM_NODE_FEATURES), dtype=np.float32)
    node_kind = np.concatenate(
        [file_node_kinds(events.paths, path_root, touched_roots),
         np.zeros(n_procs, dtype=np.int8)]
    )
    dur = np.maximum(t_last - t_first, 0.0)
    x[:, 0] = node_kind == 0  # is_process
    x[:, 1] = node_kind == 1  # is_file
    x[:, 2] = np.log1p(in_deg)
    x[:, 3] = np.log1p(out_deg)
    x[:, 4] = np.log1p(cnt_read)
    x[:, 5] = np.log1p(cnt_write)
    x[:, 6] = np.log1p(cnt_rename)
    x[:, 7] = np.log1p(cnt_unlink)
    x[:, 8] = np.log1p(cnt_open)
    x[:, 9] = np.log1p(bytes_read) / 16.0
    x[:, 10] = np.log1p(bytes_write) / 16.0
    x[:, 11] = bytes_write / np.maximum(bytes_read + bytes_write, 1.0)
    x[:, 12] = cnt_rename / np.maximum(cnt_write + cnt_rename, 1.0)  # write-to-rename mix
    x[:, 13] = suspicious
    x[:, 14] = double_ext
    x[:, 15] = dur / span
    x[:, 16] = (t_first - t0) / span
    x[:, 17] = (t_last - t0) / span
    x[:, 18] = np.log1p(cnt_total / np.maximum(dur, 1.0))  # burstiness
    x[:, 19] = np.log1p(dur / np.maximum(cnt_total, 1.0))  # mean interarrival
    x[:, 20] = note
    x[:, 21] = recon
    x[:, 22] = ((cnt_rename > 0) & (cnt_unlink > 0)).astype(np.float32)
    x[:, 23] = np.log1p(peer)
    x[:, 24] = np.log1p(cnt_total)
    x[:, 25] = np.log1p(bytes_write / np.maximum(cnt_write, 1.0)) / 16.0
    x[:, 26] = np.log1p(cnt_exec)
    if proc_identity_enabled():
        x[:, 27] = ed["trusted_proc"]
    # 28..31 reserved

    # ---- labels -----------------------------------------------------------
    y_node = None
    y_edge = None
    if y_event is not None and n_ev:
        y_node_f = np.zeros(n_nodes, dtype=np.float32)
        sel = (ev_file >= 0) & (y_event > 0.5)
        if sel.any():
            y_node_f[np.unique(ev_file[sel])] = 1.0
        selp = (ev_proc >= 0) & (y_event > 0.5)
        if selp.any():
            y_node_f[np.unique(ev_proc[selp])] = 1.0
        y_node = y_node_f
        if edge_index.shape[1]:
            y_edge = (y_node[edge_index[0]] * y_node[edge_index[1]]).astype(np.float32)

    node_key = np.concatenate(
        [touched_roots, upids.astype(np.int64)]
    ) if n_nodes else np.empty(0, dtype=np.int64)

    return TemporalGraph(
        x=x,
        edge_index=edge_index,
        edge_weight=edge_weight,
        edge_ts=edge_ts,
        node_kind=node_kind,
        node_key=node_key,
        y_node=y_node,
        y_edge=y_edge,
        t0=t0,
        t1=t1,
    )


def sliding_windows(
    events: EventArray,
    window_s: float = 30.0,
    stride_s: float = 15.0,
):
    """Yield (t_start, EventArray) windows over a time-sorted trace."""
    if len(events) == 0:
        return
    t0 = float(events.ts[0])
    t_end = float(events.ts[-1])
    t = t0
    while t <= t_end:
        win = events.time_window(t, t + window_s)
        if len(win):
            yield t, win
        t += stride_s
