"""Incremental delta-graph compaction (per-delta summaries).

The streaming store (reference README.md:114: "30 s delta compaction") holds
the window as immutable ~5 s delta chunks.  The full host build
(`build_graph_parts` + `build_edges_and_flags`) re-scans every event each
scoring tick; this module caches a small per-delta summary once per sealed
delta and re-derives the window graph each tick by merging summaries —
O(unique keys per delta) instead of O(events in window):

  * rename pairs             -> scipy connected components per tick (renames
                                can leave the window, so unions must be
                                recomputed from the surviving deltas);
  * unique (proc, path, dir) edge keys with partial recency sums
    sum(exp((ts - t_d0)/tau)) and per-key max ts — exact up to fp
    associativity: the per-delta partial is rescaled by exp((t_d0 - t1)/tau)
    at merge time so no absolute-epoch exponentials are ever formed;
  * per-path byte sums and indicator counters (additive across deltas).

Only the per-event node-id arrays (`ev_file`/`ev_proc`, consumed by the GPU
feature-compaction kernel) are still O(window events): two LUT gathers.

Parity: tests/test_graph.py asserts the merged (parts, ed) against the full
rebuild on identical event sets (edge order, ids and degrees exact; weights
to fp tolerance).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np

from ..data.trace import SYSCALL_IDS, EventArray
from .constructor import (_READ_LIKE, _path_flags, aggregate_sparse_keys,
                          proc_identity_enabled, rename_path_roots,
                          trusted_proc_flags)

_SCALE_GUARD = 60.0  # max (t1 - t_d0)/tau before a partial sum underflows


class DeltaSummary:
    __slots__ = (
        "t0", "t1", "n_ev", "ren_a", "ren_b", "upids", "upaths",
        "key_pid", "key_path", "key_dir", "key_wsum", "key_tmax",
        "mb_ids", "mb_vals", "n_writes", "n_renames",
    )


def summarize_delta(d: EventArray, causality_tau_s: float = 10.0) -> DeltaSummary:
    """One pass over a sealed delta -> merge-ready summary."""
    s = DeltaSummary()
    s.n_ev = len(d)
    s.t0 = float(d.ts[0]) if s.n_ev else 0.0
    s.t1 = float(d.ts[-1]) if s.n_ev else 0.0

    ren_mask = (d.syscall == SYSCALL_IDS["rename"]) & (d.new_path_id >= 0)
    ra, rb = d.path_id[ren_mask], d.new_path_id[ren_mask]
    keep = (ra >= 0) & (rb >= 0)
    s.ren_a, s.ren_b = ra[keep], rb[keep]
    s.n_renames = int(ren_mask.sum())
    s.n_writes = int((d.syscall == SYSCALL_IDS["write"]).sum())

    pid_mask = d.path_id >= 0
    np_mask = d.new_path_id >= 0
    upaths = np.unique(np.concatenate([d.path_id[pid_mask], d.new_path_id[np_mask]]))
    s.upaths = upaths.astype(np.int64)
    s.upids = np.unique(d.pid).astype(np.int64) if s.n_ev else np.empty(0, np.int64)

    # per-path byte sums (sparse pairs — the global path table keeps growing)
    if pid_mask.any():
        mb = np.bincount(d.path_id[pid_mask], weights=d.nbytes[pid_mask].astype(np.float64))
        nz = np.nonzero(mb)[0]
        s.mb_ids, s.mb_vals = nz.astype(np.int64), mb[nz]
    else:
        s.mb_ids = np.empty(0, np.int64)
        s.mb_vals = np.empty(0, np.float64)

    # unique (pid, path, dir) keys with recency partials relative to t_d0
    valid = pid_mask
    if valid.any():
        pids = d.pid[valid].astype(np.int64)
        paths = d.path_id[valid].astype(np.int64)
        read_like = np.zeros(int(d.syscall.max()) + 1, dtype=bool)
        for sc in _READ_LIKE:
            if sc < len(read_like):
                read_like[sc] = True
        dirs = read_like[d.syscall[valid]].astype(np.int64)
        ts_v = d.ts[valid]
        # compact local key: pids/paths are arbitrary -> local dense ranks
        up = np.unique(pids)
        ua = np.unique(paths)
        pk = np.searchsorted(up, pids)
        ak = np.searchsorted(ua, paths)
        key = (pk * len(ua) + ak) * 2 + dirs
        kspace = 2 * len(up) * len(ua)
        rec = np.exp((ts_v - s.t0) / causality_tau_s)
        # guarded aggregation (bincount / radix) — same size guard as the
        # full rebuild, so a delta with many procs*paths never allocates
        # kspace-sized dense arrays
        uk, s.key_wsum, s.key_tmax = aggregate_sparse_keys(key, rec, ts_v, kspace)
        kd = uk % 2
        kp = uk // 2
        s.key_pid = up[kp // len(ua)]
        s.key_path = ua[kp % len(ua)]
        s.key_dir = kd
    else:
        s.key_pid = s.key_path = s.key_dir = np.empty(0, np.int64)
        s.key_wsum = s.key_tmax = np.empty(0, np.float64)
    return s


class IncrementalWindowState:
    """Summary cache keyed by delta identity (deltas are immutable).

    Also holds the stable-prefix EDGE cache: the aggregated edge table over
    the sealed summaries (everything but the trailing open delta), stored in
    the tick-stable key domain (pid, path_root, dir) with recency sums at a
    reference time.  Exponential decay is multiplicative in time —
    sum_i exp(-(t1-ts_i)/tau) = exp(-(t1-t_ref)/tau) * sum_i
    exp(-(t_ref-ts_i)/tau) — so a later tick rescales the cached sums with
    ONE multiply instead of re-aggregating ~0.5M per-delta keys.  The cache
    invalidates when the sealed-summary set changes (seal/expiry, every
    ~delta_s) or when the window's rename-pair set changes (path roots, and
    with them the stable key domain, can merge)."""

    def __init__(self, causality_tau_s: float = 10.0) -> None:
        self.tau = causality_tau_s
        self._cache: Dict[int, Tuple[EventArray, DeltaSummary]] = {}
        self._edge_cache: Optional[dict] = None
        # (ren_a, ren_b, n_paths, path_root) — rename CC memo; path_root is
        # shared read-only with callers via parts["path_root"]
        self._root_cache: Optional[tuple] = None

    @staticmethod
    def _token(prefix: list) -> tuple:
        # identity + shape guard: id() alone could be reused after gc
        return tuple((id(s), s.n_ev, s.t0, s.t1) for s in prefix)

    def edge_cache_get(self, prefix: list, ren_a: np.ndarray, ren_b: np.ndarray):
        c = self._edge_cache
        if c is None or c["token"] != self._token(prefix):
            return None
        if not (np.array_equal(c["ren_a"], ren_a) and np.array_equal(c["ren_b"], ren_b)):
            return None
        return c

    def edge_cache_put(self, prefix: list, ren_a: np.ndarray, ren_b: np.ndarray,
                       pid: np.ndarray, root: np.ndarray, kdir: np.ndarray,
                       conf: np.ndarray, last: np.ndarray, t_ref: float) -> None:
        self._edge_cache = {
            "token": self._token(prefix),
            "ren_a": ren_a.copy(), "ren_b": ren_b.copy(),
            "pid": pid, "root": root, "dir": kdir,
            "conf": conf.copy(), "last": last.copy(), "t_ref": t_ref,
        }

    def summaries(self, deltas: List[EventArray]) -> List[DeltaSummary]:
        out = []
        live = set()
        for d in deltas:
            k = id(d)
            live.add(k)
            hit = self._cache.get(k)
            if hit is None or hit[0] is not d:
                hit = (d, summarize_delta(d, self.tau))
                self._cache[k] = hit
            out.append(hit[1])
        for k in list(self._cache):
            if k not in live:
                del self._cache[k]
        return out


def merge_window(
    events: EventArray,
    summaries: List[DeltaSummary],
    causality_tau_s: float = 10.0,
    device=None,
    dev_cols=None,
    state: Optional["IncrementalWindowState"] = None,
) -> Tuple[dict, dict]:
    """Merge per-delta summaries into (parts, ed) — drop-in for
    build_graph_parts + build_edges_and_flags over the same events.

    `events` is the compacted window (same delta set, already needed for the
    GPU feature/sequence kernels); only its per-event id columns are touched
    here (two LUT gathers), never re-aggregated.
    """
    summaries = [s for s in summaries if s.n_ev]
    n_ev = len(events)
    t0 = float(events.ts[0]) if n_ev else 0.0
    t1 = float(events.ts[-1]) if n_ev else 0.0
    span = max(t1 - t0, 1e-6)
    n_paths = len(events.paths)

    # ---- rename components over the surviving deltas ----------------------
    if summaries:
        ra = np.concatenate([s.ren_a for s in summaries])
        rb = np.concatenate([s.ren_b for s in summaries])
    else:
        ra = rb = np.empty(0, np.int64)
    if state is not None:
        rc = state._root_cache
        if (rc is not None and rc[2] == n_paths
                and np.array_equal(rc[0], ra) and np.array_equal(rc[1], rb)):
            path_root = rc[3]
        else:
            path_root = rename_path_roots(ra, rb, n_paths)
            state._root_cache = (ra.copy(), rb.copy(), n_paths, path_root)
    else:
        path_root = rename_path_roots(ra, rb, n_paths)

    # ---- node tables from merged unique sets ------------------------------
    seen = np.zeros(n_paths, dtype=bool)
    for s in summaries:
        seen[path_root[s.upaths]] = True
    touched_roots = np.nonzero(seen)[0].astype(np.int64)
    n_files = len(touched_roots)
    upids = (
        np.unique(np.concatenate([s.upids for s in summaries]))
        if summaries
        else np.empty(0, np.int64)
    )
    n_procs = len(upids)
    n_nodes = n_files + n_procs

    root_to_file = np.full(n_paths, -1, dtype=np.int64)
    if n_files:
        root_to_file[touched_roots] = np.arange(n_files, dtype=np.int64)
    pi = events.path_id
    pid_lut = None
    if n_procs and int(upids.max()) < (1 << 22) and int(upids.min()) >= 0:
        pid_lut = np.full(int(upids.max()) + 1, -1, dtype=np.int64)
        pid_lut[upids] = np.arange(n_procs, dtype=np.int64)

    def pid_to_local(p: np.ndarray) -> np.ndarray:
        if pid_lut is not None:
            return pid_lut[p]
        return np.searchsorted(upids, p)

    if device is not None and n_ev:
        # per-event node-id maps on the GPU: the feature-compaction kernel is
        # their only consumer there, and two 600k-row gathers cost ~10 ms of
        # host time vs ~50 us on-device (LUTs are tiny; pi/pid ship instead
        # of the derived columns — same bytes over PCIe)
        import torch

        # path_id/pid come from the HBM delta ring when available (shipped
        # once per sealed delta) instead of re-crossing PCIe every tick
        if dev_cols is not None:
            t_pi = dev_cols["path_id"]
            t_pid = dev_cols["pid"]
        else:
            t_pi = torch.from_numpy(pi).to(device, non_blocking=True)
            t_pid = torch.from_numpy(events.pid).to(device, non_blocking=True)
        t_root = torch.from_numpy(path_root).to(device, non_blocking=True)
        t_rtf = torch.from_numpy(root_to_file).to(device, non_blocking=True)
        ev_file = torch.where(
            t_pi >= 0,
            t_rtf[t_root[t_pi.clamp(min=0)]],
            torch.full((), -1, dtype=torch.int64, device=device),
        )
        if pid_lut is not None:
            t_lut = torch.from_numpy(pid_lut).to(device, non_blocking=True)
            ev_proc = n_files + t_lut[t_pid]
        else:
            t_up = torch.from_numpy(upids).to(device, non_blocking=True)
            ev_proc = n_files + torch.searchsorted(t_up, t_pid)
    else:
        ev_file = np.where(pi >= 0, root_to_file[path_root[np.clip(pi, 0, None)]], -1)
        ev_proc = (
            n_files + pid_to_local(events.pid) if n_ev else np.empty(0, np.int64)
        )

    parts = {
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

    # ---- edge aggregation from per-delta partials -------------------------
    def _agg(sums: List[DeltaSummary]):
        """Aggregate a summary subset into (uk, conf, last), sorted by the
        current node-domain key ((proc*n_files + file)*2 + dir)."""
        proc_l, file_l, dir_l, wsum_l, tmax_l = [], [], [], [], []
        for s in sums:
            if not len(s.key_pid):
                continue
            age = (t1 - s.t0) / causality_tau_s
            scale = np.exp(-age) if age < _SCALE_GUARD else 0.0
            fnode = root_to_file[path_root[s.key_path]]
            pnode = pid_to_local(s.key_pid)
            proc_l.append(pnode)
            file_l.append(fnode)
            dir_l.append(s.key_dir)
            wsum_l.append(s.key_wsum * scale)
            tmax_l.append(s.key_tmax)
        if not proc_l:
            e = np.empty(0, np.int64)
            return e, np.empty(0, np.float64), np.empty(0, np.float64)
        key = (np.concatenate(proc_l) * n_files + np.concatenate(file_l)) * 2 \
            + np.concatenate(dir_l)
        kspace = 2 * n_procs * n_files
        # guarded aggregation mirroring the full rebuild's key_space
        # check: the production run_monitor tick must never
        # allocate kspace-dense arrays for a wide window
        return aggregate_sparse_keys(
            key, np.concatenate(wsum_l), np.concatenate(tmax_l), kspace)

    if summaries and n_nodes:
        # stable-prefix cache (see IncrementalWindowState): the sealed
        # summaries' aggregation is reused across ticks — only the trailing
        # (open) delta is re-aggregated and merged in.
        prefix, tail = summaries[:-1], summaries[-1:]
        agg_p = None
        can_cache = state is not None and len(prefix) >= 1
        if can_cache:
            hit = state.edge_cache_get(prefix, ra, rb)
            if hit is not None:
                # remap stable (pid, root, dir) into the current node domain:
                # both maps are monotone (upids/touched_roots sorted), so the
                # cached sort order is preserved
                key_c = (pid_to_local(hit["pid"]) * n_files
                         + root_to_file[hit["root"]]) * 2 + hit["dir"]
                age = (t1 - hit["t_ref"]) / causality_tau_s
                rescale = np.exp(-age) if age < _SCALE_GUARD else 0.0
                agg_p = (key_c, hit["conf"] * rescale, hit["last"].copy())
        if agg_p is None:
            agg_p = _agg(prefix)
            if can_cache and len(agg_p[0]):
                uk_p = agg_p[0]
                kp = uk_p // 2
                state.edge_cache_put(
                    prefix, ra, rb,
                    pid=upids[kp // n_files],
                    root=touched_roots[kp % n_files],
                    kdir=uk_p % 2,
                    conf=agg_p[1], last=agg_p[2], t_ref=t1,
                )
        uk_t, conf_t, last_t = _agg(tail)
        uk, e_conf, e_last = agg_p
        if len(uk_t):
            if len(uk):
                # merge two sorted unique key sets (sum conf, max last)
                pos = np.searchsorted(uk, uk_t)
                dup = pos < uk.size
                dup[dup] &= uk[pos[dup]] == uk_t[dup]
                pd = pos[dup]
                e_conf[pd] += conf_t[dup]
                e_last[pd] = np.maximum(e_last[pd], last_t[dup])
                ins = ~dup
                if ins.any():
                    uk = np.insert(uk, pos[ins], uk_t[ins])
                    e_conf = np.insert(e_conf, pos[ins], conf_t[ins])
                    e_last = np.insert(e_last, pos[ins], last_t[ins])
            else:
                uk, e_conf, e_last = uk_t, conf_t, last_t
        if len(uk):
            kd = uk % 2
            kp = uk // 2
            e_proc = n_files + kp // n_files
            e_file = kp % n_files
            src = np.where(kd == 0, e_proc, e_file)
            dst = np.where(kd == 0, e_file, e_proc)
            edge_index = np.stack([src, dst]).astype(np.int64)
            edge_weight = (1.0 - np.exp(-e_conf)).astype(np.float32)
            edge_ts = ((e_last - t0) / span).astype(np.float32)
        else:
            edge_index = np.zeros((2, 0), dtype=np.int64)
            edge_weight = np.zeros(0, dtype=np.float32)
            edge_ts = np.zeros(0, dtype=np.float32)
    else:
        edge_index = np.zeros((2, 0), dtype=np.int64)
        edge_weight = np.zeros(0, dtype=np.float32)
        edge_ts = np.zeros(0, dtype=np.float32)

    if edge_index.shape[1] and n_nodes:
        out_deg = np.bincount(edge_index[0], minlength=n_nodes).astype(np.float64)
        in_deg = np.bincount(edge_index[1], minlength=n_nodes).astype(np.float64)
        peer = out_deg + in_deg
    else:
        peer = np.zeros(n_nodes, dtype=np.float64)
        in_deg = np.zeros(n_nodes, dtype=np.float64)
        out_deg = np.zeros(n_nodes, dtype=np.float64)

    suspicious, note, recon, double_ext = _path_flags(
        events.paths, path_root, root_to_file, n_files
    )
    if proc_identity_enabled() and n_ev:
        # host per-event proc map (the device branch keeps ev_proc on-GPU,
        # but the comm allowlist is string-domain host work either way)
        ev_proc_host = n_files + pid_to_local(events.pid)
        trusted = trusted_proc_flags(events, ev_proc_host, n_files, n_procs, n_nodes)
    else:
        trusted = np.zeros(n_nodes, dtype=np.float32)
    pad = np.zeros(n_procs, dtype=np.float32)
    ed = {
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
        "trusted_proc": trusted,
    }
    return parts, ed
