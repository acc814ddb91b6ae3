"""HBM-resident delta-graph compaction (GPU feature path).

The division of labour for the streaming edge store ("HBM-
resident ring of delta graphs, GPU compaction kernel"):

  * host: string-domain work — path interning, rename union-find, edge key
    dedup, regex indicator flags (build_graph_parts / build_edges_and_flags);
  * device: everything numeric — the per-event scatter-accumulate over
    millions of events and assembly of the 32-dim feature matrix
    (ops/hip/event_scatter.hip), producing `x` directly in HBM where the
    GNN consumes it.

`gpu_window_graph()` returns the same tensors as TemporalGraph.to_torch();
tests assert column-for-column parity with the CPU build.
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from .constructor import (build_edges_and_flags, build_graph_parts,
                          file_node_kinds, proc_identity_enabled)
from ..data.trace import EventArray


def gpu_window_graph(
    events: EventArray,
    device: torch.device,
    causality_tau_s: float = 10.0,
    dtype: torch.dtype = torch.float32,
    parts: Optional[dict] = None,
    ed: Optional[dict] = None,
    dev_cols: Optional[dict] = None,
) -> Optional[dict]:
    """Build one window's graph with GPU feature compaction.

    Returns dict(x, edge_index, edge_weight, edge_ts, node_kind, node_key)
    on `device`, or None for an empty window.  `parts`/`ed` let a caller that
    already did the host-side identity/edge work (e.g. the streaming engine)
    skip recomputing it.
    """
    from ..ops.native import load_extension

    ext = load_extension(required=True)
    if parts is None:
        parts = build_graph_parts(events, causality_tau_s)
    if parts["n_nodes"] == 0:
        return None
    if ed is None:
        ed = build_edges_and_flags(parts, causality_tau_s)
    n_nodes = parts["n_nodes"]
    n_files, n_procs = parts["n_files"], parts["n_procs"]
    t0, span = parts["t0"], parts["span"]

    node_kind = np.concatenate(
        [file_node_kinds(events.paths, parts["path_root"], parts["touched_roots"]),
         np.zeros(n_procs, dtype=np.int8)]
    )
    flags = (
        ed["suspicious"].astype(np.uint8)
        | (ed["note"].astype(np.uint8) << 1)
        | (ed["recon"].astype(np.uint8) << 2)
        | (ed["double_ext"].astype(np.uint8) << 3)
    )
    if proc_identity_enabled() and "trusted_proc" in ed:
        # bit 4 -> x[:, 27] (feature_assemble_kernel); gate is host-side so
        # the kernel mapping stays unconditional
        flags = flags | (ed["trusted_proc"].astype(np.uint8) << 4)

    def dev(a, dt=None):
        if isinstance(a, torch.Tensor):  # already staged on-device (merge_window)
            t = a
        else:
            t = torch.from_numpy(np.ascontiguousarray(a))
        if dt is not None:
            t = t.to(dt)
        return t.to(device, non_blocking=True)

    if dev_cols is not None:
        # HBM delta ring: event columns already resident on-device
        t_sc = dev_cols["syscall"]
        t_nb = dev_cols["nbytes_f32"]
        t_ts_ms = ((dev_cols["ts"] - t0) * 1000.0).round().to(torch.int32)
    else:
        t_sc = dev(events.syscall)
        t_nb = dev(events.nbytes.astype(np.float32))
        t_ts_ms = dev(np.round((events.ts - t0) * 1000.0).astype(np.int32))
    x = ext.event_features(
        dev(parts["ev_file"]),
        dev(parts["ev_proc"]),
        t_sc,
        t_nb,
        t_ts_ms,
        n_nodes,
        dev(ed["in_deg"].astype(np.float32)),
        dev(ed["out_deg"].astype(np.float32)),
        dev(ed["peer"].astype(np.float32)),
        dev(flags),
        dev(node_kind),
        float(span),
    )
    node_key = np.concatenate([parts["touched_roots"], parts["upids"].astype(np.int64)])
    return {
        "x": x.to(dtype),
        "edge_index": dev(ed["edge_index"]),
        "edge_weight": dev(ed["edge_weight"]),
        "edge_ts": dev(ed["edge_ts"]),
        "node_kind": dev(node_kind),
        "node_key": dev(node_key),
    }
