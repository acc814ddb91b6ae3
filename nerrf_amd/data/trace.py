"""Trace loading and canonical columnar representation.

The engine consumes syscall traces from three sources:
  * live gRPC ``Tracker/StreamEvents`` (nerrf_amd.serve.tracker_client),
  * ND-JSON benchmark artifacts in the upstream layout — one JSON object per
    line with keys ``timestamp,event,path,size,pid[,phase,file_type]``
    (reference artifact schema: /root/reference/benchmarks/m0/results/m0_trace.jsonl),
  * ``datasets/traces/toy_trace.csv`` (same columns, CSV).

Everything is normalised into :class:`EventArray` — a columnar batch
(structure-of-arrays) so that host→device staging is a handful of contiguous
copies instead of per-event Python objects.
"""
from __future__ import annotations

import csv
import json
import re
from dataclasses import dataclass
from pathlib import Path
from typing import Dict, Iterable, List, Optional, Sequence

import numpy as np

# Syscall ids 1..3 mirror the kernel tracker's encoding
# (reference: /root/reference/tracker/bpf/tracepoints.c:43-80); 4+ are the
# documented M2 extensions (reference docs tracker/implementation.mdx:520-563).
SYSCALL_IDS: Dict[str, int] = {
    "openat": 1,
    "write": 2,
    "rename": 3,
    "read": 4,
    "unlink": 5,
    "chmod": 6,
    "close": 7,
    "mkdir": 8,
    "exec": 9,
    # network egress (graph spec lists socket nodes: reference docs
    # architecture.mdx:36-43); destinations enter the path domain as
    # "tcp://host:port" strings
    "connect": 10,
    "sendto": 11,
}
SYSCALL_NAMES = {v: k for k, v in SYSCALL_IDS.items()}
UNKNOWN_SYSCALL = 0

# Event names used by the upstream jsonl artifacts -> canonical syscall.
# The second block is the recorded m0/m1 simulator vocabulary
# (reference sim_lockbit_m1.py TRACE: records, present verbatim in
# /root/reference/benchmarks/m{0,1}/results/*_trace.jsonl): seed/encrypt
# phases are write flows, the recon enumerations are exec-like.
_JSONL_EVENT_MAP = {
    "open": "openat",
    "openat": "openat",
    "create": "openat",
    "write": "write",
    "encrypt": "write",
    "rename": "rename",
    "delete": "unlink",
    "unlink": "unlink",
    "read": "read",
    "recon": "exec",
    "exec": "exec",
    "chmod": "chmod",
    # recorded-artifact vocabulary
    "file_created": "write",
    "file_encrypt_start": "write",
    "file_encrypt_complete": "rename",  # path is the NEW .lockbit3 name
    "ransom_note_created": "write",
    "process_enum": "exec",
    "network_enum": "exec",
    "user_enum": "exec",
    "disk_enum": "exec",
    "mount_enum": "exec",
    "lateral_movement_start": "exec",
    "lateral_movement_complete": "exec",
}

# the recorded simulator renames <stem>.dat -> <stem>.lockbit3 and logs only
# the new name on file_encrypt_complete (sim_lockbit_m1.py encrypt loop);
# reconstruct the rename pair so rename-union and the extension indicators
# see the same graph a live tracker would have produced
_ENCRYPTED_SUFFIX = re.compile(r"\.(lockbit\w*|encrypted|locked|crypt\w*)$", re.IGNORECASE)


def _coerce_ts(v) -> float:
    """Accept epoch floats or ISO-8601 strings (the recorded artifacts use
    naive-UTC ISO, e.g. '2025-08-30T14:07:06.542871')."""
    if isinstance(v, (int, float)):
        return float(v)
    s = str(v).strip()
    try:
        return float(s)
    except ValueError:
        pass
    from datetime import datetime, timezone

    if s.endswith("Z"):
        s = s[:-1]
    dt = datetime.fromisoformat(s)
    if dt.tzinfo is None:
        dt = dt.replace(tzinfo=timezone.utc)
    return dt.timestamp()


class StringTable:
    """Bidirectional string <-> dense-id interner (paths, comms)."""

    def __init__(self) -> None:
        self._to_id: Dict[str, int] = {}
        self._strings: List[str] = []

    def intern(self, s: str) -> int:
        idx = self._to_id.get(s)
        if idx is None:
            idx = len(self._strings)
            self._to_id[s] = idx
            self._strings.append(s)
        return idx

    def lookup(self, idx: int) -> str:
        return self._strings[idx]

    def get(self, s: str) -> Optional[int]:
        return self._to_id.get(s)

    def __len__(self) -> int:
        return len(self._strings)

    @property
    def strings(self) -> List[str]:
        return self._strings


@dataclass
class EventArray:
    """Columnar syscall-event batch (all arrays share length N)."""

    ts: np.ndarray  # float64 seconds (UTC)
    pid: np.ndarray  # int64
    syscall: np.ndarray  # int8 (SYSCALL_IDS)
    path_id: np.ndarray  # int64 into paths table
    new_path_id: np.ndarray  # int64, -1 if absent
    nbytes: np.ndarray  # int64
    ret_val: np.ndarray  # int64
    comm_id: np.ndarray  # int64 into comms table
    paths: StringTable
    comms: StringTable

    def __len__(self) -> int:
        return int(self.ts.shape[0])

    def sort_by_time(self) -> "EventArray":
        if len(self.ts) == 0 or bool(np.all(self.ts[1:] >= self.ts[:-1])):
            return self  # streaming deltas arrive near-sorted; skip the sort
        order = np.argsort(self.ts, kind="stable")
        return EventArray(
            ts=self.ts[order],
            pid=self.pid[order],
            syscall=self.syscall[order],
            path_id=self.path_id[order],
            new_path_id=self.new_path_id[order],
            nbytes=self.nbytes[order],
            ret_val=self.ret_val[order],
            comm_id=self.comm_id[order],
            paths=self.paths,
            comms=self.comms,
        )

    def slice(self, start: int, stop: int) -> "EventArray":
        return EventArray(
            ts=self.ts[start:stop],
            pid=self.pid[start:stop],
            syscall=self.syscall[start:stop],
            path_id=self.path_id[start:stop],
            new_path_id=self.new_path_id[start:stop],
            nbytes=self.nbytes[start:stop],
            ret_val=self.ret_val[start:stop],
            comm_id=self.comm_id[start:stop],
            paths=self.paths,
            comms=self.comms,
        )

    def time_window(self, t0: float, t1: float) -> "EventArray":
        """Events with t0 <= ts < t1 (assumes time-sorted)."""
        lo = int(np.searchsorted(self.ts, t0, side="left"))
        hi = int(np.searchsorted(self.ts, t1, side="left"))
        return self.slice(lo, hi)


class EventArrayBuilder:
    def __init__(self, paths: Optional[StringTable] = None, comms: Optional[StringTable] = None):
        self.paths = paths if paths is not None else StringTable()
        self.comms = comms if comms is not None else StringTable()
        self._ts: List[float] = []
        self._pid: List[int] = []
        self._sys: List[int] = []
        self._path: List[int] = []
        self._newp: List[int] = []
        self._bytes: List[int] = []
        self._ret: List[int] = []
        self._comm: List[int] = []

    def add(
        self,
        ts: float,
        pid: int,
        syscall: str,
        path: str = "",
        new_path: str = "",
        nbytes: int = 0,
        ret_val: int = 0,
        comm: str = "",
    ) -> None:
        self._ts.append(float(ts))
        self._pid.append(int(pid))
        self._sys.append(SYSCALL_IDS.get(syscall, UNKNOWN_SYSCALL))
        self._path.append(self.paths.intern(path) if path else -1)
        self._newp.append(self.paths.intern(new_path) if new_path else -1)
        self._bytes.append(int(nbytes))
        self._ret.append(int(ret_val))
        self._comm.append(self.comms.intern(comm) if comm else -1)

    def add_wire_event(self, ev) -> None:  # nerrf_amd.wire.codec.Event
        self.add(
            ts=ev.timestamp,
            pid=ev.pid,
            syscall=ev.syscall,
            path=ev.path,
            new_path=ev.new_path,
            nbytes=ev.bytes,
            ret_val=ev.ret_val,
            comm=ev.comm,
        )

    def build(self, sort: bool = True) -> EventArray:
        arr = EventArray(
            ts=np.asarray(self._ts, dtype=np.float64),
            pid=np.asarray(self._pid, dtype=np.int64),
            syscall=np.asarray(self._sys, dtype=np.int8),
            path_id=np.asarray(self._path, dtype=np.int64),
            new_path_id=np.asarray(self._newp, dtype=np.int64),
            nbytes=np.asarray(self._bytes, dtype=np.int64),
            ret_val=np.asarray(self._ret, dtype=np.int64),
            comm_id=np.asarray(self._comm, dtype=np.int64),
            paths=self.paths,
            comms=self.comms,
        )
        return arr.sort_by_time() if sort else arr

    def __len__(self) -> int:
        return len(self._ts)


def _normalise_event_name(name: str) -> str:
    return _JSONL_EVENT_MAP.get(name, name)


def load_jsonl(path: str | Path) -> EventArray:
    """Load an ND-JSON trace in the upstream benchmark artifact schema."""
    builder = EventArrayBuilder()
    with open(path, "r", encoding="utf-8") as fh:
        for line in fh:
            line = line.strip()
            if not line:
                continue
            try:
                rec = json.loads(line)
            except json.JSONDecodeError:
                continue  # tolerate truncated/corrupt lines in field traces
            name = str(rec.get("event", ""))
            path = str(rec.get("path", ""))
            new_path = str(rec.get("new_path", ""))
            if name == "file_encrypt_complete" and not new_path:
                m = _ENCRYPTED_SUFFIX.search(path)
                if m:
                    # reconstruct the logged-new-name-only rename pair
                    new_path = path
                    path = path[: m.start()] + ".dat"
            builder.add(
                ts=_coerce_ts(rec.get("timestamp", 0.0)),
                pid=int(rec.get("pid", 0)),
                syscall=_normalise_event_name(name),
                path=path,
                new_path=new_path,
                nbytes=int(rec.get("size", rec.get("bytes", 0)) or 0),
            )
    return builder.build()


def load_csv(path: str | Path) -> EventArray:
    """Load a CSV trace (toy_trace.csv layout: same columns as jsonl)."""
    builder = EventArrayBuilder()
    with open(path, "r", encoding="utf-8", newline="") as fh:
        for rec in csv.DictReader(fh):
            builder.add(
                ts=_coerce_ts(rec.get("timestamp", 0.0) or 0.0),
                pid=int(rec.get("pid", 0) or 0),
                syscall=_normalise_event_name(str(rec.get("event", ""))),
                path=str(rec.get("path", "")),
                new_path=str(rec.get("new_path", "") or ""),
                nbytes=int(float(rec.get("size", 0) or 0)),
            )
    return builder.build()


def load_trace(path: str | Path) -> EventArray:
    p = Path(path)
    if p.suffix == ".csv":
        return load_csv(p)
    return load_jsonl(p)


def from_wire_events(events: Iterable) -> EventArray:
    builder = EventArrayBuilder()
    for ev in events:
        builder.add_wire_event(ev)
    return builder.build()


def write_csv(path: str | Path, arr: EventArray) -> None:
    with open(path, "w", encoding="utf-8", newline="") as fh:
        w = csv.writer(fh)
        w.writerow(["timestamp", "event", "path", "new_path", "size", "pid"])
        for i in range(len(arr)):
            pid_ = int(arr.pid[i])
            path_ = arr.paths.lookup(int(arr.path_id[i])) if arr.path_id[i] >= 0 else ""
            newp = arr.paths.lookup(int(arr.new_path_id[i])) if arr.new_path_id[i] >= 0 else ""
            w.writerow(
                [
                    f"{arr.ts[i]:.6f}",
                    SYSCALL_NAMES.get(int(arr.syscall[i]), "unknown"),
                    path_,
                    newp,
                    int(arr.nbytes[i]),
                    pid_,
                ]
            )


def concat(arrays: Sequence[EventArray]) -> EventArray:
    """Concatenate event batches sharing no tables — re-interns strings."""
    builder = EventArrayBuilder()
    for arr in arrays:
        for i in range(len(arr)):
            builder.add(
                ts=float(arr.ts[i]),
                pid=int(arr.pid[i]),
                syscall=SYSCALL_NAMES.get(int(arr.syscall[i]), "unknown"),
                path=arr.paths.lookup(int(arr.path_id[i])) if arr.path_id[i] >= 0 else "",
                new_path=arr.paths.lookup(int(arr.new_path_id[i])) if arr.new_path_id[i] >= 0 else "",
                nbytes=int(arr.nbytes[i]),
                ret_val=int(arr.ret_val[i]),
                comm=arr.comms.lookup(int(arr.comm_id[i])) if arr.comm_id[i] >= 0 else "",
            )
    return builder.build()
