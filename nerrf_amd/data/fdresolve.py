"""Userspace fd->path fallback resolver for replayed traces.

The eBPF program resolves write/read paths in-kernel via the openat-time
fd map (tracker/bpf/nerrf_tracepoints.c, reference M2 spec
implementation.mdx:520-563).  Traces captured WITHOUT that map — the
upstream tracker's recorded streams, or replays from older captures —
carry write events with no path.  This resolver reconstructs them after
the fact from the event stream itself:

  * exact mode: when openat events carry the returned fd in `ret_val`
    (the daemon fills ret_val from sys_exit), maintain a per-(pid, fd)
    open-file table and look writes up by (pid, fd in `ret_val`);
  * heuristic mode: when fds were not recorded (the upstream artifact
    schema), attribute each pathless write/read to the most recent
    still-plausible openat by the same pid (last-opened-wins — the same
    approximation the reference's own docs describe for M1 data).

Pure columnar numpy; used by the replay loaders and unit-tested on
synthetic streams (tests/test_native_ingest.py).
"""
from __future__ import annotations

import numpy as np

from .trace import SYSCALL_IDS, EventArray


def resolve_fd_paths(arr: EventArray, use_ret_fd: bool = True) -> EventArray:
    """Return a copy of `arr` with pathless write/read events resolved.

    `use_ret_fd`: treat openat `ret_val` >= 0 as the returned fd and
    write/read `ret_val` as the consumed fd (exact mode).  Falls back to
    last-open-by-pid per event when the fd key is unknown.
    """
    n = len(arr)
    if n == 0:
        return arr
    path_id = arr.path_id.copy()
    sc = arr.syscall
    is_open = sc == SYSCALL_IDS["openat"]
    is_rw = (sc == SYSCALL_IDS["write"]) | (sc == SYSCALL_IDS["read"])
    is_close = sc == SYSCALL_IDS.get("close", -99)

    fd_table: dict = {}       # (pid, fd) -> path_id
    last_open: dict = {}      # pid -> path_id
    for i in range(n):
        p = int(arr.pid[i])
        if is_open[i] and path_id[i] >= 0:
            last_open[p] = int(path_id[i])
            if use_ret_fd and arr.ret_val[i] >= 0:
                fd_table[(p, int(arr.ret_val[i]))] = int(path_id[i])
        elif is_close[i]:
            fd_table.pop((p, int(arr.ret_val[i])), None)
        elif is_rw[i] and path_id[i] < 0:
            resolved = -1
            if use_ret_fd and arr.ret_val[i] >= 0:
                resolved = fd_table.get((p, int(arr.ret_val[i])), -1)
            if resolved < 0:
                resolved = last_open.get(p, -1)
            path_id[i] = resolved
    return EventArray(
        paths=arr.paths, comms=arr.comms, ts=arr.ts, pid=arr.pid,
        syscall=arr.syscall, path_id=path_id, new_path_id=arr.new_path_id,
        nbytes=arr.nbytes, ret_val=arr.ret_val, comm_id=arr.comm_id,
    )
