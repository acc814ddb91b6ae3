"""Ground-truth label derivation.

Upstream ground truth is an attack-window CSV (start/end timestamps + target
directory) rather than per-event labels (reference:
/root/reference/benchmarks/m1/scripts/m1_minikube_bootstrap.sh:219-224).
Per-event / per-node labels are derived as documented in the threat model
(event inside window AND path under target, or issued by the attack process —
reference docs threat-model.mdx:108-118).
"""
from __future__ import annotations

from typing import Optional

import numpy as np

from .synth import AttackWindow
from .trace import EventArray


def event_labels(arr: EventArray, window: Optional[AttackWindow]) -> np.ndarray:
    """Binary malicious label per event (float32, shape [N])."""
    n = len(arr)
    y = np.zeros(n, dtype=np.float32)
    if window is None or n == 0:
        return y
    in_window = (arr.ts >= window.t_start) & (arr.ts < window.t_end)
    if not in_window.any():
        return y
    # path under target dir?
    target = window.target_dir.rstrip("/") + "/"
    path_is_target = np.zeros(len(arr.paths) + 1, dtype=bool)
    for pid_, s in enumerate(arr.paths.strings):
        if s.startswith(target):
            path_is_target[pid_] = True
    under_target = path_is_target[np.where(arr.path_id >= 0, arr.path_id, len(arr.paths))]
    under_target |= path_is_target[
        np.where(arr.new_path_id >= 0, arr.new_path_id, len(arr.paths))
    ]
    y[in_window & under_target] = 1.0
    return y


def malicious_pids(arr: EventArray, y_event: np.ndarray) -> np.ndarray:
    """PIDs that issued at least one malicious event."""
    if y_event.sum() == 0:
        return np.empty(0, dtype=np.int64)
    return np.unique(arr.pid[y_event > 0.5])
