"""Per-file event sequences for the BiLSTM branch.

Spec: the sequence model consumes the last 100 events per file (reference
docs architecture.mdx:55-59; rolling-window example threat-model.mdx:192-203).
Sequences are emitted as a dense [B, T, E] tensor (padded, time-major inside
the model) plus lengths, so thousands of per-file streams batch into one
fused-LSTM launch.
"""
from __future__ import annotations

import re
from dataclasses import dataclass
from typing import Optional, Tuple

import numpy as np

from .trace import EventArray

SEQ_LEN = 100
NUM_SEQ_FEATURES = 16

_SUSPICIOUS_EXT = re.compile(r"\.(lockbit\w*|encrypted|locked|crypt\w*)$", re.IGNORECASE)


@dataclass
class SequenceBatch:
    feats: np.ndarray  # [B, T, E] float32
    lengths: np.ndarray  # [B] int64
    file_path_id: np.ndarray  # [B] int64 (representative path id)
    labels: Optional[np.ndarray] = None  # [B] float32


def build_sequences(
    events: EventArray,
    y_event: Optional[np.ndarray] = None,
    seq_len: int = SEQ_LEN,
    min_events: int = 2,
) -> SequenceBatch:
    """Group events by file, keep the trailing `seq_len` per file."""
    n = len(events)
    sus = np.zeros(len(events.paths), dtype=np.float32)
    for i, s in enumerate(events.paths.strings):
        if _SUSPICIOUS_EXT.search(s):
            sus[i] = 1.0

    valid = events.path_id >= 0
    idx = np.nonzero(valid)[0]
    if idx.size == 0:
        return SequenceBatch(
            feats=np.zeros((0, seq_len, NUM_SEQ_FEATURES), dtype=np.float32),
            lengths=np.zeros(0, dtype=np.int64),
            file_path_id=np.zeros(0, dtype=np.int64),
            labels=None if y_event is None else np.zeros(0, dtype=np.float32),
        )
    pids = events.path_id[idx]
    order = np.argsort(pids, kind="stable")  # stable keeps time order per file
    idx = idx[order]
    pids = pids[order]
    boundaries = np.nonzero(np.diff(pids))[0] + 1
    groups = np.split(idx, boundaries)
    groups = [g for g in groups if g.size >= min_events]
    b = len(groups)

    feats = np.zeros((b, seq_len, NUM_SEQ_FEATURES), dtype=np.float32)
    lengths = np.zeros(b, dtype=np.int64)
    fids = np.zeros(b, dtype=np.int64)
    labels = np.zeros(b, dtype=np.float32) if y_event is not None else None

    for bi, g in enumerate(groups):
        g = g[-seq_len:]
        t = g.size
        lengths[bi] = t
        fids[bi] = events.path_id[g[0]]
        sc = events.syscall[g].astype(np.int64)
        f = feats[bi, :t]
        one_hot_cols = np.clip(sc, 0, 9)
        f[np.arange(t), one_hot_cols] = 1.0
        f[:, 10] = np.log1p(events.nbytes[g]) / 16.0
        ts = events.ts[g]
        dt = np.diff(ts, prepend=ts[0])
        f[:, 11] = np.log1p(dt)
        f[:, 12] = sus[events.path_id[g]]
        np_ids = events.new_path_id[g]
        f[:, 13] = np.where(np_ids >= 0, sus[np.clip(np_ids, 0, None)], 0.0)
        pid_seq = events.pid[g]
        f[1:, 14] = (pid_seq[1:] == pid_seq[:-1]).astype(np.float32)
        if labels is not None:
            labels[bi] = float(y_event[g].max())

    return SequenceBatch(feats=feats, lengths=lengths, file_path_id=fids, labels=labels)
