"""Per-file event sequences for the BiLSTM branch.

Spec: the sequence model consumes the last 100 events per file (reference
docs architecture.mdx:55-59; rolling-window example threat-model.mdx:192-203).
Sequences are emitted as a dense [B, T, E] tensor (padded, time-major inside
the model) plus lengths, so thousands of per-file streams batch into one
fused-LSTM launch.

Fully vectorised (no per-file Python loop): the trailing-`seq_len` selection
per file and every feature channel are computed with flat numpy indexing —
~20x faster than the groupby loop on 16k-file windows, which dominated the
serving path's host time.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np

from .trace import EventArray

SEQ_LEN = 100
NUM_SEQ_FEATURES = 16


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
    from ..graph.constructor import _string_flag_bits

    sus = (_string_flag_bits(events.paths) & 1).astype(np.float32)

    idx = np.nonzero(events.path_id >= 0)[0]
    if idx.size == 0:
        return SequenceBatch(
            feats=np.zeros((0, seq_len, NUM_SEQ_FEATURES), dtype=np.float32),
            lengths=np.zeros(0, dtype=np.int64),
            file_path_id=np.zeros(0, dtype=np.int64),
            labels=None if y_event is None else np.zeros(0, dtype=np.float32),
        )
    pids = events.path_id[idx]
    # stable sort keeps time order per file; narrow keys hit numpy's radix
    # path (uint16: measured 10x faster than the int64 mergesort on 600k keys)
    sort_key = pids.astype(np.uint16) if pids.max() < 2**16 else pids
    order = np.argsort(sort_key, kind="stable")
    idx = idx[order]
    pids = pids[order]

    # group extents over the sorted-by-file index array
    starts = np.concatenate([[0], np.nonzero(np.diff(pids))[0] + 1])
    ends = np.concatenate([starts[1:], [len(pids)]])
    sizes = ends - starts
    keep = sizes >= min_events
    starts, ends, sizes = starts[keep], ends[keep], sizes[keep]
    b = len(starts)
    if b == 0:
        return SequenceBatch(
            feats=np.zeros((0, seq_len, NUM_SEQ_FEATURES), dtype=np.float32),
            lengths=np.zeros(0, dtype=np.int64),
            file_path_id=np.zeros(0, dtype=np.int64),
            labels=None if y_event is None else np.zeros(0, dtype=np.float32),
        )
    lengths = np.minimum(sizes, seq_len).astype(np.int64)
    sel_start = ends - lengths

    # flat selection: for sequence g, positions 0..len_g-1 map to
    # idx[sel_start_g + pos]
    total = int(lengths.sum())
    seq_of = np.repeat(np.arange(b), lengths)
    pos = np.arange(total) - np.repeat(np.cumsum(lengths) - lengths, lengths)
    flat = np.repeat(sel_start, lengths) + pos
    ev = idx[flat]

    feats = np.zeros((b, seq_len, NUM_SEQ_FEATURES), dtype=np.float32)
    flatf = feats.reshape(-1)
    # single flat linear-index scatter per channel (a triple fancy-index
    # assignment iterates element-wise and was ~6x slower)
    base = (seq_of * seq_len + pos) * NUM_SEQ_FEATURES
    sc = events.syscall[ev].astype(np.int64)
    # one-hot channels 0..9; network ids (connect/sendto, 10+) share channel
    # 9 until the per-event feature map widens (same clip in the torch path)
    flatf[base + np.clip(sc, 0, 9)] = 1.0
    flatf[base + 10] = np.log1p(events.nbytes[ev]) / 16.0
    ts = events.ts[ev]
    prev = idx[np.maximum(flat - 1, 0)]
    flatf[base + 11] = np.log1p(ts - np.where(pos > 0, events.ts[prev], ts))
    flatf[base + 12] = sus[events.path_id[ev]]
    np_ids = events.new_path_id[ev]
    flatf[base + 13] = np.where(np_ids >= 0, sus[np.clip(np_ids, 0, None)], 0.0)
    flatf[base + 14] = ((pos > 0) & (events.pid[ev] == events.pid[prev])).astype(np.float32)

    fids = events.path_id[idx[sel_start]]
    labels = None
    if y_event is not None:
        labels = np.zeros(b, dtype=np.float32)
        np.maximum.at(labels, seq_of, y_event[ev])

    return SequenceBatch(feats=feats, lengths=lengths, file_path_id=fids, labels=labels)


def build_sequences_torch(
    events: EventArray,
    device,
    seq_len: int = SEQ_LEN,
    min_events: int = 2,
    dtype=None,
    dev_cols=None,
):
    """GPU sequence assembly: same semantics as build_sequences, in torch.

    The serving host path spent ~57 ms per 600k-event window on the numpy
    build; on the GPU the grouping is a radix argsort plus a handful of
    flat scatters (~2 ms), and the 25 MB feature tensor never crosses PCIe.
    Returns (feats [B,T,E] on device, lengths [B] cpu, file_path_id [B] cpu).
    """
    import torch

    from ..graph.constructor import _string_flag_bits

    dev = torch.device(device)
    sus_np = (_string_flag_bits(events.paths) & 1).astype(np.float32)

    def empty():
        f = torch.zeros(0, seq_len, NUM_SEQ_FEATURES, device=dev, dtype=dtype or torch.float32)
        z = torch.zeros(0, dtype=torch.int64)
        return f, z, z

    if not len(events):
        return empty()
    if dev_cols is not None:  # HBM delta ring: columns already on-device
        t_path = dev_cols["path_id"]
        t_newp = dev_cols["new_path_id"]
        t_sc = dev_cols["syscall"]
        t_nb = dev_cols["nbytes_f32"]
        t_ts = dev_cols["ts"]
        t_pid = dev_cols["pid"]
    else:
        t_path = torch.from_numpy(events.path_id).to(dev, non_blocking=True)
        t_newp = torch.from_numpy(events.new_path_id).to(dev, non_blocking=True)
        t_sc = torch.from_numpy(events.syscall).to(dev, non_blocking=True)
        t_nb = torch.from_numpy(np.ascontiguousarray(events.nbytes, dtype=np.float32)).to(dev, non_blocking=True)
        t_ts = torch.from_numpy(events.ts).to(dev, non_blocking=True)  # float64: epoch ts
        t_pid = torch.from_numpy(events.pid).to(dev, non_blocking=True)
    t_sus = torch.from_numpy(sus_np).to(dev, non_blocking=True)

    idx = (t_path >= 0).nonzero(as_tuple=True)[0]
    if idx.numel() == 0:
        return empty()
    pids = t_path[idx]
    order = torch.argsort(pids, stable=True)
    idx = idx[order]
    pids = pids[order]
    n = idx.numel()
    new_grp = torch.ones(n, dtype=torch.bool, device=dev)
    new_grp[1:] = pids[1:] != pids[:-1]
    starts = new_grp.nonzero(as_tuple=True)[0]
    ends = torch.cat([starts[1:], torch.tensor([n], device=dev)])
    sizes = ends - starts
    keep = sizes >= min_events
    starts, ends, sizes = starts[keep], ends[keep], sizes[keep]
    b = int(starts.numel())
    if b == 0:
        return empty()
    lengths = sizes.clamp(max=seq_len)
    sel_start = ends - lengths
    seq_of = torch.repeat_interleave(torch.arange(b, device=dev), lengths)
    cum = torch.cumsum(lengths, 0) - lengths
    pos = torch.arange(seq_of.numel(), device=dev) - cum[seq_of]
    flat = sel_start[seq_of] + pos
    ev = idx[flat]
    prev = idx[(flat - 1).clamp(min=0)]

    feats = torch.zeros(b * seq_len * NUM_SEQ_FEATURES, device=dev, dtype=torch.float32)
    base = (seq_of * seq_len + pos) * NUM_SEQ_FEATURES
    feats[base + t_sc[ev].long().clamp(0, 9)] = 1.0  # same clip as numpy path
    feats[base + 10] = torch.log1p(t_nb[ev]) / 16.0
    ts_ev = t_ts[ev]
    dt = ts_ev - torch.where(pos > 0, t_ts[prev], ts_ev)
    feats[base + 11] = torch.log1p(dt).float()
    feats[base + 12] = t_sus[t_path[ev]]
    np_ids = t_newp[ev]
    feats[base + 13] = torch.where(np_ids >= 0, t_sus[np_ids.clamp(min=0)], torch.zeros((), device=dev))
    feats[base + 14] = ((pos > 0) & (t_pid[ev] == t_pid[prev])).float()
    feats = feats.view(b, seq_len, NUM_SEQ_FEATURES)
    if dtype is not None:
        feats = feats.to(dtype)
    fids = t_path[idx[sel_start]].cpu()
    return feats, lengths.cpu(), fids
