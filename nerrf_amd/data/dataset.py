"""Window-batch dataset: scenario traces -> training batches.

Each training example is one 30 s sliding window rendered as
(graph tensors + per-file sequences + labels).  Batches are prebuilt on the
CPU (numpy) and staged to the device as a handful of contiguous copies.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Iterator, List, Optional

import numpy as np
import torch

from ..graph.constructor import build_graph, sliding_windows
from ..graph.sampling import edge_reverse_index, reverse_index, sample_fanout, to_csr
from .labels import event_labels
from .sequences import build_sequences
from .synth import AttackWindow, SynthConfig, generate
from .trace import EventArray


@dataclass
class WindowBatch:
    """One window's tensors (numpy, CPU)."""

    x: np.ndarray
    nbr_idx: np.ndarray
    nbr_w: np.ndarray
    edge_index: np.ndarray
    edge_weight: np.ndarray
    edge_ts: np.ndarray
    y_node: np.ndarray
    y_edge: np.ndarray
    seq_feats: np.ndarray
    seq_lengths: np.ndarray
    y_seq: np.ndarray
    n_events: int
    seq_path_id: Optional[np.ndarray] = None  # [B] path id per sequence
    rev_dst: Optional[np.ndarray] = None  # reverse index for gather backward
    rev_src: Optional[np.ndarray] = None
    rev_w: Optional[np.ndarray] = None
    e0_rev: Optional[tuple] = None  # edge endpoint reverse (dst, src, w)
    e1_rev: Optional[tuple] = None

    def to_torch(self, device="cpu", dtype=torch.float32) -> Dict[str, torch.Tensor]:
        def t(a, dt=None):
            x = torch.from_numpy(a)
            if dt is not None:
                x = x.to(dt)
            return x.to(device, non_blocking=True)

        return {
            "x": t(self.x, dtype),
            "nbr_idx": t(self.nbr_idx),
            "nbr_w": t(self.nbr_w, torch.float32),
            "edge_index": t(self.edge_index),
            "edge_weight": t(self.edge_weight, torch.float32),
            "edge_ts": t(self.edge_ts, torch.float32),
            "y_node": t(self.y_node, torch.float32),
            "y_edge": t(self.y_edge, torch.float32),
            "seq_feats": t(self.seq_feats, dtype),
            "seq_lengths": t(self.seq_lengths),
            "y_seq": t(self.y_seq, torch.float32),
            "n_events": torch.tensor(self.n_events),
            "nbr_rev": None
            if self.rev_dst is None
            else (t(self.rev_dst), t(self.rev_src), t(self.rev_w, torch.float32)),
            "edge_rev": None
            if self.e0_rev is None
            else (
                tuple(t(a) for a in self.e0_rev),
                tuple(t(a) for a in self.e1_rev),
            ),
        }


def window_to_batch(
    events: EventArray,
    window: Optional[AttackWindow],
    fanout: int = 16,
    seq_len: int = 100,
    seed: int = 0,
) -> WindowBatch:
    y_ev = event_labels(events, window)
    g = build_graph(events, window, y_ev)
    csr = to_csr(g.edge_index, g.num_nodes, g.edge_weight)
    nbr_idx, nbr_w = sample_fanout(csr, fanout, seed=seed)
    rev_dst, rev_src, rev_w = reverse_index(nbr_idx, nbr_w)
    e0_rev = edge_reverse_index(g.edge_index[0]) if g.num_edges else None
    e1_rev = edge_reverse_index(g.edge_index[1]) if g.num_edges else None
    seqs = build_sequences(events, y_ev, seq_len=seq_len)
    return WindowBatch(
        x=g.x,
        nbr_idx=nbr_idx,
        nbr_w=nbr_w,
        edge_index=g.edge_index,
        edge_weight=g.edge_weight,
        edge_ts=g.edge_ts,
        y_node=g.y_node if g.y_node is not None else np.zeros(g.num_nodes, dtype=np.float32),
        y_edge=g.y_edge if g.y_edge is not None else np.zeros(g.num_edges, dtype=np.float32),
        seq_feats=seqs.feats,
        seq_lengths=seqs.lengths,
        y_seq=seqs.labels if seqs.labels is not None else np.zeros(len(seqs.lengths), dtype=np.float32),
        n_events=len(events),
        seq_path_id=seqs.file_path_id,
        rev_dst=rev_dst,
        rev_src=rev_src,
        rev_w=rev_w,
        e0_rev=e0_rev,
        e1_rev=e1_rev,
    )


def synth_window_batches(
    n_scenarios: int = 8,
    window_s: float = 30.0,
    stride_s: float = 15.0,
    duration_s: float = 120.0,
    benign_rate_hz: float = 800.0,
    attack_fraction: float = 0.6,
    fanout: int = 16,
    seq_len: int = 100,
    base_seed: int = 0,
    kinds: tuple = ("lockbit", "supply_chain"),
    benign_kinds: tuple = ("lockbit", "benign_rotate", "benign_backup", "benign_build"),
    config_jitter: bool = False,
) -> List[WindowBatch]:
    """Prebuild window batches from synthetic scenarios.

    Attack scenarios alternate over `kinds`; clean scenarios alternate over
    `benign_kinds` (plain background plus the hard negatives — log rotation
    and backup daemons — so the model learns to NOT fire on attack
    lookalikes: the FP-undo < 5% target).

    `config_jitter` varies duration and event rate per scenario (x0.5-x2
    around the given values) so the heads stay calibrated on
    off-distribution window configurations — the round-2 calibration
    sweep showed head maxima drifting at unseen duration/rate combos."""
    batches: List[WindowBatch] = []
    for i in range(n_scenarios):
        # interleave attack/benign at the requested ratio over ANY
        # scenario count, spread so small n still gets both classes
        _rank = [0, 2, 4, 6, 8, 1, 3, 5, 7, 9]  # rank of i%10 in a strided order
        is_attack = _rank[i % 10] < round(attack_fraction * 10)
        dur_i, rate_i = duration_s, benign_rate_hz
        if config_jitter:
            # deterministic per-scenario jitter over a 4x span
            dur_i = duration_s * (0.5 + 1.5 * ((i * 29) % 10) / 9.0)
            rate_i = benign_rate_hz * (0.5 + 1.5 * ((i * 37) % 10) / 9.0)
        cfg = SynthConfig(
            duration_s=dur_i,
            benign_rate_hz=rate_i,
            attack=is_attack,
            seed=base_seed + 7919 * i,
            attack_start_frac=0.2 + 0.5 * ((i * 13) % 10) / 10.0,
            kind=kinds[i % len(kinds)] if is_attack else benign_kinds[i % len(benign_kinds)],
        )
        arr, win = generate(cfg)
        for j, (t0, evw) in enumerate(sliding_windows(arr, window_s, stride_s)):
            batches.append(window_to_batch(evw, win, fanout=fanout, seq_len=seq_len, seed=base_seed + i * 131 + j))
    return batches


def iterate_epochs(
    batches: List[WindowBatch],
    epochs: int,
    device="cpu",
    dtype=torch.float32,
    shuffle: bool = True,
    seed: int = 0,
) -> Iterator[Dict[str, torch.Tensor]]:
    rng = np.random.default_rng(seed)
    for _ in range(epochs):
        order = rng.permutation(len(batches)) if shuffle else np.arange(len(batches))
        for i in order:
            yield batches[int(i)].to_torch(device=device, dtype=dtype)


def collate_windows(batches: List[WindowBatch]) -> WindowBatch:
    """Disjoint union of window batches into one training batch.

    Graphs merge block-diagonally (node/edge ids offset, no cross-window
    edges or samples — the model output over the union equals the
    concatenation of per-window outputs), sequences concatenate along B.
    Reverse indexes stay sorted because every window's node ids are offset
    above the previous window's range.
    """
    if len(batches) == 1:
        return batches[0]
    node_off = 0
    edge_off = 0
    xs, nbr_i, nbr_w, ei, ew, ets, yn, ye = [], [], [], [], [], [], [], []
    sf, sl, ysq = [], [], []
    rds, rss, rws = [], [], []
    e0d, e0s, e0w, e1d, e1s, e1w = [], [], [], [], [], []
    n_events = 0
    have_rev = all(b.rev_dst is not None for b in batches)
    have_erev = all(b.e0_rev is not None for b in batches)
    for b in batches:
        n = b.x.shape[0]
        e = b.edge_index.shape[1]
        xs.append(b.x)
        nbr_i.append(b.nbr_idx + node_off)
        nbr_w.append(b.nbr_w)
        ei.append(b.edge_index + node_off)
        ew.append(b.edge_weight)
        ets.append(b.edge_ts)
        yn.append(b.y_node)
        ye.append(b.y_edge)
        sf.append(b.seq_feats)
        sl.append(b.seq_lengths)
        ysq.append(b.y_seq)
        if have_rev:
            rds.append(b.rev_dst + node_off)
            rss.append(b.rev_src + node_off)
            rws.append(b.rev_w)
        if have_erev:
            e0d.append(b.e0_rev[0] + node_off)
            e0s.append(b.e0_rev[1] + edge_off)
            e0w.append(b.e0_rev[2])
            e1d.append(b.e1_rev[0] + node_off)
            e1s.append(b.e1_rev[1] + edge_off)
            e1w.append(b.e1_rev[2])
        node_off += n
        edge_off += e
        n_events += b.n_events
    return WindowBatch(
        x=np.concatenate(xs),
        nbr_idx=np.concatenate(nbr_i),
        nbr_w=np.concatenate(nbr_w),
        edge_index=np.concatenate(ei, axis=1),
        edge_weight=np.concatenate(ew),
        edge_ts=np.concatenate(ets),
        y_node=np.concatenate(yn),
        y_edge=np.concatenate(ye),
        seq_feats=np.concatenate(sf),
        seq_lengths=np.concatenate(sl),
        y_seq=np.concatenate(ysq),
        n_events=n_events,
        seq_path_id=None,  # per-window path tables do not merge
        rev_dst=np.concatenate(rds) if have_rev else None,
        rev_src=np.concatenate(rss) if have_rev else None,
        rev_w=np.concatenate(rws) if have_rev else None,
        e0_rev=(np.concatenate(e0d), np.concatenate(e0s), np.concatenate(e0w)) if have_erev else None,
        e1_rev=(np.concatenate(e1d), np.concatenate(e1s), np.concatenate(e1w)) if have_erev else None,
    )
