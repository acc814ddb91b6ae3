"""Evaluation metrics.

Definitions follow the reference threat model (threat-model.mdx:279-318):
precision / recall / F1 / ROC-AUC for detection; MTTR (detect -> recover) and
data-loss for the recovery loop.  ROC-AUC is implemented directly (rank
statistic) so the hot path has no sklearn dependency; tests cross-check
against sklearn.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

import numpy as np


def roc_auc(y_true: np.ndarray, y_score: np.ndarray) -> float:
    """Mann-Whitney U / rank-based AUC with tie handling."""
    y_true = np.asarray(y_true).astype(bool).ravel()
    y_score = np.asarray(y_score, dtype=np.float64).ravel()
    n_pos = int(y_true.sum())
    n_neg = int((~y_true).sum())
    if n_pos == 0 or n_neg == 0:
        return float("nan")
    order = np.argsort(y_score, kind="mergesort")
    sorted_scores = y_score[order]
    # average ranks with ties
    ranks = np.empty(len(y_score), dtype=np.float64)
    i = 0
    r = 1.0
    n = len(y_score)
    while i < n:
        j = i
        while j + 1 < n and sorted_scores[j + 1] == sorted_scores[i]:
            j += 1
        avg = (r + r + (j - i)) / 2.0
        ranks[order[i : j + 1]] = avg
        r += j - i + 1
        i = j + 1
    sum_pos = ranks[y_true].sum()
    return float((sum_pos - n_pos * (n_pos + 1) / 2.0) / (n_pos * n_neg))


def precision_recall_f1(
    y_true: np.ndarray, y_score: np.ndarray, threshold: float = 0.5
) -> Dict[str, float]:
    y_true = np.asarray(y_true).astype(bool).ravel()
    pred = np.asarray(y_score).ravel() >= threshold
    tp = int((pred & y_true).sum())
    fp = int((pred & ~y_true).sum())
    fn = int((~pred & y_true).sum())
    precision = tp / (tp + fp) if tp + fp else 0.0
    recall = tp / (tp + fn) if tp + fn else 0.0
    f1 = 2 * precision * recall / (precision + recall) if precision + recall else 0.0
    return {"precision": precision, "recall": recall, "f1": f1, "tp": tp, "fp": fp, "fn": fn}


def best_f1(y_true: np.ndarray, y_score: np.ndarray, n_grid: int = 101) -> Dict[str, float]:
    """F1 at the best threshold over a probability grid."""
    best = {"f1": -1.0, "threshold": 0.5}
    for thr in np.linspace(0.0, 1.0, n_grid):
        m = precision_recall_f1(y_true, y_score, float(thr))
        if m["f1"] > best["f1"]:
            best = {**m, "threshold": float(thr)}
    return best


@dataclass
class RecoveryMetrics:
    """MTTR / data-loss bookkeeping for a recovery run."""

    detect_ts: float
    recover_ts: float
    bytes_lost: int
    files_restored: int
    files_total: int

    @property
    def mttr_s(self) -> float:
        return max(self.recover_ts - self.detect_ts, 0.0)

    def as_dict(self) -> Dict[str, float]:
        return {
            "mttr_s": self.mttr_s,
            "bytes_lost": float(self.bytes_lost),
            "files_restored": float(self.files_restored),
            "files_total": float(self.files_total),
            "restore_rate": self.files_restored / max(self.files_total, 1),
        }


def operating_point(y: np.ndarray, scores: np.ndarray, max_fp_frac: float = 0.05) -> Dict[str, float]:
    """Fixed-FP-budget operating point (reference README.md:23-27:
    "FP-undo < 5%"): the lowest threshold whose flagged set stays under
    `max_fp_frac` false positives, i.e. precision >= 1 - max_fp_frac.
    Returns threshold, precision, recall (and flagged count) there."""
    y = np.asarray(y).astype(bool)
    s = np.asarray(scores, dtype=np.float64)
    order = np.argsort(-s, kind="stable")
    ys = y[order]
    tp = np.cumsum(ys)
    fp = np.cumsum(~ys)
    n = len(ys)
    prec = tp / np.maximum(tp + fp, 1)
    ok = np.nonzero(prec >= 1.0 - max_fp_frac)[0]
    if len(ok) == 0 or not y.any():
        return {"threshold": 1.0, "precision": 1.0, "recall": 0.0, "flagged": 0.0}
    k = int(ok[-1])  # largest flagged set still within budget
    thr = float(s[order][k])
    return {
        "threshold": thr,
        "precision": float(prec[k]),
        "recall": float(tp[k] / max(int(y.sum()), 1)),
        "flagged": float(k + 1),
    }


def detection_report(
    y_node: np.ndarray,
    node_scores: np.ndarray,
    y_seq: Optional[np.ndarray] = None,
    seq_scores: Optional[np.ndarray] = None,
    y_edge: Optional[np.ndarray] = None,
    edge_scores: Optional[np.ndarray] = None,
) -> Dict[str, float]:
    rep: Dict[str, float] = {}
    rep["node_auc"] = roc_auc(y_node, node_scores)
    rep.update({f"node_{k}": v for k, v in best_f1(y_node, node_scores).items()})
    if y_seq is not None and seq_scores is not None and len(y_seq):
        rep["seq_auc"] = roc_auc(y_seq, seq_scores)
        rep.update({f"seq_{k}": v for k, v in best_f1(y_seq, seq_scores).items()})
    if y_edge is not None and edge_scores is not None and len(y_edge):
        rep["edge_auc"] = roc_auc(y_edge, edge_scores)
    return rep
