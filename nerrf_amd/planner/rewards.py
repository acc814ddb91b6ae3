"""Rollback-plan reward model.

Spec (reference README.md:115, architecture.mdx:64-72, threat-model.mdx:206-222):
  reward = restoration gain - side effects  ==  -(data_loss + 0.1 * downtime)

Concrete MDP used by both the CPU planner and the CDNA4 batched kernel:

State after executing a partial plan:
  * each candidate file group g has expected encrypted-but-unrecovered bytes;
  * the attack process is alive or killed; while alive, encryption continues
    at `attack_rate_mbps` against remaining un-encrypted target data;
  * every action costs wall time (downtime) and wrong actions on clean files
    add false-positive side effects (counted as data made unavailable).

All quantities are expectations under the detector's anomaly scores, so the
planner maximises  -(E[data_loss_MB] + 0.1 * downtime_s + fp_penalty_MB).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List

import numpy as np

# action ids (shared with the HIP kernel — keep in sync with mcts.hip)
A_STOP = 0
A_KILL = 1
A_RESTORE = 2  # restore everything from backup (reference threat-model.mdx:206-222)
A_REVERT_BASE = 3  # A_REVERT_BASE + g : revert file group g


@dataclass
class PlannerParams:
    n_groups: int = 8
    downtime_weight: float = 0.1
    revert_time_s: float = 0.05  # per-file rename-back (reference measured ~1 ms/file; batched overhead)
    kill_time_s: float = 0.5
    fp_weight: float = 1.0  # MB-equivalent penalty per wrongly-reverted MB
    attack_rate_mbps: float = 2.0  # reference simulator rate limit
    horizon_s: float = 60.0
    max_depth: int = 10
    # restore-from-backup (reference threat-model.mdx:206-222: "cost 100 —
    # data loss risk, confidence 1.0"): recovers every group at a large
    # fixed downtime plus the data written since the last backup
    restore_time_s: float = 30.0
    restore_loss_mb: float = 16.0


@dataclass
class PlannerState:
    """Detector output aggregated into file groups (score-bucketed)."""

    group_score: np.ndarray  # [G] mean anomaly score of group
    group_mb: np.ndarray  # [G] total MB in group
    group_files: np.ndarray  # [G] file count
    proc_score: float  # attack-process anomaly score
    remaining_clean_mb: float  # target data not yet encrypted

    @property
    def n_groups(self) -> int:
        return int(len(self.group_score))


def build_state(
    file_scores: np.ndarray,
    file_mb: np.ndarray,
    proc_score: float,
    remaining_clean_mb: float,
    n_groups: int = 8,
) -> PlannerState:
    """Bucket files by anomaly score into n_groups candidate groups."""
    order = np.argsort(-file_scores, kind="stable")
    gs = np.zeros(n_groups, dtype=np.float64)
    gm = np.zeros(n_groups, dtype=np.float64)
    gf = np.zeros(n_groups, dtype=np.float64)
    if len(order):
        split = np.array_split(order, n_groups)
        for g, ids in enumerate(split):
            if len(ids):
                gs[g] = float(file_scores[ids].mean())
                gm[g] = float(file_mb[ids].sum())
                gf[g] = float(len(ids))
    return PlannerState(
        group_score=gs,
        group_mb=gm,
        group_files=gf,
        proc_score=float(proc_score),
        remaining_clean_mb=float(remaining_clean_mb),
    )


def simulate_plan(
    state: PlannerState,
    actions: List[int],
    params: PlannerParams,
) -> float:
    """Expected reward of executing `actions` in order.

    Mirrors the device function `eval_plan` in ops/hip/mcts.hip — any change
    here must be mirrored there (tests assert bit-level agreement in fp32).
    """
    g = state.n_groups
    unrec = state.group_score * state.group_mb  # expected encrypted MB per group
    reverted = np.zeros(g, dtype=np.float64)
    fp_mb = 0.0
    downtime = 0.0
    alive = True
    restored = False
    staleness = 0.0
    ongoing = 0.0  # extra MB encrypted while we act
    for a in actions:
        if a == A_STOP:
            break
        if a == A_KILL:
            if alive:
                downtime += params.kill_time_s
                alive = False
            continue
        if a == A_RESTORE:
            if restored:
                continue
            dt = params.restore_time_s
            if alive:
                ongoing += params.attack_rate_mbps * dt * state.proc_score
            downtime += dt
            reverted[:] = 1.0
            staleness = params.restore_loss_mb
            restored = True
            continue
        gi = a - A_REVERT_BASE
        if gi < 0 or gi >= g or reverted[gi] > 0:
            continue  # invalid / duplicate: no-op
        dt = params.revert_time_s * max(state.group_files[gi], 1.0)
        if alive:
            ongoing += params.attack_rate_mbps * dt * state.proc_score
        downtime += dt
        reverted[gi] = 1.0
        fp_mb += (1.0 - state.group_score[gi]) * state.group_mb[gi] * 0.05
    if alive:  # attack keeps running until the horizon
        ongoing += params.attack_rate_mbps * params.horizon_s * state.proc_score
        # the service-degradation charge scales with our belief the process
        # is actually malicious: a clean state pays ~nothing for inaction
        downtime += params.horizon_s * state.proc_score
    loss = (
        float(((1.0 - reverted) * unrec).sum())
        + min(ongoing, state.remaining_clean_mb)
        + staleness
    )
    return -(loss + params.downtime_weight * downtime + params.fp_weight * fp_mb)
