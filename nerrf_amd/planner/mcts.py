"""MCTS rollback planner.

Spec (reference architecture.mdx:64-72): 500-1000 simulations, reward =
restoration gain - side effects, <= 5 min planning budget, output = ranked
undo candidates (file reversion / process kill / restore).

Design: root-parallel MCTS — S independent trees (the "1024 parallel sims"
of BASELINE.json config 4), each tree arena-allocated in flat arrays, UCB1
tree policy, xorshift32 rollouts, terminal rewards from rewards.simulate_plan.
Root statistics are summed across trees for the final ranking.  The same
algorithm runs on CPU (this file, numpy) and on GPU
(ops/hip/mcts.hip: one wave per tree, arenas in HBM) — the CPU version is the
numerics ground truth for the kernel tests.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import numpy as np

from .rewards import (A_KILL, A_RESTORE, A_REVERT_BASE, A_STOP, PlannerParams,
                      PlannerState, simulate_plan)

UCB_C = 1.2


def xorshift32(s: int) -> int:
    """The shared CPU/GPU rollout RNG (must match mcts.hip bit-for-bit)."""
    s &= 0xFFFFFFFF
    s ^= (s << 13) & 0xFFFFFFFF
    s ^= s >> 17
    s ^= (s << 5) & 0xFFFFFFFF
    return s & 0xFFFFFFFF


@dataclass
class PlanResult:
    plan: List[int]  # greedy most-visited action path (ends before STOP)
    ranked_actions: List[Tuple[int, float, int]]  # (action, mean value, visits) at root
    root_value: float
    simulations: int

    def describe(self, n_groups: int) -> List[str]:
        out = []
        for a in self.plan:
            if a == A_KILL:
                out.append("kill_process")
            elif a == A_RESTORE:
                out.append("restore_from_backup")
            elif a >= A_REVERT_BASE:
                out.append(f"revert_group_{a - A_REVERT_BASE}")
        return out


class _Tree:
    """Flat-array MCTS tree arena (mirrors the GPU layout)."""

    __slots__ = ("parent", "action", "visits", "value", "children", "n_nodes", "n_actions", "cap")

    def __init__(self, n_actions: int, cap: int) -> None:
        self.parent = np.full(cap, -1, dtype=np.int32)
        self.action = np.full(cap, -1, dtype=np.int32)
        self.visits = np.zeros(cap, dtype=np.int32)
        self.value = np.zeros(cap, dtype=np.float32)
        self.children = np.full((cap, n_actions), -1, dtype=np.int32)
        self.n_nodes = 1  # root
        self.n_actions = n_actions
        self.cap = cap


def _path_actions(tree: _Tree, node: int) -> List[int]:
    path = []
    while node != 0:
        path.append(int(tree.action[node]))
        node = int(tree.parent[node])
    path.reverse()
    return path


def run_mcts(
    state: PlannerState,
    params: Optional[PlannerParams] = None,
    n_sims: int = 1024,
    sims_per_tree: int = 32,
    seed: int = 0,
) -> PlanResult:
    """Root-parallel MCTS: n_sims total simulations over n_sims//sims_per_tree trees."""
    params = params or PlannerParams()
    n_actions = 3 + state.n_groups  # STOP, KILL, RESTORE, revert g
    n_trees = max(1, n_sims // sims_per_tree)
    root_visits = np.zeros(n_actions, dtype=np.int64)
    root_value = np.zeros(n_actions, dtype=np.float64)
    total_sims = 0

    for ti in range(n_trees):
        tree = _Tree(n_actions, cap=sims_per_tree * params.max_depth + 2)
        rng_state = (seed * 2654435761 + ti * 40503 + 1) & 0xFFFFFFFF
        for si in range(sims_per_tree):
            # ---- selection ----
            node = 0
            depth = 0
            while depth < params.max_depth:
                kids = tree.children[node]
                untried = np.nonzero(kids < 0)[0]
                if len(untried):
                    # ---- expansion: first untried action (deterministic) ----
                    a = int(untried[0])
                    new = tree.n_nodes
                    tree.n_nodes += 1
                    tree.parent[new] = node
                    tree.action[new] = a
                    tree.children[node][a] = new
                    node = new
                    depth += 1
                    break
                # fully expanded: UCB1
                n_parent = max(int(tree.visits[node]), 1)
                best, best_u = 0, -np.inf
                for a in range(n_actions):
                    ch = int(kids[a])
                    nv = int(tree.visits[ch])
                    if nv == 0:
                        u = np.inf
                    else:
                        q = float(tree.value[ch]) / nv
                        u = q + UCB_C * np.sqrt(np.log(n_parent) / nv)
                    if u > best_u:
                        best, best_u = a, u
                node = int(kids[best])
                depth += 1
                if tree.action[node] == A_STOP:
                    break
            # ---- rollout ----
            path = _path_actions(tree, node)
            actions = list(path)
            rng_state = xorshift32(rng_state ^ (si * 747796405 + 2891336453 & 0xFFFFFFFF))
            r = rng_state
            while len(actions) < params.max_depth and (not actions or actions[-1] != A_STOP):
                r = xorshift32(r)
                a = r % n_actions
                actions.append(int(a))
                if a == A_STOP:
                    break
            reward = simulate_plan(state, actions, params)
            # ---- backup ----
            nd = node
            while nd >= 0:
                tree.visits[nd] += 1
                tree.value[nd] += reward
                nd = int(tree.parent[nd])
            total_sims += 1
        # accumulate root stats
        for a in range(n_actions):
            ch = int(tree.children[0][a])
            if ch >= 0:
                root_visits[a] += int(tree.visits[ch])
                root_value[a] += float(tree.value[ch])

    return _result_from_root_stats(state, params, root_visits, root_value, total_sims)


def _result_from_root_stats(
    state: PlannerState,
    params: PlannerParams,
    root_visits: np.ndarray,  # [n_actions] summed over trees
    root_value: np.ndarray,
    total_sims: int,
) -> PlanResult:
    n_actions = 3 + state.n_groups
    mean_val = np.where(root_visits > 0, root_value / np.maximum(root_visits, 1), -np.inf)
    ranked = sorted(
        [(int(a), float(mean_val[a]), int(root_visits[a])) for a in range(n_actions)],
        key=lambda t: (-t[2], -t[1]),
    )
    # greedy plan: walk the visit ranking, keep an action only if it improves
    # the simulated reward of the plan built so far
    plan: List[int] = []
    best_reward = simulate_plan(state, [A_STOP], params)
    for a, v, n in ranked:
        if a == A_STOP or len(plan) >= params.max_depth:
            continue
        cand = plan + [a]
        r = simulate_plan(state, cand + [A_STOP], params)
        if r > best_reward:
            plan = cand
            best_reward = r
    return PlanResult(plan=plan, ranked_actions=ranked, root_value=float(best_reward), simulations=total_sims)


def run_mcts_gpu(
    state: PlannerState,
    params: Optional[PlannerParams] = None,
    n_sims: int = 1024,
    sims_per_tree: int = 32,
    seed: int = 0,
    device: str = "cuda",
) -> PlanResult:
    """Batched MCTS on the CDNA4 kernel (ops/hip/mcts.hip): one wave per tree."""
    import torch

    from ..ops.native import load_extension

    ext = load_extension(required=True)
    params = params or PlannerParams()
    n_trees = max(1, n_sims // sims_per_tree)
    pd = {
        "n_groups": state.n_groups,
        "max_depth": params.max_depth,
        "sims_per_tree": sims_per_tree,
        "downtime_weight": params.downtime_weight,
        "revert_time_s": params.revert_time_s,
        "kill_time_s": params.kill_time_s,
        "fp_weight": params.fp_weight,
        "attack_rate_mbps": params.attack_rate_mbps,
        "horizon_s": params.horizon_s,
        "restore_time_s": params.restore_time_s,
        "restore_loss_mb": params.restore_loss_mb,
        "ucb_c": UCB_C,
        "seed": seed & 0xFFFFFFFF,
    }
    dev = torch.device(device)
    gs = torch.from_numpy(state.group_score.astype(np.float32)).to(dev)
    gm = torch.from_numpy(state.group_mb.astype(np.float32)).to(dev)
    gf = torch.from_numpy(state.group_files.astype(np.float32)).to(dev)
    rv, rw = ext.mcts_search(gs, gm, gf, float(state.proc_score), float(state.remaining_clean_mb), pd, n_trees)
    root_visits = rv.sum(dim=0).cpu().numpy().astype(np.int64)
    root_value = rw.sum(dim=0).double().cpu().numpy()
    return _result_from_root_stats(state, params, root_visits, root_value, n_trees * sims_per_tree)
