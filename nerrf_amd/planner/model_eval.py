"""Model-in-the-loop plan refinement: batched counterfactual leaf scoring.

SURVEY.md §7 asks for "batched leaf evaluation through the GNN+LSTM": the
MCTS rollouts use the closed-form reward (rewards.simulate_plan), which
trusts the detector's per-group scores as fixed expectations.  This module
closes the loop with the model itself: for the top candidate plans it
builds the COUNTERFACTUAL post-plan window graph — reverted groups' file
nodes with their attack-pattern channels cleared, the attacker process
neutralised when the plan kills it — and re-scores all candidates through
the GNN node head in ONE batched forward (disjoint graph union, exactly
like training batches).  A plan whose counterfactual still scores hot
(e.g. reverting decoys while the real target group stays encrypted) gets
its reward debited by the residual model risk.

This runs once per planning call (a single extra forward for ~8 candidate
plans), not per simulation — the spec budget is 5 minutes/plan
(reference architecture.mdx:68-72); this costs milliseconds.
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np

from .rewards import A_KILL, A_RESTORE, A_REVERT_BASE, A_STOP, PlannerParams, PlannerState

# feature channels an executed revert resets on a file node — the attack
# signature channels of graph/constructor.build_graph (write/rename/unlink
# counters, byte flows, suspicious/double-ext/note flags, rename+unlink
# combo, per-write size).  Degrees/temporal channels stay: the history of
# events is real even after the files are restored.
REVERT_CLEAR_CHANNELS = (5, 6, 7, 10, 11, 12, 13, 14, 20, 22, 25)
# channels a KILL clears on process nodes (activity counters + recon flag)
KILL_CLEAR_CHANNELS = (4, 5, 6, 7, 8, 9, 10, 18, 21, 24, 26)


def counterfactual_features(
    x: np.ndarray,
    plan: Sequence[int],
    group_nodes: Sequence[np.ndarray],
    proc_nodes: np.ndarray,
) -> np.ndarray:
    """Apply a plan's effects to a copy of the node-feature matrix."""
    cf = x.copy()
    restored = any(a == A_RESTORE for a in plan)
    for a in plan:
        if a == A_STOP:
            break
        if a == A_KILL and len(proc_nodes):
            cf[np.asarray(proc_nodes)[:, None], list(KILL_CLEAR_CHANNELS)] = 0.0
        elif a >= A_REVERT_BASE:
            gi = a - A_REVERT_BASE
            if 0 <= gi < len(group_nodes) and len(group_nodes[gi]):
                cf[np.asarray(group_nodes[gi])[:, None], list(REVERT_CLEAR_CHANNELS)] = 0.0
    if restored:
        for nodes in group_nodes:
            if len(nodes):
                cf[np.asarray(nodes)[:, None], list(REVERT_CLEAR_CHANNELS)] = 0.0
    return cf


def refine_plans_with_model(
    model,
    x: np.ndarray,              # [N, F] window node features
    nbr_idx: np.ndarray,        # [N, K] sampled neighbor index
    nbr_w: np.ndarray,          # [N, K]
    group_nodes: Sequence[np.ndarray],  # planner group -> node ids
    proc_nodes: np.ndarray,     # attacker-candidate process node ids
    candidates: List[List[int]],
    state: PlannerState,
    params: Optional[PlannerParams] = None,
    risk_weight: float = 10.0,
    device: str = "cpu",
    dtype=None,
) -> List[tuple]:
    """Re-rank candidate plans by closed-form reward minus batched model risk.

    Returns [(plan, combined, closed_form, residual_risk)] sorted best-first.
    The model forward runs ONCE over the disjoint union of every candidate's
    counterfactual graph.
    """
    import torch

    from .rewards import simulate_plan

    params = params or PlannerParams()
    n = x.shape[0]
    if n == 0 or not candidates:
        return [(p, simulate_plan(state, p, params), simulate_plan(state, p, params), 0.0)
                for p in candidates]

    cfs = [counterfactual_features(x, p, group_nodes, proc_nodes) for p in candidates]
    big_x = np.concatenate(cfs, axis=0)
    # disjoint union: each copy's neighbor ids shift by its node offset
    big_idx = np.concatenate(
        [nbr_idx + i * n for i in range(len(cfs))], axis=0
    )
    big_w = np.concatenate([nbr_w] * len(cfs), axis=0)

    dev = torch.device(device)
    dt = dtype if dtype is not None else next(model.parameters()).dtype
    batch = {
        "x": torch.from_numpy(big_x).to(dev, dt),
        "nbr_idx": torch.from_numpy(big_idx).to(dev),
        "nbr_w": torch.from_numpy(big_w.astype(np.float32)).to(dev),
    }
    with torch.no_grad():
        h = model.gnn.encode(batch["x"], batch["nbr_idx"], batch["nbr_w"])
        node_logit = model.gnn.node_head(h).squeeze(-1)
    scores = torch.sigmoid(node_logit.float()).cpu().numpy().reshape(len(cfs), n)

    watched = (
        np.concatenate([np.asarray(g) for g in group_nodes if len(g)])
        if any(len(g) for g in group_nodes)
        else np.arange(n)
    )
    out = []
    for plan, sc in zip(candidates, scores):
        closed = simulate_plan(state, list(plan) + [A_STOP], params)
        residual = float(sc[watched].mean()) if len(watched) else 0.0
        out.append((list(plan), closed - risk_weight * residual, closed, residual))
    out.sort(key=lambda t: -t[1])
    return out
