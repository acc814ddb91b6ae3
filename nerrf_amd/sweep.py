"""Hyper-parameter search over the joint model.

The reference's training roadmap calls for "GraphSAGE-T + hparam search"
(reference ROADMAP.md:62-69); this is a deterministic grid/random search over
the config tree with ROC-AUC selection, runnable on CPU or GPU:

    python -m nerrf_amd.sweep --trials 8 --set data.n_scenarios=4
"""
from __future__ import annotations

import argparse
import itertools
import json
from typing import Dict, List, Sequence

import numpy as np

from .config import load_config
from .train import run_training

DEFAULT_SPACE: Dict[str, Sequence] = {
    "optim.lr": [3e-4, 1e-3, 3e-3],
    "model.sage.hidden": [64, 128],
    "model.sage.layers": [8, 28],
    "model.pos_weight": [2.0, 4.0, 8.0],
}


def run_sweep(
    base_overrides: List[str],
    space: Dict[str, Sequence] = DEFAULT_SPACE,
    trials: int = 8,
    seed: int = 0,
    metric: str = "node_auc",
) -> List[dict]:
    keys = sorted(space)
    grid = list(itertools.product(*(space[k] for k in keys)))
    rng = np.random.default_rng(seed)
    picks = rng.permutation(len(grid))[: min(trials, len(grid))]
    results = []
    for t, gi in enumerate(picks):
        combo = dict(zip(keys, grid[gi]))
        overrides = base_overrides + [f"{k}={v}" for k, v in combo.items()]
        cfg = load_config(None, overrides)
        cfg.run.checkpoint_dir = f"{cfg.run.checkpoint_dir}_sweep{t}"
        report = run_training(cfg)
        results.append({"trial": t, "params": combo, "metrics": report})
        print(json.dumps(results[-1], default=float))
    results.sort(key=lambda r: -float(r["metrics"].get(metric, float("-inf"))))
    print(json.dumps({"best": results[0]}, default=float))
    return results


def main(argv=None) -> None:
    ap = argparse.ArgumentParser(description="nerrf-amd hparam sweep")
    ap.add_argument("--trials", type=int, default=8)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--metric", default="node_auc")
    ap.add_argument("--set", dest="overrides", action="append", default=[])
    args = ap.parse_args(argv)
    run_sweep(args.overrides, trials=args.trials, seed=args.seed, metric=args.metric)


if __name__ == "__main__":
    main()
