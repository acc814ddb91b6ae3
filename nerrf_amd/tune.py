"""Hyperparameter search over the joint-model training pipeline.

The reference's roadmap commits the training pipeline to "GraphSAGE-T +
hparam search" (ROADMAP.md:62-69, SURVEY.md §2b "Training pipeline"); this
is that component: deterministic random search over a dotted-override
space, each trial a full run_training() on its own checkpoint dir,
selected by the holdout seq-F1 + node-AUC composite (the same monitor run_training's best-epoch keeping uses).

A space maps dotted config keys (config.py override syntax) to either a
list of choices or a (low, high) tuple sampled log-uniformly:

    space = {
        "optim.lr": (1e-4, 1e-2),
        "model.sage.dropout": [0.0, 0.1, 0.2],
        "model.pos_weight": [2.0, 4.0, 8.0],
    }

Deterministic: trial i's sample depends only on (seed, i), so a search is
reproducible and resumable by trial index.

CLI: ``./nerrf tune --trials 8 --epochs 2`` (cli.py) or
``python -m nerrf_amd.tune``.
"""
from __future__ import annotations

import json
import math
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence

import numpy as np

# the default search space: the knobs the reference's roadmap names
# (learning rate, model capacity/regularisation, loss balance)
DEFAULT_SPACE: Dict[str, Any] = {
    "optim.lr": (2e-4, 5e-3),
    "optim.weight_decay": (1e-5, 1e-3),
    "model.sage.dropout": [0.0, 0.1, 0.2],
    "model.pos_weight": [2.0, 4.0, 8.0],
    "model.w_seq": [0.5, 1.0, 2.0],
}

# selection metric: same composite run_training uses for best-epoch keeping
SELECT_KEYS = ("seq_f1", "node_auc")


def sample_trial(space: Dict[str, Any], seed: int, trial: int) -> Dict[str, str]:
    """Deterministic sample -> dotted-override strings for load_config."""
    rng = np.random.default_rng(hash((seed, trial)) & 0x7FFFFFFF)
    out: Dict[str, str] = {}
    for key in sorted(space):
        spec = space[key]
        if isinstance(spec, tuple) and len(spec) == 2:
            lo, hi = float(spec[0]), float(spec[1])
            v = math.exp(rng.uniform(math.log(lo), math.log(hi)))
            out[key] = f"{v:.6g}"
        else:
            out[key] = str(spec[int(rng.integers(0, len(spec)))])
    return out


@dataclass
class TrialResult:
    trial: int
    overrides: Dict[str, str]
    metrics: Dict[str, float]
    score: float
    checkpoint_dir: str


@dataclass
class SearchResult:
    trials: List[TrialResult] = field(default_factory=list)

    @property
    def best(self) -> TrialResult:
        return max(self.trials, key=lambda t: t.score)

    def to_json(self) -> str:
        return json.dumps(
            {
                "best": self.best.trial,
                "trials": [
                    {
                        "trial": t.trial,
                        "overrides": t.overrides,
                        "score": round(t.score, 6),
                        "metrics": {k: round(v, 6) for k, v in t.metrics.items()},
                        "checkpoint_dir": t.checkpoint_dir,
                    }
                    for t in self.trials
                ],
            },
            indent=2,
        )


def _score(metrics: Dict[str, float]) -> float:
    return float(sum(metrics.get(k, 0.0) for k in SELECT_KEYS))


def random_search(
    n_trials: int = 8,
    space: Optional[Dict[str, Any]] = None,
    base_overrides: Sequence[str] = (),
    out_dir: str = "hparam_search",
    seed: int = 0,
) -> SearchResult:
    """Run `n_trials` trainings, one per sampled config; returns all trial
    metrics with the best by holdout seq-F1 + node-AUC.  Each trial gets
    `out_dir/trial_NN` as its checkpoint dir; a summary lands at
    `out_dir/search.json` after every trial (crash-resumable evidence)."""
    from .config import load_config
    from .train import run_training

    space = dict(DEFAULT_SPACE if space is None else space)
    os.makedirs(out_dir, exist_ok=True)
    result = SearchResult()
    for trial in range(n_trials):
        ov = sample_trial(space, seed, trial)
        ckpt = os.path.join(out_dir, f"trial_{trial:02d}")
        overrides = (
            list(base_overrides)
            + [f"{k}={v}" for k, v in ov.items()]
            + [f"run.checkpoint_dir={ckpt}"]
        )
        cfg = load_config(None, overrides)
        metrics = run_training(cfg)
        tr = TrialResult(
            trial=trial,
            overrides=ov,
            metrics={k: float(v) for k, v in metrics.items()},
            score=_score(metrics),
            checkpoint_dir=ckpt,
        )
        result.trials.append(tr)
        with open(os.path.join(out_dir, "search.json"), "w") as f:
            f.write(result.to_json())
    return result


def main(argv: Optional[Sequence[str]] = None) -> int:
    import argparse

    ap = argparse.ArgumentParser(description="joint-model hparam search")
    ap.add_argument("--trials", type=int, default=8)
    ap.add_argument("--epochs", type=int, default=3)
    ap.add_argument("--scenarios", type=int, default=8)
    ap.add_argument("--out", default="hparam_search")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--set", action="append", default=[], metavar="K=V",
                    help="extra dotted overrides applied to every trial")
    args = ap.parse_args(argv)
    base = [
        f"optim.epochs={args.epochs}",
        f"data.n_scenarios={args.scenarios}",
    ] + list(args.set)
    res = random_search(
        n_trials=args.trials, base_overrides=base, out_dir=args.out,
        seed=args.seed,
    )
    print(res.to_json())
    b = res.best
    print(f"# best: trial {b.trial} score={b.score:.4f} -> {b.checkpoint_dir}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
