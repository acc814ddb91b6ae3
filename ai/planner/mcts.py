"""Shim: the reference layout's ai/planner/mcts.py -> nerrf_amd."""
from nerrf_amd.planner.mcts import PlanResult, run_mcts, run_mcts_gpu  # noqa: F401
