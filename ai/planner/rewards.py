"""Shim: the reference layout's ai/planner/rewards.py -> nerrf_amd."""
from nerrf_amd.planner.rewards import (  # noqa: F401
    A_KILL,
    A_RESTORE,
    A_REVERT_BASE,
    A_STOP,
    PlannerParams,
    PlannerState,
    build_state,
    simulate_plan,
)
