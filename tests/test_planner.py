"""MCTS rollback planner tests (CPU reference)."""
import numpy as np
import pytest

from nerrf_amd.planner.mcts import PlanResult, run_mcts, xorshift32
from nerrf_amd.planner.rewards import (
    A_KILL,
    A_REVERT_BASE,
    A_STOP,
    PlannerParams,
    build_state,
    simulate_plan,
)


def _attack_state(n_hot=20, n_cold=20, hot=0.95, cold=0.03):
    scores = np.concatenate([np.full(n_hot, hot), np.full(n_cold, cold)])
    mb = np.full(n_hot + n_cold, 2.0)
    return build_state(scores, mb, proc_score=0.97, remaining_clean_mb=50.0)


def test_xorshift_known_values():
    # reference values of the classic xorshift32 sequence from seed 1
    s = 1
    seq = []
    for _ in range(3):
        s = xorshift32(s)
        seq.append(s)
    assert seq == [270369, 67634689, 2647435461]


def test_reward_do_nothing_vs_perfect():
    st = _attack_state()
    p = PlannerParams()
    nothing = simulate_plan(st, [A_STOP], p)
    kill_all = simulate_plan(st, [A_KILL] + [A_REVERT_BASE + g for g in range(st.n_groups)] + [A_STOP], p)
    assert kill_all > nothing
    # killing the process alone already removes the ongoing-damage term
    assert simulate_plan(st, [A_KILL, A_STOP], p) > nothing


def test_reward_duplicate_revert_is_noop():
    st = _attack_state()
    p = PlannerParams()
    r1 = simulate_plan(st, [A_KILL, A_REVERT_BASE, A_STOP], p)
    r2 = simulate_plan(st, [A_KILL, A_REVERT_BASE, A_REVERT_BASE, A_STOP], p)
    assert r1 == r2


def test_mcts_finds_kill_and_reverts_hot_groups():
    st = _attack_state()
    res = run_mcts(st, n_sims=1024, seed=1)
    assert isinstance(res, PlanResult)
    assert res.simulations == 1024
    assert A_KILL in res.plan  # stopping the attacker dominates
    # the plan beats doing nothing by a wide margin
    p = PlannerParams()
    assert res.root_value > simulate_plan(st, [A_STOP], p) + 10
    # descriptions render
    assert "kill_process" in res.describe(st.n_groups)


def test_mcts_clean_system_plans_nothing():
    scores = np.full(30, 0.01)
    mb = np.full(30, 1.0)
    st = build_state(scores, mb, proc_score=0.01, remaining_clean_mb=100.0)
    res = run_mcts(st, n_sims=512, seed=3)
    # on a clean system the planner must not take destructive actions
    assert all(a == A_KILL or a >= A_REVERT_BASE for a in res.plan)
    assert len([a for a in res.plan if a >= A_REVERT_BASE]) == 0


def test_mcts_deterministic_given_seed():
    st = _attack_state()
    r1 = run_mcts(st, n_sims=256, seed=7)
    r2 = run_mcts(st, n_sims=256, seed=7)
    assert r1.plan == r2.plan
    assert r1.ranked_actions == r2.ranked_actions


def test_build_state_buckets():
    scores = np.linspace(1, 0, 64)
    mb = np.ones(64)
    st = build_state(scores, mb, 0.5, 10.0, n_groups=8)
    assert st.n_groups == 8
    assert st.group_mb.sum() == pytest.approx(64.0)
    # score-ordered buckets: group 0 hottest
    assert st.group_score[0] > st.group_score[-1]


def test_reward_kill_stops_ongoing_encryption():
    """While the attack process lives, remaining clean data keeps getting
    encrypted at attack_rate; an early kill caps the loss."""
    import numpy as np

    from nerrf_amd.planner.rewards import (
        A_KILL, A_STOP, PlannerParams, PlannerState, simulate_plan,
    )

    st = PlannerState(
        group_score=np.array([0.9]), group_mb=np.array([10.0]),
        group_files=np.array([5.0]), proc_score=0.95,
        remaining_clean_mb=100.0,
    )
    p = PlannerParams(n_groups=1)
    r_idle = simulate_plan(st, [A_STOP], p)
    r_kill = simulate_plan(st, [A_KILL, A_STOP], p)
    assert r_kill > r_idle  # killing beats watching the encryption continue


def test_empty_state_no_destructive_actions():
    """Zeroed detector state: the plan must contain no reverts.  (The
    reward model currently charges horizon downtime while the process is
    alive regardless of proc_score, so a bare KILL can appear — accepted
    contract, same as test_mcts_clean_system_plans_nothing; scaling that
    charge by proc_score is a round-2 change that must land in
    rewards.py and mcts.hip together to keep bit-parity.)"""
    import numpy as np

    from nerrf_amd.planner.mcts import run_mcts
    from nerrf_amd.planner.rewards import A_REVERT_BASE, PlannerState

    st = PlannerState(
        group_score=np.zeros(4), group_mb=np.zeros(4),
        group_files=np.zeros(4), proc_score=0.0, remaining_clean_mb=0.0,
    )
    res = run_mcts(st, n_sims=128)
    assert not [a for a in res.plan if a >= A_REVERT_BASE]


def test_restore_action_semantics():
    """RESTORE recovers every group at a fixed downtime + backup-staleness
    loss (reference threat-model.mdx:206-222 candidate 3)."""
    from nerrf_amd.planner.rewards import (
        A_KILL, A_RESTORE, A_STOP, PlannerParams, PlannerState, simulate_plan,
    )
    import numpy as np

    p = PlannerParams(n_groups=4)
    st = PlannerState(
        group_score=np.array([0.95, 0.9, 0.85, 0.8]),
        group_mb=np.array([40.0, 40.0, 40.0, 40.0]),
        group_files=np.array([20.0, 20.0, 20.0, 20.0]),
        proc_score=0.9,
        remaining_clean_mb=500.0,
    )
    r_restore = simulate_plan(st, [A_KILL, A_RESTORE, A_STOP], p)
    r_idle = simulate_plan(st, [A_STOP], p)
    # heavy widespread encryption: restore beats doing nothing
    assert r_restore > r_idle
    # restoring twice is a no-op
    assert simulate_plan(st, [A_KILL, A_RESTORE, A_RESTORE, A_STOP], p) == r_restore
    # with per-file reverts cheap and few files, targeted reverts beat the
    # blunt restore (the reference's "highest reward/cost ratio" example)
    from nerrf_amd.planner.rewards import A_REVERT_BASE

    small = PlannerState(
        group_score=np.array([0.95, 0.0, 0.0, 0.0]),
        group_mb=np.array([5.0, 1.0, 1.0, 1.0]),
        group_files=np.array([3.0, 1.0, 1.0, 1.0]),
        proc_score=0.9,
        remaining_clean_mb=500.0,
    )
    r_rev = simulate_plan(small, [A_KILL, A_REVERT_BASE + 0, A_STOP], p)
    r_res = simulate_plan(small, [A_KILL, A_RESTORE, A_STOP], p)
    assert r_rev > r_res


def test_idle_on_clean_state_is_cheap():
    """NEXT.md gap: a zero-belief process must not be charged the full
    horizon downtime — the degradation charge scales with proc_score."""
    from nerrf_amd.planner.rewards import (
        A_KILL, A_STOP, PlannerParams, PlannerState, simulate_plan,
    )
    import numpy as np

    p = PlannerParams(n_groups=2)
    clean = PlannerState(
        group_score=np.zeros(2), group_mb=np.ones(2), group_files=np.ones(2),
        proc_score=0.0, remaining_clean_mb=100.0,
    )
    r_idle = simulate_plan(clean, [A_STOP], p)
    r_kill = simulate_plan(clean, [A_KILL, A_STOP], p)
    # on a fully clean state, doing nothing now beats a spurious KILL
    assert r_idle > r_kill
    assert r_idle == 0.0


def test_mcts_prefers_restore_when_everything_encrypted():
    from nerrf_amd.planner.mcts import run_mcts
    from nerrf_amd.planner.rewards import A_RESTORE, PlannerParams, PlannerState
    import numpy as np

    p = PlannerParams(n_groups=8, revert_time_s=2.0)  # slow per-file reverts
    st = PlannerState(
        group_score=np.full(8, 0.95),
        group_mb=np.full(8, 50.0),
        group_files=np.full(8, 40.0),  # 40 files/group x 2 s >> restore_time
        proc_score=0.95,
        remaining_clean_mb=100.0,
    )
    res = run_mcts(st, p, n_sims=2048, seed=3)
    assert A_RESTORE in res.plan
    assert "restore_from_backup" in res.describe(st.n_groups)


def test_counterfactual_features_clear_only_targeted_nodes():
    import numpy as np

    from nerrf_amd.planner.model_eval import (
        KILL_CLEAR_CHANNELS, REVERT_CLEAR_CHANNELS, counterfactual_features,
    )
    from nerrf_amd.planner.rewards import A_KILL, A_REVERT_BASE, A_STOP

    x = np.ones((10, 32), dtype=np.float32)
    groups = [np.array([0, 1]), np.array([2, 3])]
    procs = np.array([8, 9])
    cf = counterfactual_features(x, [A_KILL, A_REVERT_BASE + 1, A_STOP], groups, procs)
    # group 1's nodes cleared on revert channels; group 0 untouched
    assert (cf[2, list(REVERT_CLEAR_CHANNELS)] == 0).all()
    assert (cf[0, list(REVERT_CLEAR_CHANNELS)] == 1).all()
    # proc nodes cleared on kill channels
    assert (cf[8, list(KILL_CLEAR_CHANNELS)] == 0).all()
    # untouched nodes identical
    assert (cf[4:8] == 1).all()
    assert (x == 1).all()  # input not mutated


def test_model_refinement_prefers_reverting_hot_group():
    """With the trained checkpoint, a plan that reverts the actually-hot
    files leaves lower residual model risk than one reverting cold files."""
    from pathlib import Path

    import numpy as np
    import pytest

    ckpt = Path("checkpoints/pretrained")
    if not (ckpt / "checkpoint.json").exists():
        pytest.skip("no vendored checkpoint")
    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.serve.engine import StreamingEngine, load_model_from_checkpoint
    from nerrf_amd.planner.rewards import A_KILL, A_REVERT_BASE

    model = load_model_from_checkpoint(str(ckpt))
    arr, _ = generate(SynthConfig(seed=6, duration_s=40, benign_rate_hz=40,
                                  n_victim_files=8))
    eng = StreamingEngine(model=model, device="cpu", window_s=1e9)
    eng.ingest_events(arr)
    det = eng.score_window()
    assert det.alarm and det.refine_ctx is not None

    from nerrf_amd.planner.model_eval import refine_plans_with_model
    from nerrf_amd.planner.rewards import PlannerParams, build_state

    paths = list(det.file_scores.keys())
    scores = np.array([det.file_scores[p] for p in paths])
    mb = np.array([max(det.file_mb.get(p, 0.01), 0.01) for p in paths])
    params = PlannerParams()
    state = build_state(scores, mb, proc_score=0.9, remaining_clean_mb=10.0,
                        n_groups=params.n_groups)
    order = np.argsort(-scores, kind="stable")
    split = np.array_split(order, params.n_groups)
    p2n = det.refine_ctx["path_to_node"]
    group_nodes = [
        np.array([p2n[paths[i]] for i in ids if paths[i] in p2n], dtype=np.int64)
        for ids in split
    ]
    hot = [A_KILL, A_REVERT_BASE + 0]        # group 0 = highest scores
    cold = [A_KILL, A_REVERT_BASE + params.n_groups - 1]
    ranked = refine_plans_with_model(
        model, det.refine_ctx["x"], det.refine_ctx["nbr_idx"],
        det.refine_ctx["nbr_w"], group_nodes, det.refine_ctx["proc_nodes"],
        [hot, cold], state, params, device="cpu",
    )
    res = {tuple(p): resid for p, _c, _cl, resid in ranked}
    assert res[tuple(hot)] < res[tuple(cold)]


def test_engine_plan_with_model_refinement_end_to_end():
    import numpy as np

    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.models.graphsage import SageConfig
    from nerrf_amd.models.joint import JointConfig, NerrfJointModel
    from nerrf_amd.models.lstm import LSTMConfig
    from nerrf_amd.serve.engine import StreamingEngine

    model = NerrfJointModel(JointConfig(sage=SageConfig(layers=3, hidden=32),
                                        lstm=LSTMConfig(hidden=32)))
    arr, _ = generate(SynthConfig(seed=2, duration_s=40, benign_rate_hz=40,
                                  n_victim_files=8))
    eng = StreamingEngine(model=model, device="cpu", window_s=1e9)
    eng.ingest_events(arr)
    det = eng.score_window()
    assert det.alarm
    plan = eng.plan(det, n_sims=256)
    assert plan.plan  # non-empty remediation
    # refinement off still works
    plan2 = eng.plan(det, n_sims=256, model_refine=False)
    assert plan2.simulations >= 256
