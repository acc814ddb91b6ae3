"""Parity on the reference's RECORDED benchmark artifacts (VERDICT r1 gap 1).

These tests load the actual traces the upstream tracker captured in minikube
(`/root/reference/benchmarks/m{0,1}/results/`), replay them through the
engine, and assert the recorded attack is detected inside the recorded
ground-truth window — the only validation data the reference ships.
Skipped where the reference checkout is absent (e.g. on a GPU box snapshot).
"""
from pathlib import Path

import pytest

REF = Path("/root/reference/benchmarks")

needs_reference = pytest.mark.skipif(
    not REF.exists(), reason="reference benchmark artifacts not present"
)


@needs_reference
@pytest.mark.parametrize("stem", ["m0", "m1"])
def test_recorded_trace_loads_with_real_schema(stem):
    from nerrf_amd.data.trace import SYSCALL_IDS, load_jsonl

    ev = load_jsonl(REF / stem / "results" / f"{stem}_trace.jsonl")
    assert len(ev) == {"m0": 88, "m1": 149}[stem]
    # ISO timestamps parsed to real epochs (2025-08-30 ~= 1.7565e9)
    assert 1.75e9 < float(ev.ts.min()) < 1.77e9
    # encrypt-complete events reconstructed as rename pairs .dat -> .lockbit3
    ren = ev.syscall == SYSCALL_IDS["rename"]
    n_files = {"m0": 25, "m1": 45}[stem]
    assert int(ren.sum()) == n_files
    news = {ev.paths.lookup(int(i)) for i in ev.new_path_id[ren]}
    assert all(p.endswith(".lockbit3") for p in news)
    olds = {ev.paths.lookup(int(i)) for i in ev.path_id[ren]}
    assert all(p.endswith(".dat") for p in olds)


@needs_reference
@pytest.mark.parametrize("stem", ["m0", "m1"])
def test_detects_recorded_attack_inside_ground_truth_window(stem):
    from nerrf_amd.harness.reference_parity import detect_on_recorded_run

    rep = detect_on_recorded_run(REF / stem / "results")
    assert rep["first_alarm_ts"] is not None
    assert rep["alarm_within_window"]
    # every encrypted file the run recorded is identified across the replay
    assert rep["encrypted_file_recall"] == 1.0
    # the alarm fires DURING encryption, not after the fact
    assert rep["data_loss_mb_at_alarm"] < rep["total_encrypted_mb"]
    assert rep["meets_data_loss_target"]  # <= 128 MB, reference README.md:23-27
    assert rep["latency_from_first_encrypt_s"] < 30.0


@needs_reference
def test_replay_recovery_emits_reference_schema(tmp_path):
    from nerrf_amd.harness.reference_parity import load_recorded_run, replay_recovery

    out = tmp_path / "recovery.json"
    rep = replay_recovery(REF / "m1" / "results", tmp_path, out_json=out)
    ref = load_recorded_run(REF / "m1" / "results")["recovery"]
    # exact reference key layout is a subset of ours
    for k in ref:
        assert k in rep, k
    assert rep["recovered_files"] == ref["recovered_files"] == 45
    # ours decrypts + sha256-gates (strictly more work than the recorded
    # rename-back), and must still be in the same performance class
    assert rep["decrypted"] and rep["sandbox_validated"] and rep["sha256_ok"]
    assert rep["recovery_duration_ms"] < 5000.0
    assert out.exists()


@needs_reference
def test_detection_with_vendored_checkpoint(tmp_path):
    """The trained checkpoint must not regress indicator-driven detection
    on the recorded data (model scores can only raise the alarm earlier)."""
    ckpt = Path("checkpoints/pretrained")
    if not (ckpt / "checkpoint.json").exists():
        pytest.skip("no vendored checkpoint")
    from nerrf_amd.harness.reference_parity import detect_on_recorded_run
    from nerrf_amd.serve.engine import load_model_from_checkpoint

    model = load_model_from_checkpoint(str(ckpt))
    rep = detect_on_recorded_run(REF / "m1" / "results", model=model)
    assert rep["alarm_within_window"]
    assert rep["encrypted_file_recall"] == 1.0


@needs_reference
def test_detection_with_proc_identity_stack(monkeypatch):
    """The proc-identity stack (channel active + its checkpoint) also
    detects the recorded m0/m1 attacks inside their ground-truth windows —
    the future default-swap is CPU-validated against the only real data."""
    ckpt = Path("checkpoints/pretrained_procid")
    if not (ckpt / "checkpoint.json").exists():
        pytest.skip("no proc-identity checkpoint")
    from nerrf_amd.harness.reference_parity import detect_on_recorded_run
    from nerrf_amd.serve.engine import load_model_from_checkpoint

    monkeypatch.setenv("NERRF_PROC_IDENTITY", "1")
    model = load_model_from_checkpoint(str(ckpt))
    for stem in ("m0", "m1"):
        rep = detect_on_recorded_run(REF / stem / "results", model=model)
        assert rep["alarm_within_window"], stem
        assert rep["encrypted_file_recall"] == 1.0, stem
