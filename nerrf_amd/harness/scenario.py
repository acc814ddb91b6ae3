"""End-to-end benchmark scenario: seed -> attack -> detect -> plan -> recover.

The in-process equivalent of the reference's minikube benchmark loop
(behavior of benchmarks/m1/scripts/m1_minikube_bootstrap.sh: deploy victim,
run simulator, collect trace + ground truth, time the rollback, emit
metadata/recovery-result artifacts) — producing the same artifact schema
(*_trace.jsonl, *_ground_truth.csv, metadata.json, *_recovery_results.json)
without k8s.
"""
from __future__ import annotations

import csv
import json
import time
from pathlib import Path
from typing import Dict, Optional

from ..data.trace import load_trace
from .attack_sim import run_attack, seed_files, verify_manifest


def run_scenario(
    work_dir: str | Path,
    n_files: int = 16,
    file_kb: int = 32,
    device: str = "cpu",
    n_sims: int = 512,
    artifacts_dir: Optional[str | Path] = None,
    model=None,
) -> Dict:
    """Returns a metrics report; writes benchmark artifacts when requested."""
    from ..serve.engine import StreamingEngine

    work = Path(work_dir)
    uploads = work / "uploads"
    art = Path(artifacts_dir) if artifacts_dir else work / "results"
    art.mkdir(parents=True, exist_ok=True)

    # ---- phase 1: seed victim files ---------------------------------------
    manifest = seed_files(uploads, n_files=n_files, file_kb=file_kb, seed=1)

    # ---- phase 2: attack ---------------------------------------------------
    trace_path = art / "trace.jsonl"
    report = run_attack(uploads, trace_path=trace_path)
    with open(art / "ground_truth.csv", "w", newline="") as fh:
        w = csv.writer(fh)
        w.writerow(["attack_start", "attack_end", "target_dir", "files", "bytes"])
        w.writerow([report.t_start, report.t_end, str(uploads), len(report.files_attacked), report.bytes_attacked])

    # ---- phase 3: detect ---------------------------------------------------
    t_ingest0 = time.perf_counter()
    engine = StreamingEngine(device=device, model=model)
    trace = load_trace(trace_path)
    engine.ingest_events(trace)
    det = engine.score_window()
    t_detect = time.perf_counter() - t_ingest0

    # ---- phase 4: plan -----------------------------------------------------
    t_plan0 = time.perf_counter()
    plan = engine.plan(det, n_sims=n_sims)
    t_plan = time.perf_counter() - t_plan0

    # ---- phase 5: recover (sandbox gate + live restore) --------------------
    result = engine.respond(det, plan, str(uploads), manifest=manifest)
    checks = verify_manifest(manifest)
    recovered_ok = all(checks.values())
    mttr_s = t_detect + t_plan + result.duration_ms / 1000.0

    out = {
        "events": len(trace),
        "files_attacked": len(report.files_attacked),
        "mb_attacked": report.bytes_attacked / 1e6,
        "alarm": det.alarm,
        "indicators": det.indicators,
        "plan": plan.describe(engine.planner_params.n_groups),
        "plan_value": plan.root_value,
        "detect_s": t_detect,
        "plan_s": t_plan,
        "recovery_ms": result.duration_ms,
        "files_per_sec": result.files_per_sec,
        "mb_per_sec": result.mb_per_sec,
        "sandbox_validated": result.sandbox_validated,
        "recovered_ok": recovered_ok,
        "mttr_s": mttr_s,
        "data_loss_mb": 0.0 if recovered_ok else report.bytes_attacked / 1e6,
    }
    with open(art / "metadata.json", "w") as fh:
        json.dump({"scale": {"files": n_files, "file_kb": file_kb}, "report": out}, fh, indent=2)
    with open(art / "recovery_results.json", "w") as fh:
        json.dump(result.as_dict(), fh, indent=2)
    return out
