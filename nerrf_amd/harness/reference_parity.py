"""Parity evaluation on the reference's RECORDED benchmark artifacts.

The reference ships one real dataset per milestone: the traces its tracker
actually captured in minikube, with ground truth and timed recovery
(`/root/reference/benchmarks/m0/results/` — 88 events, and
`/root/reference/benchmarks/m1/results/` — 149 events;
`m1_ground_truth.csv` gives the attack window, `m1_recovery_results.json:6-11`
the published recovery numbers).  This module closes the loop on that data:

  1. `detect_on_recorded_run` replays the recorded trace through the
     streaming engine tick by tick and reports WHEN the engine first alarms
     relative to the recorded ground-truth window, plus how many of the
     recorded encrypted files it identifies and the MB already encrypted at
     alarm time (the reference's DataLoss metric, target <= 128 MB,
     README.md:23-27).
  2. `replay_recovery` reconstructs the recorded victim file set (same
     names and sizes, from `file_list.txt`), re-runs the attack end state
     with our simulator and times OUR rollback on it, emitting a JSON in
     the exact key layout of the reference's `*_recovery_results.json` so
     the two are directly comparable.

Nothing here imports reference code; only its published data artifacts are
read.
"""
from __future__ import annotations

import csv
import json
import re
import time
from pathlib import Path
from typing import Dict, List, Optional, Tuple

import numpy as np

from ..data.trace import EventArray, load_jsonl

_LOCKBIT_EXT = re.compile(r"\.(lockbit\w*)$", re.IGNORECASE)


def load_ground_truth(csv_path: str | Path) -> dict:
    """One-row CSV: start_ts,end_ts,...,attack_family,target_path,..."""
    with open(csv_path, newline="") as fh:
        row = next(iter(csv.DictReader(fh)))
    return {
        "start_ts": float(row["start_ts"]),
        "end_ts": float(row["end_ts"]),
        "attack_family": row.get("attack_family", ""),
        "target_path": row.get("target_path", ""),
        "duration_sec": float(row.get("duration_sec", 0) or 0),
        "scale": row.get("scale", row.get("platform", "")),
    }


def load_file_list(path: str | Path) -> List[Tuple[str, int]]:
    """Parse the recorded `ls -l`-style file inventory -> [(path, bytes)].

    Lines look like:
    `-rw-r--r-- 1 root root 2397226 Aug 30 14:08 /app/uploads/x.lockbit3`.
    """
    out: List[Tuple[str, int]] = []
    for line in Path(path).read_text().splitlines():
        parts = line.split()
        if len(parts) < 9 or not parts[0].startswith("-"):
            continue
        try:
            size = int(parts[4])
        except ValueError:
            continue
        out.append((parts[-1], size))
    return out


def load_recorded_run(results_dir: str | Path) -> dict:
    """Load every artifact of one recorded run (m0 or m1 layout)."""
    d = Path(results_dir)
    stem = "m1" if (d / "m1_trace.jsonl").exists() else "m0"
    run = {
        "stem": stem,
        "events": load_jsonl(d / f"{stem}_trace.jsonl"),
        "ground_truth": load_ground_truth(d / f"{stem}_ground_truth.csv"),
        "metadata": json.loads((d / "metadata.json").read_text()),
    }
    rec = d / f"{stem}_recovery_results.json"
    if rec.exists():
        # the recorded file uses bare-leading-dot floats (".044"), which
        # strict json rejects; patch to valid literals before parsing
        txt = re.sub(r":\s*\.(\d)", r": 0.\1", rec.read_text())
        txt = re.sub(r':\s*"\.(\d+)"', r': "0.\1"', txt)
        run["recovery"] = json.loads(txt)
    fl = d / "file_list.txt"
    if fl.exists():
        run["file_list"] = load_file_list(fl)
    return run


def detect_on_recorded_run(
    results_dir: str | Path,
    model=None,
    window_s: float = 30.0,
    tick_s: float = 5.0,
    alarm_threshold: float = 0.7,
) -> dict:
    """Replay the recorded trace through the engine; report detection timing.

    The engine sees the events exactly as recorded (ingested in bulk; the
    tick loop trims the 30 s window at each step, like the production
    monitor would have at the time).  Returns a report with the first-alarm
    tick, its latency inside the recorded ground-truth window, recall over
    the recorded encrypted file set, and the DataLoss proxy (MB whose
    encryption completed before the alarm).
    """
    from ..serve.engine import StreamingEngine

    run = load_recorded_run(results_dir)
    ev: EventArray = run["events"]
    gt = run["ground_truth"]
    eng = StreamingEngine(
        model=model, device="cpu", window_s=window_s, alarm_threshold=alarm_threshold
    )

    t_lo, t_hi = float(ev.ts.min()), float(ev.ts.max())
    first_alarm: Optional[float] = None
    alarm_indicators: Dict[str, float] = {}
    detected_enc: set = set()
    fed = 0  # replay faithfully: only events recorded up to the tick are fed
    for t in np.arange(t_lo + tick_s, t_hi + tick_s, tick_s):
        hi = int(np.searchsorted(ev.ts, float(t), side="right"))
        if hi > fed:
            eng.ingest_events(ev.slice(fed, hi))
            fed = hi
        det = eng.score_window(now=float(t))
        if det.alarm:
            if first_alarm is None:
                first_alarm = float(t)
                alarm_indicators = dict(det.indicators)
            detected_enc.update(det.encrypted_paths)

    # recorded encrypted set: every *.lockbit* the run left on disk
    recorded_enc = {
        p for (p, _sz) in run.get("file_list", []) if _LOCKBIT_EXT.search(p)
    }
    recall = (
        len(detected_enc & recorded_enc) / len(recorded_enc) if recorded_enc else None
    )

    # DataLoss proxy: bytes whose encryption had completed by the alarm
    # (rename events carry the reconstructed .dat -> .lockbit3 pair)
    from ..data.trace import SYSCALL_IDS

    ren = (ev.syscall == SYSCALL_IDS["rename"]) & (ev.new_path_id >= 0)
    loss_mb = None
    if first_alarm is not None:
        loss_mb = float(ev.nbytes[ren & (ev.ts <= first_alarm)].sum()) / 1e6
    total_enc_mb = float(ev.nbytes[ren].sum()) / 1e6
    t_first_encrypt = float(ev.ts[ren].min()) if ren.any() else None

    return {
        "results_dir": str(results_dir),
        "stem": run["stem"],
        "events": len(ev),
        "ground_truth_window": [gt["start_ts"], gt["end_ts"]],
        "first_alarm_ts": first_alarm,
        "alarm_within_window": (
            first_alarm is not None and gt["start_ts"] <= first_alarm <= gt["end_ts"]
        ),
        "detection_latency_s": (
            None if first_alarm is None else first_alarm - gt["start_ts"]
        ),
        # ground truth starts before trace capture did; latency from the
        # first recorded encryption is the actionable figure
        "latency_from_first_encrypt_s": (
            None
            if first_alarm is None or t_first_encrypt is None
            else first_alarm - t_first_encrypt
        ),
        "alarm_indicators": alarm_indicators,
        "encrypted_files_recorded": len(recorded_enc),
        "encrypted_files_detected": len(detected_enc & recorded_enc),
        "encrypted_file_recall": recall,
        "data_loss_mb_at_alarm": loss_mb,
        "total_encrypted_mb": total_enc_mb,
        "data_loss_target_mb": 128.0,  # reference README.md:23-27
        "meets_data_loss_target": loss_mb is not None and loss_mb <= 128.0,
    }


def seed_recorded_files(
    target_dir: str | Path, file_list: List[Tuple[str, int]], seed: int = 0
) -> Dict[str, str]:
    """Recreate the recorded victim set: same basenames (encrypted ext
    stripped back to .dat) and byte sizes, random content; returns the
    {path: sha256} manifest."""
    import hashlib

    target = Path(target_dir)
    target.mkdir(parents=True, exist_ok=True)
    rng = np.random.default_rng(seed)
    manifest: Dict[str, str] = {}
    for path, size in file_list:
        name = Path(path).name
        m = _LOCKBIT_EXT.search(name)
        if not m:
            continue  # ransom note / stray entries
        orig = target / (name[: m.start()] + ".dat")
        data = rng.integers(0, 256, size=size, dtype=np.uint8).tobytes()
        orig.write_bytes(data)
        manifest[str(orig)] = hashlib.sha256(data).hexdigest()
    return manifest


def replay_recovery(
    results_dir: str | Path,
    workdir: str | Path,
    out_json: Optional[str | Path] = None,
) -> dict:
    """Reconstruct the recorded victim set, attack it, time OUR rollback.

    Emits the reference's recovery-results schema
    (`m1_recovery_results.json` keys) so the published numbers
    (m1: 44 ms / 45 files / 1022.72 files/s / 2500 MB/s) are directly
    comparable — with the difference, noted in the payload, that our
    rollback DECRYPTS (XOR inverse + sha256 sandbox gate) rather than just
    renaming, i.e. it does strictly more work per file.
    """
    from ..harness.attack_sim import run_attack
    from ..serve.rollback import execute_rollback

    run = load_recorded_run(results_dir)
    file_list = run.get("file_list", [])
    victim = Path(workdir) / "uploads"
    manifest = seed_recorded_files(victim, file_list)
    run_attack(victim)
    res = execute_rollback(victim, manifest=manifest, decrypt=True, validate_in_sandbox=True)

    total_mb = sum(sz for _p, sz in file_list if _LOCKBIT_EXT.search(_p)) / 1e6
    payload = {
        "timestamp": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
        "platform": "nerrf_amd-replay",
        "scale": run["ground_truth"].get("scale", ""),
        "recovered_files": res.files_restored,
        "recovery_duration_ms": round(res.duration_ms, 3),
        "recovery_duration_sec": round(res.duration_ms / 1000.0, 6),
        "avg_recovery_per_file_ms": round(
            res.duration_ms / max(res.files_restored, 1), 3
        ),
        "recovery_rate_fps": round(res.files_per_sec, 2),
        "total_size_mb": f"{total_mb:.0f}",
        "throughput_mbps": round(res.mb_per_sec, 2),
        # extras the reference schema has no slot for
        "decrypted": True,
        "sandbox_validated": res.sandbox_validated,
        "sha256_ok": res.sha256_ok,
        "reference_recovery": run.get("recovery"),
    }
    if out_json is not None:
        Path(out_json).write_text(json.dumps(payload, indent=2))
    return payload


def run_reference_parity(
    reference_benchmarks: str | Path,
    workdir: str | Path,
    model=None,
    out_json: Optional[str | Path] = None,
) -> dict:
    """Full parity report over every recorded run under
    `<reference>/benchmarks/*/results`."""
    report: dict = {"runs": {}}
    for results in sorted(Path(reference_benchmarks).glob("m*/results")):
        stem = results.parent.name
        wd = Path(workdir) / stem
        wd.mkdir(parents=True, exist_ok=True)
        report["runs"][stem] = {
            "detection": detect_on_recorded_run(results, model=model),
            "recovery": replay_recovery(results, wd),
        }
    if out_json is not None:
        Path(out_json).write_text(json.dumps(report, indent=2))
    return report
