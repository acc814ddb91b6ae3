"""Persisted detection registry — the `nerrf undo --id <attack>` contract.

The reference's CLI addresses a previously identified attack by id
(reference README.md:82 / ROADMAP.md:86: ``nerrf undo --id <attack>``);
this registry is where those ids come from: the monitor records every
alarm (detection summary + ranked plan + target directory) as one JSON
file under a state directory, `nerrf status` lists them, and
`nerrf undo --id` replays the recorded response without re-scoring.

Records are small (scores capped at the top 200 files) and atomic
(write-temp + rename), so a crashed monitor never leaves a torn record.
"""
from __future__ import annotations

import json
import os
import time
from pathlib import Path
from typing import Dict, List, Optional

DEFAULT_STATE_DIR = ".nerrf/detections"
_MAX_FILE_SCORES = 200


class DetectionRegistry:
    def __init__(self, state_dir: str = DEFAULT_STATE_DIR) -> None:
        self.state_dir = Path(state_dir)

    def record(self, det, plan=None, target_dir: str = "",
               n_groups: int = 0) -> str:
        """Persist one alarm; returns its attack id (atk-<utc>-<seq>)."""
        self.state_dir.mkdir(parents=True, exist_ok=True)
        stamp = time.strftime("%Y%m%dT%H%M%S", time.gmtime())
        seq = 0
        while True:
            attack_id = f"atk-{stamp}-{seq:03d}"
            path = self.state_dir / f"{attack_id}.json"
            if not path.exists():
                break
            seq += 1
        top = sorted(det.file_scores.items(), key=lambda kv: -kv[1])
        rec = {
            "attack_id": attack_id,
            "t_detect": det.t_detect,
            "alarm": bool(det.alarm),
            "window_events": int(det.window_events),
            "indicators": {k: float(v) for k, v in det.indicators.items()},
            "encrypted_paths": list(det.encrypted_paths)[:_MAX_FILE_SCORES],
            "file_scores": {p: round(float(s), 4) for p, s in top[:_MAX_FILE_SCORES]},
            "target_dir": target_dir,
        }
        if plan is not None:
            rec["plan"] = {
                "actions": plan.describe(n_groups) if n_groups else [],
                "raw_actions": [int(a) for a in plan.plan],
                "root_value": float(plan.root_value),
                "simulations": int(plan.simulations),
            }
        tmp = path.with_suffix(".tmp")
        tmp.write_text(json.dumps(rec, indent=2))
        os.replace(tmp, path)
        return attack_id

    def load(self, attack_id: str) -> Optional[Dict]:
        p = self.state_dir / f"{attack_id}.json"
        if not p.exists():
            return None
        return json.loads(p.read_text())

    def list(self) -> List[Dict]:
        """Newest-first summaries of every recorded detection."""
        out = []
        if not self.state_dir.is_dir():
            return out
        for p in sorted(self.state_dir.glob("atk-*.json"), reverse=True):
            try:
                r = json.loads(p.read_text())
            except ValueError:
                continue
            out.append({
                "attack_id": r.get("attack_id", p.stem),
                "t_detect": r.get("t_detect"),
                "alarm": r.get("alarm"),
                "target_dir": r.get("target_dir", ""),
                "n_flagged": len(r.get("file_scores", {})),
            })
        return out
