"""Detection ablation: rule indicators alone vs model heads alone vs the
combined alarm, per scenario family (window-level alarm decision).

Quantifies the defence-in-depth design (docs/threat-model.md): the alarm is
max(indicator_score, min(node_max, seq_max)), so a rule bypass and a cold
model must BOTH happen to silence detection.
"""
import sys, os, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from nerrf_amd.data.synth import SynthConfig, generate
from nerrf_amd.serve.engine import StreamingEngine, load_model_from_checkpoint

FAMILIES = ["lockbit", "supply_chain", "supply_chain_net"]
NEGATIVES = ["benign_rotate", "benign_backup", "benign_build"]
THRESH = 0.7


def score(kind, model, seed):
    eng = StreamingEngine(model=model, device="cpu")
    eng.store.window_s = 1e9
    arr, _ = generate(SynthConfig(kind=kind, duration_s=40.0, benign_rate_hz=60.0, seed=seed))
    eng.ingest_events(arr)
    det = eng.score_window()
    ind = det.indicators
    ind_score = min(1.0, 0.6 * float(ind["suspicious_ext_count"] > 0)
                    + 0.3 * float(ind["ransom_note"]) + 0.4 * float(ind["write_to_rename"] > 0.1)
                    + 0.7 * float(ind.get("exfil_dest_count", 0) > 0))
    model_scores = [s for s in det.file_scores.values()]
    model_max = max(model_scores, default=0.0)
    return {
        "indicators_alarm": ind_score >= THRESH,
        "model_alarm": model_max >= THRESH,  # model heads only (no rules)
        "combined_alarm": det.alarm,
    }


def main():
    model = load_model_from_checkpoint(
        os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                     "checkpoints", "pretrained"))
    out = {}
    for kind in FAMILIES + NEGATIVES:
        rows = [score(kind, model, seed) for seed in (11, 23, 37)]
        out[kind] = {k: sum(r[k] for r in rows) / len(rows) for k in rows[0]}
    print(json.dumps(out, indent=2))


if __name__ == "__main__":
    main()
