"""Labelled synthetic trace dataset exporter (parquet + manifest).

The reference README maps a `datasets/` directory of "100 h labelled cloud
traces" that the repository does not ship; this tool manufactures the
equivalent from the deterministic scenario generators: N scenarios across
the attack/benign families, exported as one parquet (or CSV fallback) of
per-event rows with ground-truth labels, plus a JSON manifest of per-
scenario metadata (family, seed, attack window).

    python tools/export_dataset.py --out datasets/synthetic \
        --scenarios 24 --duration 120

Deterministic by (seed, scenario index): re-running reproduces the bytes.
"""
from __future__ import annotations

import argparse
import json
import os
import sys

import numpy as np

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

from nerrf_amd.data.labels import event_labels  # noqa: E402
from nerrf_amd.data.synth import SynthConfig, generate  # noqa: E402

FAMILIES = ("lockbit", "supply_chain", "supply_chain_net",
            "benign_rotate", "benign_backup", "benign_build")


def export(out_dir: str, n_scenarios: int = 12, duration_s: float = 120.0,
           benign_rate_hz: float = 400.0, base_seed: int = 0) -> dict:
    os.makedirs(out_dir, exist_ok=True)
    cols = {k: [] for k in ("scenario", "kind", "ts", "pid", "syscall",
                            "path", "new_path", "nbytes", "comm", "label")}
    manifest = []
    from nerrf_amd.data.trace import SYSCALL_IDS

    id_to_sys = {v: k for k, v in SYSCALL_IDS.items()}
    for i in range(n_scenarios):
        kind = FAMILIES[i % len(FAMILIES)]
        attack = not kind.startswith("benign")
        cfg = SynthConfig(duration_s=duration_s, benign_rate_hz=benign_rate_hz,
                          seed=base_seed + 101 * i, kind=kind, attack=attack)
        arr, win = generate(cfg)
        y = event_labels(arr, win) if win is not None else np.zeros(len(arr), dtype=np.int8)
        lk = arr.paths.strings
        ck = arr.comms.strings
        cols["scenario"].extend([i] * len(arr))
        cols["kind"].extend([kind] * len(arr))
        cols["ts"].extend(arr.ts.tolist())
        cols["pid"].extend(arr.pid.tolist())
        cols["syscall"].extend(id_to_sys.get(int(s), "unknown") for s in arr.syscall)
        cols["path"].extend(lk[p] if p >= 0 else "" for p in arr.path_id)
        cols["new_path"].extend(lk[p] if p >= 0 else "" for p in arr.new_path_id)
        cols["nbytes"].extend(arr.nbytes.tolist())
        cols["comm"].extend(ck[c] if c >= 0 else "" for c in arr.comm_id)
        cols["label"].extend(np.asarray(y, dtype=np.int8).tolist())
        manifest.append({
            "scenario": i, "kind": kind, "seed": cfg.seed,
            "events": len(arr),
            "attack_window": None if win is None else
            {"t_start": win.t_start, "t_end": win.t_end,
             "target_dir": win.target_dir},
        })

    n_rows = len(cols["ts"])
    try:
        import pyarrow as pa
        import pyarrow.parquet as pq

        table = pa.table(cols)
        data_path = os.path.join(out_dir, "events.parquet")
        pq.write_table(table, data_path)
    except ImportError:  # pragma: no cover - pyarrow is in the image
        import csv

        data_path = os.path.join(out_dir, "events.csv")
        with open(data_path, "w", newline="") as f:
            w = csv.writer(f)
            keys = list(cols)
            w.writerow(keys)
            for r in range(n_rows):
                w.writerow([cols[k][r] for k in keys])
    meta = {
        "rows": n_rows,
        "scenarios": manifest,
        "hours": round(n_scenarios * duration_s / 3600.0, 3),
        "data": os.path.basename(data_path),
    }
    with open(os.path.join(out_dir, "manifest.json"), "w") as f:
        json.dump(meta, f, indent=2)
    return meta


def main(argv=None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="datasets/synthetic")
    ap.add_argument("--scenarios", type=int, default=12)
    ap.add_argument("--duration", type=float, default=120.0)
    ap.add_argument("--rate", type=float, default=400.0)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args(argv)
    meta = export(args.out, args.scenarios, args.duration, args.rate, args.seed)
    print(json.dumps({k: v for k, v in meta.items() if k != "scenarios"}))
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
