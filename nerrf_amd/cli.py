"""nerrf command-line interface.

Commands (spec: reference ROADMAP.md:86, README.md:82 — `nerrf undo`,
`nerrf status` — plus the operational verbs this engine adds):

  nerrf status                       engine/store/checkpoint status + recorded detections
  nerrf undo --dir D                 detect + plan + sandbox-validate + restore
  nerrf undo --id ATK                replay a recorded response (serve/registry.py)
  nerrf simulate --dir D             run the reversible LockBit attack sim
  nerrf scenario --dir D             full e2e: seed -> attack -> detect ->
                                     plan -> rollback -> sha256 verify
  nerrf train ...                    training entrypoint (= ai/train.py)
  nerrf serve --trace T              tracker-sim + streaming engine demo

Entry point: `python -m nerrf_amd.cli <cmd>` (also installed as ./nerrf).
"""
from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path


def cmd_status(args) -> int:
    import torch

    from .ops.native import native_available

    info = {
        "version": __import__("nerrf_amd").__version__,
        "torch": torch.__version__,
        "gpu_available": torch.cuda.is_available(),
        "native_kernels_built": native_available(),
    }
    if args.checkpoint and Path(args.checkpoint, "checkpoint.json").exists():
        info["checkpoint"] = json.loads(Path(args.checkpoint, "checkpoint.json").read_text())
    from .serve.registry import DEFAULT_STATE_DIR, DetectionRegistry

    info["detections"] = DetectionRegistry(
        getattr(args, "state_dir", None) or DEFAULT_STATE_DIR).list()[:20]
    print(json.dumps(info, indent=2))
    return 0


def cmd_simulate(args) -> int:
    from .harness.attack_sim import run_attack, seed_files

    if args.seed_files:
        manifest = seed_files(args.dir, n_files=args.n_files, file_kb=args.file_kb)
        Path(args.dir, ".nerrf_manifest.json").write_text(json.dumps(manifest))
        print(f"seeded {len(manifest)} files in {args.dir}")
    report = run_attack(args.dir, trace_path=args.trace_out)
    print(
        json.dumps(
            {
                "files_attacked": len(report.files_attacked),
                "bytes_attacked": report.bytes_attacked,
                "duration_s": report.t_end - report.t_start,
                "trace_events": len(report.trace_events),
            }
        )
    )
    return 0


def cmd_undo(args) -> int:
    from .data.trace import load_trace
    from .serve.engine import StreamingEngine

    t0 = time.time()
    if args.id:
        # replay a RECORDED response (reference contract: nerrf undo --id):
        # the monitor already scored, planned and persisted the attack —
        # execute its rollback without re-scoring
        from .serve.registry import DEFAULT_STATE_DIR, DetectionRegistry
        from .serve.rollback import execute_rollback

        reg = DetectionRegistry(args.state_dir or DEFAULT_STATE_DIR)
        rec = reg.load(args.id)
        if rec is None:
            print(json.dumps({"error": f"unknown attack id {args.id}",
                              "known": [r["attack_id"] for r in reg.list()]}))
            return 1
        target = args.dir or rec.get("target_dir") or ""
        if not target:
            print(json.dumps({"error": "record has no target_dir; pass --dir"}))
            return 1
        manifest = None
        mpath = Path(target, ".nerrf_manifest.json")
        if mpath.exists():
            manifest = json.loads(mpath.read_text())
        result = execute_rollback(target, manifest=manifest, decrypt=True,
                                  validate_in_sandbox=True)
        out = {
            "attack_id": rec["attack_id"],
            "recorded_plan": rec.get("plan", {}).get("actions"),
            "indicators": rec.get("indicators"),
            "mttr_s": time.time() - t0,
            **result.as_dict(),
        }
        print(json.dumps(out, indent=2))
        return 0 if result.files_failed == 0 else 2
    if not args.dir:
        print(json.dumps({"error": "--dir is required without --id"}))
        return 1
    model = None
    if args.checkpoint:
        from .serve.engine import load_model_from_checkpoint

        model = load_model_from_checkpoint(args.checkpoint)
    engine = StreamingEngine(model=model, device=args.device)
    if args.trace:
        engine.ingest_events(load_trace(args.trace))
    det = engine.score_window()
    if not det.alarm and not args.force:
        print(json.dumps({"alarm": False, "msg": "no attack detected; use --force to undo anyway"}))
        return 1
    plan = engine.plan(det, n_sims=args.sims)
    manifest = None
    mpath = Path(args.dir, ".nerrf_manifest.json")
    if mpath.exists():
        manifest = json.loads(mpath.read_text())
    result = engine.respond(det, plan, args.dir, manifest=manifest)
    out = {
        "alarm": det.alarm,
        "indicators": det.indicators,
        "plan": plan.describe(engine.planner_params.n_groups),
        "plan_value": plan.root_value,
        "mttr_s": time.time() - t0,
        **result.as_dict(),
    }
    print(json.dumps(out, indent=2))
    return 0 if result.files_failed == 0 else 2


def cmd_scenario(args) -> int:
    from .harness.scenario import run_scenario

    report = run_scenario(
        work_dir=args.dir,
        n_files=args.n_files,
        file_kb=args.file_kb,
        device=args.device,
        n_sims=args.sims,
    )
    print(json.dumps(report, indent=2))
    return 0 if report["recovered_ok"] else 2


def cmd_serve(args) -> int:
    from .serve.engine import StreamingEngine

    if args.tracker:
        # live mode: consume an existing Tracker/StreamEvents endpoint and
        # monitor continuously (deploy/engine-deployment.yaml wiring)
        import torch

        model = None
        if args.checkpoint:
            from .serve.engine import load_model_from_checkpoint

            model = load_model_from_checkpoint(args.checkpoint)
        # bf16 on GPU enables the fused-MFMA scoring path
        dtype = torch.bfloat16 if args.device != "cpu" else torch.float32
        engine = StreamingEngine(model=model, device=args.device, dtype=dtype)
        from .serve.registry import DEFAULT_STATE_DIR, DetectionRegistry

        for status in engine.run_monitor(
            interval_s=args.interval,
            max_iterations=args.iterations,
            tracker_address=args.tracker,
            registry=DetectionRegistry(args.state_dir or DEFAULT_STATE_DIR),
            target_dir=args.target_dir,
        ):
            print(json.dumps(status), flush=True)
        return 0

    if not args.trace:
        print(json.dumps({"error": "serve needs --tracker <addr> or --trace <file>"}))
        return 2
    from .data.trace import load_trace
    from .serve.tracker_sim import TrackerSimServer

    trace = load_trace(args.trace)
    server = TrackerSimServer(trace, rate_multiplier=args.rate)
    server.start()
    print(f"tracker-sim listening on {server.address}")
    engine = StreamingEngine(device=args.device)
    n = engine.ingest_from_tracker(server.address, max_events=args.max_events, timeout_s=args.timeout)
    det = engine.score_window()
    server.stop()
    attack_id = None
    if det.alarm and args.state_dir:
        # persist for `nerrf undo --id` (same record the live monitor makes)
        from .serve.registry import DetectionRegistry

        plan = engine.plan(det, n_sims=256, use_gpu=(args.device != "cpu"))
        attack_id = DetectionRegistry(args.state_dir).record(
            det, plan, target_dir=args.target_dir,
            n_groups=engine.planner_params.n_groups,
        )
    print(
        json.dumps(
            {
                "events_ingested": n,
                "alarm": det.alarm,
                "attack_id": attack_id,
                "indicators": det.indicators,
                "suspicious_files": det.encrypted_paths[:10],
            },
            indent=2,
        )
    )
    return 0


def cmd_eval(args) -> int:
    """Reproduce the detection-quality table: per-family ROC-AUC / F1 on
    held-out synthetic scenarios, plus max model scores on the benign
    hard negatives (spec targets: ROC-AUC >= 0.90, seq F1 >= 0.95)."""
    import torch

    from .data.dataset import synth_window_batches
    from .serve.engine import load_model_from_checkpoint
    from .train import evaluate

    device = args.device
    dtype = torch.bfloat16 if (device != "cpu" and args.dtype == "bf16") else torch.float32
    model = load_model_from_checkpoint(args.checkpoint).to(device, dtype).eval()
    out = {}
    for kind in args.families.split(","):
        hb = synth_window_batches(n_scenarios=args.scenarios, attack_fraction=0.67,
                                  base_seed=555000, kinds=(kind,))
        rep = evaluate(model, hb, device, dtype)
        out[kind] = {k: round(float(v), 4) for k, v in rep.items()
                     if "auc" in k or k.endswith("f1")}
    for kind in args.negatives.split(","):
        if not kind:
            continue
        hb = synth_window_batches(n_scenarios=2, attack_fraction=0.0,
                                  base_seed=777000, benign_kinds=(kind,))
        mx = 0.0
        with torch.no_grad():
            for b in hb:
                tb = b.to_torch(device, dtype)
                nl, _, sl = model(tb)
                mx = max(mx, float(torch.sigmoid(nl.float()).max()))
                if sl is not None and sl.numel():
                    mx = max(mx, float(torch.sigmoid(sl.float()).max()))
        out[f"negative:{kind}"] = {"max_model_score": round(mx, 4)}
    print(json.dumps(out, indent=2))
    return 0


def cmd_train(args, extra) -> int:
    from .train import main as train_main

    train_main(extra)
    return 0


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="nerrf", description="nerrf-amd: undo computing engine")
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("status", help="engine status")
    p.add_argument("--checkpoint", default=None)

    p = sub.add_parser("simulate", help="run the reversible attack simulator")
    p.add_argument("--dir", required=True)
    p.add_argument("--seed-files", action="store_true")
    p.add_argument("--n-files", type=int, default=24)
    p.add_argument("--file-kb", type=int, default=64)
    p.add_argument("--trace-out", default=None)

    p = sub.add_parser("undo", help="detect + plan + rollback a directory")
    p.add_argument("--dir", default=None, help="target directory (defaults to the recorded one with --id)")
    p.add_argument("--id", default=None, help="recorded attack id (see `nerrf status`)")
    p.add_argument("--state-dir", default=None, help="detection registry dir")
    p.add_argument("--trace", default=None, help="trace file to ingest first")
    p.add_argument("--device", default="cpu")
    p.add_argument("--sims", type=int, default=1024)
    p.add_argument("--force", action="store_true")
    p.add_argument("--checkpoint", default=None, help="trained model checkpoint dir")

    p = sub.add_parser("scenario", help="full e2e attack->detect->recover")
    p.add_argument("--dir", required=True)
    p.add_argument("--n-files", type=int, default=16)
    p.add_argument("--file-kb", type=int, default=32)
    p.add_argument("--device", default="cpu")
    p.add_argument("--sims", type=int, default=512)

    p = sub.add_parser("serve", help="streaming engine (live tracker or trace demo)")
    p.add_argument("--trace", default=None, help="trace file (demo mode)")
    p.add_argument("--tracker", default=None, help="live Tracker/StreamEvents address")
    p.add_argument("--interval", type=float, default=5.0)
    p.add_argument("--iterations", type=int, default=None)
    p.add_argument("--checkpoint", default=None)
    p.add_argument("--rate", type=float, default=0.0)
    p.add_argument("--max-events", type=int, default=None)
    p.add_argument("--timeout", type=float, default=15.0)
    p.add_argument("--device", default="cpu")
    p.add_argument("--state-dir", default=None,
                   help="persist alarms for `nerrf undo --id` (default .nerrf/detections)")
    p.add_argument("--target-dir", default="", help="recorded rollback target")

    p = sub.add_parser("eval", help="detection-quality report on synthetic scenarios")
    p.add_argument("--checkpoint", default="checkpoints/pretrained")
    p.add_argument("--families", default="lockbit,supply_chain,supply_chain_net")
    p.add_argument("--negatives", default="benign_rotate,benign_backup,benign_build")
    p.add_argument("--scenarios", type=int, default=3)
    p.add_argument("--device", default="cpu")
    p.add_argument("--dtype", default="bf16")

    sub.add_parser("train", help="training entrypoint (args passed through)")
    sub.add_parser("tune", help="hparam search (args passed through to nerrf_amd.tune)")

    if argv is None:
        argv = sys.argv[1:]
    if argv and argv[0] == "train":
        return cmd_train(None, argv[1:])
    if argv and argv[0] == "tune":
        from .tune import main as tune_main

        return tune_main(argv[1:])
    args = ap.parse_args(argv)
    return {
        "status": cmd_status,
        "simulate": cmd_simulate,
        "undo": cmd_undo,
        "scenario": cmd_scenario,
        "serve": cmd_serve,
        "eval": cmd_eval,
    }[args.cmd](args)


if __name__ == "__main__":
    sys.exit(main())
