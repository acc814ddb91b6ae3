"""Training entrypoint (also exposed as ai/train.py, the layout the
reference README promises at README.md:72-76 but never ships).

Modes:
  * --toy <trace.csv>: BASELINE config 1 — 2-layer GraphSAGE-T forward on a
    toy trace on CPU, zero GPU deps, proves schema -> graph -> model.
  * default: joint GraphSAGE-T + BiLSTM training on synthetic scenario
    windows, single GPU or DP over RCCL (torchrun), bf16 or fp32,
    checkpoint + ROC-AUC eval per epoch.
"""
from __future__ import annotations

import argparse
import json
import time
from pathlib import Path
from typing import Dict, List

import numpy as np
import torch

from .checkpoint import load_checkpoint, save_checkpoint
from .config import TrainConfig, load_config
from .data.dataset import WindowBatch, iterate_epochs, synth_window_batches
from .data.trace import load_trace
from .eval import detection_report
from .graph.constructor import build_graph
from .graph.sampling import sample_fanout, to_csr
from .models.graphsage import GraphSAGET, SageConfig
from .models.joint import NerrfJointModel
from .parallel.ddp import GradAllReducer, init_distributed


def _resolve_device(name: str) -> torch.device:
    if name == "auto":
        return torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    return torch.device(name)


def run_toy(trace_path: str) -> Dict[str, float]:
    """Config 1: tiny 2-layer GraphSAGE-T forward on CPU."""
    arr = load_trace(trace_path)
    g = build_graph(arr)
    csr = to_csr(g.edge_index, g.num_nodes, g.edge_weight)
    nbr_idx, nbr_w = sample_fanout(csr, fanout=8, seed=0)
    model = GraphSAGET(SageConfig(layers=2, hidden=64, fanout=8))
    gt = g.to_torch()
    with torch.no_grad():
        node_logit, edge_logit = model(
            gt["x"], torch.from_numpy(nbr_idx), torch.from_numpy(nbr_w),
            gt["edge_index"], gt["edge_weight"], gt["edge_ts"],
        )
    scores = torch.sigmoid(node_logit)
    out = {
        "events": len(arr),
        "nodes": g.num_nodes,
        "edges": g.num_edges,
        "score_mean": float(scores.mean()),
        "score_max": float(scores.max()),
        "edge_scores": int(edge_logit.numel() if edge_logit is not None else 0),
    }
    print(json.dumps({"mode": "toy", **out}))
    return out


@torch.no_grad()
def evaluate(model: NerrfJointModel, batches: List[WindowBatch], device, dtype) -> Dict[str, float]:
    model.eval()
    yn, sn, ys, ss, ye, se = [], [], [], [], [], []
    for b in batches:
        tb = b.to_torch(device=device, dtype=dtype)
        node_logit, edge_logit, seq_logit = model(tb)
        yn.append(tb["y_node"].cpu().numpy())
        sn.append(torch.sigmoid(node_logit.float()).cpu().numpy())
        if seq_logit is not None and seq_logit.numel():
            ys.append(tb["y_seq"].cpu().numpy())
            ss.append(torch.sigmoid(seq_logit.float()).cpu().numpy())
        if edge_logit is not None and edge_logit.numel():
            ye.append(tb["y_edge"].cpu().numpy())
            se.append(torch.sigmoid(edge_logit.float()).cpu().numpy())
    model.train()
    return detection_report(
        np.concatenate(yn), np.concatenate(sn),
        np.concatenate(ys) if ys else None, np.concatenate(ss) if ss else None,
        np.concatenate(ye) if ye else None, np.concatenate(se) if se else None,
    )


def run_training(cfg: TrainConfig, resume: str | None = None) -> Dict[str, float]:
    rank, world, local_rank = init_distributed()
    device = _resolve_device(cfg.run.device)
    if device.type == "cuda":
        torch.cuda.set_device(local_rank)
        device = torch.device(f"cuda:{local_rank}")
    dtype = torch.bfloat16 if cfg.optim.dtype == "bfloat16" else torch.float32

    d = cfg.data
    # each rank trains on its own scenario shard (data parallel over windows)
    train_batches = synth_window_batches(
        n_scenarios=d.n_scenarios,
        window_s=d.window_s,
        stride_s=d.stride_s,
        duration_s=d.duration_s,
        benign_rate_hz=d.benign_rate_hz,
        attack_fraction=d.attack_fraction,
        fanout=d.fanout,
        seq_len=d.seq_len,
        base_seed=d.seed + 100000 * rank,
        kinds=tuple(d.scenario_kinds),
        config_jitter=bool(getattr(d, "config_jitter", False)),
    )
    holdout = synth_window_batches(
        n_scenarios=cfg.run.eval_holdout,
        window_s=d.window_s,
        stride_s=d.stride_s,
        duration_s=d.duration_s,
        benign_rate_hz=d.benign_rate_hz,
        attack_fraction=0.5,
        fanout=d.fanout,
        seq_len=d.seq_len,
        base_seed=d.seed + 999331,  # disjoint from every rank's train shard
        kinds=tuple(d.scenario_kinds),
    )

    model = NerrfJointModel(cfg.model).to(device=device, dtype=dtype)
    opt = torch.optim.AdamW(model.parameters(), lr=cfg.optim.lr, weight_decay=cfg.optim.weight_decay)
    start_epoch = 0
    if resume:
        manifest, _ = load_checkpoint(resume, model, opt)
        start_epoch = int(manifest.get("epoch", 0))
        model = model.to(device=device, dtype=dtype)
    reducer = GradAllReducer(model)
    reducer.broadcast_params(model)

    step = 0
    t_start = time.perf_counter()
    events_done = 0
    report: Dict[str, float] = {}
    best_monitor = -1.0  # round-1 finding: the sequence head overfits by
    # epoch ~3; keep the best holdout epoch (seq F1 + node AUC) under
    # <checkpoint_dir>/best alongside the last epoch
    for epoch in range(start_epoch, cfg.optim.epochs):
        for batch in iterate_epochs(train_batches, 1, device=device, dtype=dtype, seed=epoch):
            node_logit, edge_logit, seq_logit = model(batch)
            losses = model.loss(node_logit, edge_logit, seq_logit, batch)
            opt.zero_grad(set_to_none=False)
            losses["total"].backward()
            reducer.finalize()
            torch.nn.utils.clip_grad_norm_(model.parameters(), cfg.optim.grad_clip)
            opt.step()
            step += 1
            events_done += int(batch["n_events"])
            if rank == 0 and step % cfg.run.log_every == 0:
                dt = time.perf_counter() - t_start
                print(
                    f"epoch {epoch} step {step} loss {losses['total'].item():.4f} "
                    f"(node {losses['node'].item():.4f} edge {losses['edge'].item():.4f} "
                    f"seq {losses['seq'].item():.4f}) evt/s {world * events_done / dt:.0f}"
                )
        if rank == 0:
            report = evaluate(model, holdout, device, dtype)
            print(f"epoch {epoch} eval: " + json.dumps({k: round(float(v), 4) for k, v in report.items()}))
            if (epoch + 1) % cfg.run.save_every_epochs == 0:
                save_checkpoint(
                    Path(cfg.run.checkpoint_dir),
                    model,
                    opt,
                    step=step,
                    epoch=epoch + 1,
                    metrics=report,
                    config=cfg,
                )
            monitor = float(report.get("seq_f1", 0.0)) + float(report.get("node_auc", 0.0))
            if monitor > best_monitor:
                best_monitor = monitor
                save_checkpoint(
                    Path(cfg.run.checkpoint_dir) / "best",
                    model,
                    opt,
                    step=step,
                    epoch=epoch + 1,
                    metrics=report,
                    config=cfg,
                )
    if rank == 0 and report:
        print("final: " + json.dumps({k: round(float(v), 4) for k, v in report.items()}))
    return report


def main(argv: List[str] | None = None) -> None:
    ap = argparse.ArgumentParser(description="nerrf-amd training")
    ap.add_argument("--config", default=None, help="YAML config path")
    ap.add_argument("--set", dest="overrides", action="append", default=[], help="a.b.c=value")
    ap.add_argument("--toy", default=None, help="toy trace CSV -> 2-layer CPU forward")
    ap.add_argument("--resume", default=None, help="checkpoint dir to resume from")
    args = ap.parse_args(argv)
    if args.toy:
        run_toy(args.toy)
        return
    cfg = load_config(args.config, args.overrides)
    run_training(cfg, resume=args.resume)


if __name__ == "__main__":
    main()
