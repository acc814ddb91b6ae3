#!/usr/bin/env python3
"""Flagship benchmark: trace-events/sec through GraphSAGE-T + BiLSTM training.

Measures the BASELINE.json metric ("trace-events/sec through GraphSAGE-T+LSTM
at 1/2/4/8 MI355X") on synthetic trace windows (the reference publishes no GPU
numbers; its headline ingest figure is 1,250 evt/s on a 4-core VM —
BASELINE.md).  Each timed step is one full bf16 training step (forward +
backward + gradient all-reduce + optimizer) of the joint model on one 30 s
window of a synthetic LockBit scenario; events/sec counts the raw trace
events the processed windows represent.

Weak scaling: every rank owns identically-shaped windows (per-GPU work fixed);
the whole-job value sums events over ranks and takes the max step time.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N ...
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

from nerrf_amd.models.graphsage import SageConfig
from nerrf_amd.models.joint import JointConfig, NerrfJointModel
from nerrf_amd.models.lstm import LSTMConfig
from nerrf_amd.parallel.ddp import GradAllReducer, init_distributed

BASELINE_EVT_S = 1250.0  # reference tracker peak throughput


def build_bench_batches(rank: int, n_windows: int, scale: str):
    """Prebuild identically-shaped-per-rank window batches."""
    shapes = {
        # benign_rate, duration, n_files, n_victims
        "tiny": (200.0, 31.0, 200, 12),  # CPU CI smoke only
        "small": (2_000.0, 60.0, 2_000, 48),
        "full": (20_000.0, 60.0, 16_000, 64),
    }[scale]
    rate, dur, n_files, n_victims = shapes
    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.data.dataset import window_to_batch
    from nerrf_amd.graph.constructor import sliding_windows

    batches = []
    i = 0
    while len(batches) < n_windows:
        cfg = SynthConfig(
            duration_s=dur,
            benign_rate_hz=rate,
            n_benign_files=n_files,
            n_victim_files=n_victims,
            attack=True,
            seed=1234 + 7919 * (rank * 97 + i),
        )
        arr, win = generate(cfg)
        min_events = int(rate * 30.0 * 0.8)  # full-rate windows only: the
        # attack tail past the benign span yields sparse windows that would
        # make step shapes uneven
        for t0, evw in sliding_windows(arr, window_s=30.0, stride_s=30.0):
            if len(evw) < min_events:
                continue
            batches.append(window_to_batch(evw, win, fanout=16, seq_len=100, seed=i))
            if len(batches) >= n_windows:
                break
        i += 1
    return batches


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--scale", choices=["tiny", "small", "full"], default="full")
    ap.add_argument("--windows", type=int, default=2, help="prebuilt window-batches per rank")
    ap.add_argument(
        "--batch-windows", type=int, default=4,
        help="30s windows fused per training batch (disjoint graph union)",
    )
    ap.add_argument("--dtype", choices=["bf16", "fp32"], default="bf16")
    ap.add_argument(
        "--graphs",
        choices=["auto", "on", "off"],
        default="off",
        help="hipGraph-capture the whole training step (measured neutral at "
        "full scale — the step is kernel-bound — so off by default)",
    )
    args = ap.parse_args()

    rank, world, local_rank = init_distributed()
    has_gpu = torch.cuda.is_available()
    if has_gpu:
        torch.cuda.set_device(local_rank)
        device = torch.device(f"cuda:{local_rank}")
        from nerrf_amd.perf import enable_tuned_gemms

        enable_tuned_gemms()
    else:
        device = torch.device("cpu")
    dtype = torch.bfloat16 if (args.dtype == "bf16" and has_gpu) else torch.float32
    scale = args.scale
    if not has_gpu and scale == "full":
        scale = "small"  # CPU hosts cannot time the full config meaningfully

    from nerrf_amd.data.dataset import collate_windows

    raw_np = build_bench_batches(rank, args.windows * args.batch_windows, scale)
    batches_np = [
        collate_windows(raw_np[i : i + args.batch_windows])
        for i in range(0, len(raw_np), args.batch_windows)
    ]
    batches = [b.to_torch(device=device, dtype=dtype) for b in batches_np]
    events_per_step = [int(b.n_events) for b in batches_np]

    use_graphs = args.graphs == "on" or (
        args.graphs == "auto" and world == 1 and has_gpu
    )
    model = NerrfJointModel(JointConfig(sage=SageConfig(), lstm=LSTMConfig())).to(
        device=device, dtype=dtype
    )
    # fused AdamW measured neutral-to-slightly-slower than foreach on
    # MI355X at this step (243.7 vs 242.5 ms, profiles/PROFILES.md) —
    # foreach stays the default; NERRF_FUSED_ADAM=1 to re-test
    fused_ok = has_gpu and os.environ.get("NERRF_FUSED_ADAM", "0") == "1"
    opt = torch.optim.AdamW(
        model.parameters(), lr=1e-3, weight_decay=1e-4,
        fused=fused_ok, foreach=not fused_ok,
        capturable=use_graphs,
    )
    reducer = GradAllReducer(model)
    reducer.broadcast_params(model)

    def step_body(b) -> None:
        node_logit, edge_logit, seq_logit = model(b)
        losses = model.loss(node_logit, edge_logit, seq_logit, b)
        opt.zero_grad(set_to_none=False)
        losses["total"].backward()
        reducer.finalize()
        opt.step()

    def step(i: int) -> int:
        step_body(batches[i % len(batches)])
        return events_per_step[i % len(batches)]

    if use_graphs:
        # hipGraph-capture one full training step per prebuilt window shape;
        # replay turns ~2.5k launches/step into one graph launch.
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for b in batches:
                    for _ in range(2):
                        step_body(b)
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            graphs = []
            for b in batches:
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    step_body(b)
                graphs.append(g)

            def step(i: int) -> int:  # noqa: F811 - graph replay path
                graphs[i % len(graphs)].replay()
                return events_per_step[i % len(batches)]

            if rank == 0:
                print(f"# hipGraph capture: {len(graphs)} step graphs", file=sys.stderr, flush=True)
        except Exception as e:  # pragma: no cover
            use_graphs = False
            if rank == 0:
                print(f"# hipGraph capture failed, eager fallback: {e}", file=sys.stderr, flush=True)

    import torch.distributed as dist

    def barrier_sync():
        if dist.is_initialized():
            dist.barrier()
        if has_gpu:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        step(i)
    barrier_sync()
    t0 = time.perf_counter()
    events_done = 0
    for i in range(args.steps):
        events_done += step(i)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # whole-job: max time over ranks, sum of events over ranks
    # (tensors must live on the backend's device: RCCL reduces GPU tensors)
    red_dev = device if (dist.is_initialized() and dist.get_backend() == "nccl") else "cpu"
    t_tensor = torch.tensor([elapsed], dtype=torch.float64, device=red_dev)
    e_tensor = torch.tensor([float(events_done)], dtype=torch.float64, device=red_dev)
    if dist.is_initialized():
        dist.all_reduce(t_tensor, op=dist.ReduceOp.MAX)
        dist.all_reduce(e_tensor, op=dist.ReduceOp.SUM)
    t_max = float(t_tensor[0])
    e_sum = float(e_tensor[0])
    value = e_sum / t_max

    if rank == 0:
        result = {
            "metric": "trace_events_per_sec_graphsage_lstm_train",
            "value": value,
            "unit": "events/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": t_max / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": value / BASELINE_EVT_S,
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": "GraphSAGE-T(28x128)+BiLSTM(2x256) joint",
                "global_batch": world * args.batch_windows,
                "seq_len": 100,
                "window_s": 30,
                "parallelism": f"dp{world}",
                "hipgraph": use_graphs,
                "scale": scale,
                "events_per_window": int(np.mean(events_per_step)),
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
