"""Streaming-inference benchmark (BASELINE config 5 evidence): events/s
through the online scoring path (delta compaction on GPU + GNN + LSTM +
MCTS planning) with the pretrained model."""
import sys, os, time, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from nerrf_amd.data.synth import SynthConfig, generate
from nerrf_amd.serve.engine import StreamingEngine, load_model_from_checkpoint
from nerrf_amd.perf import enable_tuned_gemms

enable_tuned_gemms()
dev = "cuda:0" if torch.cuda.is_available() else "cpu"
_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
model = load_model_from_checkpoint(os.path.join(_ROOT, "checkpoints", "pretrained"))
engine = StreamingEngine(model=model, device=dev, dtype=torch.bfloat16 if dev != "cpu" else torch.float32)
engine.store.window_s = 1e9

arr, _ = generate(SynthConfig(duration_s=30.0, benign_rate_hz=20000.0, n_benign_files=16000,
                              n_victim_files=64, seed=9))
t0 = time.perf_counter()
engine.ingest_events(arr)
t_ingest = time.perf_counter() - t0

# warmup + timed scoring
det = engine.score_window()
if dev != "cpu":
    torch.cuda.synchronize()
n = 8
t0 = time.perf_counter()
for _ in range(n):
    det = engine.score_window()
if dev != "cpu":
    torch.cuda.synchronize()
t_score = (time.perf_counter() - t0) / n

t0 = time.perf_counter()
plan = engine.plan(det, n_sims=1024, use_gpu=(dev != "cpu"))
t_plan = time.perf_counter() - t0

out = {
    "window_events": det.window_events,
    "ingest_s": t_ingest,
    "score_s_per_window": t_score,
    "inference_events_per_s": det.window_events / t_score,
    "alarm": det.alarm,
    "plan_s": t_plan,
    "plan_sims": plan.simulations,
    "plan": plan.describe(engine.planner_params.n_groups)[:4],
}
if getattr(engine, "last_timing", None):
    out["stage_ms"] = {k: round(v * 1e3, 2) for k, v in engine.last_timing.items()}
print(json.dumps(out))
