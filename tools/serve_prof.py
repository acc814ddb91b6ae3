"""cProfile of the serving tick's HOST side (stage timers say ~12 ms of the
51 ms GPU tick is host work in the graph_merge stage; this names it).

Run on a GPU box:  NERRF_SERVE_TIMING=1 python tools/serve_prof.py
"""
import cProfile
import io
import os
import pstats
import sys
import time

import torch

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

from nerrf_amd.data.synth import SynthConfig, generate  # noqa: E402
from nerrf_amd.perf import enable_tuned_gemms  # noqa: E402
from nerrf_amd.serve.engine import (StreamingEngine,  # noqa: E402
                                    load_model_from_checkpoint)

enable_tuned_gemms()
dev = "cuda" if torch.cuda.is_available() else "cpu"
model = load_model_from_checkpoint(os.path.join(_ROOT, "checkpoints", "pretrained"))
engine = StreamingEngine(model=model, device=dev,
                         dtype=torch.bfloat16 if dev != "cpu" else torch.float32)
engine.store.window_s = 1e9
arr, _ = generate(SynthConfig(duration_s=30.0, benign_rate_hz=20000.0,
                              n_benign_files=16000, n_victim_files=64, seed=9))
engine.ingest_events(arr)
for _ in range(3):
    engine.score_window()
if dev != "cpu":
    torch.cuda.synchronize()

pr = cProfile.Profile()
t0 = time.perf_counter()
pr.enable()
for _ in range(8):
    engine.score_window()
pr.disable()
if dev != "cpu":
    torch.cuda.synchronize()
print(f"score_window x8: {(time.perf_counter() - t0) / 8 * 1e3:.2f} ms each")
s = io.StringIO()
pstats.Stats(pr, stream=s).sort_stats("tottime").print_stats(25)
print(s.getvalue())
print("stage_ms:", {k: round(v * 1e3, 2) for k, v in (engine.last_timing or {}).items()})
