import sys, os, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from nerrf_amd.config import load_config
from nerrf_amd.train import run_training, evaluate
from nerrf_amd.data.dataset import synth_window_batches
from nerrf_amd.serve.engine import load_model_from_checkpoint

cfg = load_config(None, [
    "optim.dtype=bfloat16", "optim.epochs=3", "data.n_scenarios=10",
    "run.eval_holdout=4", "run.checkpoint_dir=gpurun_out/ckpt_mixed",
    "run.log_every=100",
])
run_training(cfg)
model = load_model_from_checkpoint("gpurun_out/ckpt_mixed").to("cuda", torch.bfloat16)
for kind in ("lockbit", "supply_chain"):
    hb = synth_window_batches(n_scenarios=3, attack_fraction=0.67, base_seed=555000,
                              kinds=(kind,))
    rep = evaluate(model, hb, "cuda", torch.bfloat16)
    print(f"family={kind}: " + json.dumps({k: round(float(v), 4) for k, v in rep.items() if "auc" in k or k.endswith("f1")}))
