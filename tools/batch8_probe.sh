#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
export PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunable_b8_.csv
export PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=200 PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=200
timeout 1200 python bench.py --steps 2 --warmup 1 --windows 1 --batch-windows 8 > gpurun_out/b8.log 2>&1
export PYTORCH_TUNABLEOP_TUNING=0
timeout 600 python - <<'PYEOF' >> gpurun_out/b8.log 2>&1
import subprocess, sys, torch, json
r = subprocess.run([sys.executable, "bench.py", "--steps", "6", "--warmup", "2", "--windows", "1", "--batch-windows", "8"],
                   capture_output=True, text=True, env=None)
print(r.stdout.strip().splitlines()[-1] if r.stdout.strip() else r.stderr[-500:])
PYEOF
timeout 600 python - >> gpurun_out/b8.log 2>&1 <<'PYEOF'
# memory headroom at batch 8
import torch, json
from bench import build_bench_batches
from nerrf_amd.data.dataset import collate_windows
from nerrf_amd.models.joint import NerrfJointModel, JointConfig
from nerrf_amd.perf import enable_tuned_gemms
enable_tuned_gemms()
raw = build_bench_batches(0, 8, "full")
b = collate_windows(raw).to_torch("cuda:0", torch.bfloat16)
m = NerrfJointModel(JointConfig()).to("cuda:0", torch.bfloat16)
opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
nl, el, sl = m(b)
losses = m.loss(nl, el, sl, b)
losses["total"].backward(); opt.step(); torch.cuda.synchronize()
print(json.dumps({"max_mem_GB": torch.cuda.max_memory_allocated()/1e9}))
PYEOF
tail -3 gpurun_out/b8.log
