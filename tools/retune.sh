#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
export PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunable_r2_.csv
export PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=200
export PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=200
timeout 1200 python bench.py --steps 2 --warmup 1 --scale full --windows 1 --batch-windows 4 > gpurun_out/retune.log 2>&1
export PYTORCH_TUNABLEOP_TUNING=0
timeout 300 python bench.py --steps 10 --warmup 3 --scale full --windows 2 >> gpurun_out/retune.log 2>&1
tail -2 gpurun_out/retune.log
