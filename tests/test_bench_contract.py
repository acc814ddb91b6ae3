"""bench.py driver-contract tests: JSON schema, single- and multi-process."""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _last_json_line(out: str) -> dict:
    lines = [ln for ln in out.strip().splitlines() if ln.startswith("{")]
    assert lines, f"no JSON line in output:\n{out[-2000:]}"
    return json.loads(lines[-1])


def test_bench_single_process_contract():
    proc = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--scale", "tiny", "--windows", "1", "--batch-windows", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=900,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    res = _last_json_line(proc.stdout)
    assert REQUIRED_KEYS <= set(res)
    assert res["n_gpus"] == 1
    assert res["value"] > 0
    assert res["scaling"] == "weak"
    assert res["higher_is_better"] is True
    assert res["config"]["parallelism"] == "dp1"
    assert res["data"] == "synthetic"


@pytest.mark.timeout(900)
def test_bench_torchrun_world2_gloo():
    """The driver's multi-rank launch shape, on CPU with gloo (world=2)."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29517",
            "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--scale", "tiny", "--windows", "1", "--batch-windows", "1",
        ],
        cwd=ROOT, capture_output=True, text=True, timeout=800, env=env,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    res = _last_json_line(proc.stdout)
    assert res["n_gpus"] == 2
    assert res["config"]["parallelism"] == "dp2"
    assert res["value"] > 0


def test_bench_torchrun_world4_gloo():
    """World-4 launch contract: the 8-GPU driver run
    differs from world-2 only in degree; exercise a deeper rank fan-out on
    gloo so bucket-order determinism holds beyond the pairwise case."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "4",
            "--master-addr", "127.0.0.1", "--master-port", "29519",
            "bench.py", "--gpus", "4", "--steps", "2", "--warmup", "1",
            "--scale", "tiny", "--windows", "1", "--batch-windows", "1",
        ],
        cwd=ROOT, capture_output=True, text=True, timeout=1200, env=env,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    res = _last_json_line(proc.stdout)
    assert res["n_gpus"] == 4
    assert res["config"]["parallelism"] == "dp4"
    assert res["value"] > 0
