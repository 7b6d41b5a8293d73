"""Driver bench contract: the exact torchrun launch the driver uses at
round end must produce one JSON line with the agreed schema (weak-scaling
whole-job value, per-rank max step time). Runs on CPU with shrunk sizes."""

import json
import os
import subprocess
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_FIELDS = [
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
]


def run_bench(cmd, env):
    out = subprocess.run(
        cmd, cwd=REPO_ROOT, env=env, capture_output=True, text=True,
        timeout=420,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert lines, out.stdout[-2000:]
    return json.loads(lines[-1])


def test_bench_single_rank():
    env = {**os.environ, "FAABRIC_BENCH_BASE_OFFSET": "12000",
           "LOG_LEVEL": "error"}
    parsed = run_bench(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "2",
         "--warmup", "1", "--batch", "8"],
        env,
    )
    for f in REQUIRED_FIELDS:
        assert f in parsed, f
    assert parsed["n_gpus"] == 1
    assert parsed["steps"] == 2
    assert parsed["value"] > 0
    assert parsed["scaling"] == "weak"
    assert parsed["config"]["global_batch"] == 8


def test_bench_torchrun_two_ranks():
    """The N>1 launch shape the driver uses (one rank per GPU; gloo/host
    plane on CPU here)."""
    env = {**os.environ, "FAABRIC_BENCH_BASE_OFFSET": "12600",
           "LOG_LEVEL": "error"}
    parsed = run_bench(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29731", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "8"],
        env,
    )
    assert parsed["n_gpus"] == 2
    assert parsed["value"] > 0
    # Whole-job aggregate: batch is per host x N
    assert parsed["config"]["global_batch"] == 16
    assert parsed["config"]["pingpong_bytes"] > 0
