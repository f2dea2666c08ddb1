"""bench.py driver contract: JSON line shape, and the torchrun multi-process
path (CPU/gloo debug mode — the same code the driver runs with RCCL on the
8-GPU node)."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                 "dtype", "data", "config"}


def _last_json_line(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output: {stdout[-1500:]}")


@pytest.mark.timeout(600)
def test_bench_single_process_cpu_debug():
    res = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--gpus", "1",
         "--steps", "2", "--warmup", "1", "--debug-cpu"],
        capture_output=True, text=True, timeout=540, cwd=ROOT)
    assert res.returncode == 0, res.stderr[-2000:]
    payload = _last_json_line(res.stdout)
    assert REQUIRED_KEYS <= set(payload.keys())
    assert payload["n_gpus"] == 1
    assert payload["scaling"] == "weak"
    assert payload["data"] == "synthetic"
    assert payload["higher_is_better"] is True
    assert payload["value"] > 0


@pytest.mark.timeout(600)
def test_bench_torchrun_two_ranks_cpu_debug():
    """Exactly the driver's N>1 launch shape, on gloo."""
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29411", os.path.join(ROOT, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1", "--debug-cpu"],
        capture_output=True, text=True, timeout=540, cwd=ROOT,
        env={**os.environ, "MASTER_ADDR": "127.0.0.1"})
    assert res.returncode == 0, res.stderr[-2000:]
    payload = _last_json_line(res.stdout)
    assert payload["n_gpus"] == 2
    assert payload["config"]["parallelism"] == "dp2"
    assert payload["config"]["global_batch"] == 8  # 4/rank in debug mode
