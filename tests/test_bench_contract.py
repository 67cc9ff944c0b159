"""The driver contract: bench.py must print ONE JSON line with the agreed
fields; exercised here on the CPU debug path."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1"],
        cwd=REPO, timeout=600, capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, r.stdout
    out = json.loads(json_lines[-1])
    for field in ["metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"]:
        assert field in out, field
    assert out["n_gpus"] == 1 and out["steps"] == 2 and out["warmup"] == 1
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["value"] > 0
    assert "global_batch" in out["config"] and "parallelism" in out["config"]


def test_bench_under_torchrun_world2():
    """The exact driver launch shape for N>1 (CPU/gloo fallback here)."""
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29733", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1"],
        cwd=REPO, timeout=600, capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-2000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1  # rank 0 only
    out = json.loads(json_lines[0])
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["config"]["global_batch"] == 16  # cpu debug batch 8 x 2
