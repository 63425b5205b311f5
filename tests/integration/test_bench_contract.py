"""Protects the driver contract: `python bench.py` emits exactly one JSON
line with the agreed schema, on CPU, within a bounded time."""
import json
import os
import subprocess
import sys

import pytest

_ROOT = os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__))))


@pytest.mark.timeout(300)
def test_bench_single_rank_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1"],
        cwd=_ROOT, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    r = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in r, key
    assert r["metric"] == "predict_req_per_s"
    assert r["n_gpus"] == 1 and r["steps"] == 3 and r["warmup"] == 1
    assert r["higher_is_better"] is True
    assert r["scaling"] == "weak"
    assert r["data"] == "synthetic"
    assert r["value"] > 0 and r["ms_per_step"] > 0
    assert r["config"]["global_batch"] == 32
    assert r["config"]["parallelism"] == "dp1"


@pytest.mark.timeout(300)
def test_bench_torchrun_two_ranks():
    """The exact launcher shape the driver uses for N>1 (gloo on CPU)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29811", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1"],
        cwd=_ROOT, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    r = json.loads(lines[0])
    assert r["n_gpus"] == 2
    assert r["config"]["parallelism"] == "dp2"
    assert r["config"]["global_batch"] == 64
