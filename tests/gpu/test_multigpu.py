"""Multi-GPU RCCL sharding tests — gated on device count (SURVEY §4
blueprint item 5). The driver's per-round boxes have 1 GPU, so these run
only on multi-GPU nodes (e.g. the round-end 8-GPU scaling tier runs
bench.py directly; this suite is the pytest-level equivalent)."""
import os
import subprocess
import sys

import pytest
import torch

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(torch.cuda.device_count() < 2,
                       reason="needs >= 2 GPUs"),
]

_ROOT = os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__))))


@pytest.mark.timeout(600)
def test_bench_two_gpus_rccl():
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29815", "bench.py", "--gpus", "2",
         "--steps", "8", "--warmup", "2"],
        cwd=_ROOT, capture_output=True, text=True, timeout=560)
    assert out.returncode == 0, out.stderr[-2000:]
    import json
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][0]
    r = json.loads(line)
    assert r["n_gpus"] == 2 and r["config"]["gpu"] is True
