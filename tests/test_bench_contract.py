"""bench.py driver-contract checks (CPU: gloo backend, world_size 2).

The round-end driver runs `python bench.py --gpus N ...` (N>1 under
torch.distributed.run); this guards that exact launch path so a scaling
run on an 8-GPU node cannot fail on plumbing.
"""

from __future__ import annotations

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _check_result(line: str, n_gpus: int) -> None:
    result = json.loads(line)
    assert result["metric"] == "map_items_per_sec"
    assert result["n_gpus"] == n_gpus
    assert result["value"] > 0
    assert result["scaling"] == "weak"
    assert result["data"] == "synthetic"
    assert result["config"]["global_batch"] > 0


def test_bench_single_process(tmp_path):
    proc = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "1",
         "--warmup", "0", "--items-per-gpu", "64"],
        cwd=REPO, capture_output=True, text=True, timeout=180,
        env={**os.environ, "MASTER_PORT": "0"},
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    line = [ln for ln in proc.stdout.splitlines() if ln.startswith("{")][-1]
    _check_result(line, 1)


@pytest.mark.timeout(240)
def test_bench_torchrun_world2(tmp_path):
    """The N>1 launch the driver uses, on CPU (gloo, 127.0.0.1)."""
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", "bench.py", "--gpus", "2",
         "--steps", "1", "--warmup", "0", "--items-per-gpu", "64"],
        cwd=REPO, capture_output=True, text=True, timeout=220,
    )
    assert proc.returncode == 0, (proc.stdout[-1000:], proc.stderr[-2000:])
    line = [ln for ln in proc.stdout.splitlines() if ln.startswith("{")][-1]
    _check_result(line, 2)
