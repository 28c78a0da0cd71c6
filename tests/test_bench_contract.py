"""bench.py driver-contract tests (CPU).

The driver launches `bench.py` standalone (N=1) and under
`torch.distributed.run --nproc-per-node N` (N>1, one rank per GPU). Both
must emit exactly one JSON line on rank 0 with the contract fields. The
multi-rank case runs here over gloo (world_size 2) — it caught
bench.py initialising CUDA before checking availability, which would have
broken the driver's 8-GPU scaling tier.
"""

from __future__ import annotations

import json
import os
import subprocess
import sys

import pytest

pytestmark = [pytest.mark.integration, pytest.mark.slow]

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_FIELDS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _last_json_line(out: str) -> dict:
    for line in reversed(out.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out}")


@pytest.mark.timeout(300)
def test_bench_single_process():
    proc = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=280,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    data = _last_json_line(proc.stdout)
    assert REQUIRED_FIELDS <= set(data)
    assert data["n_gpus"] == 1
    assert data["steps"] == 2
    assert data["value"] > 0


@pytest.mark.timeout(300)
def test_bench_two_ranks_gloo():
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29731", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=280, env=env,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    data = _last_json_line(proc.stdout)
    assert data["n_gpus"] == 2
    assert data["config"]["parallelism"] == "dp2"
    # value is the whole-job aggregate over both ranks
    assert data["value"] > 0


@pytest.mark.timeout(300)
def test_bench_tp_mode_two_ranks_gloo():
    """--tp N: one model sharded over all ranks (BASELINE config #4 path).
    Proves the TP bench mode is launchable the moment an 8-GPU node
    appears (VERDICT r1 'harden TP for the driver's 8-GPU day')."""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29733", "bench.py", "--gpus", "2", "--tp", "2",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=280, env=env,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    data = _last_json_line(proc.stdout)
    assert data["n_gpus"] == 2
    assert data["config"]["parallelism"] == "tp2"
    # ONE engine's tokens (not 2×): global_batch equals per-engine batch
    assert data["config"]["global_batch"] == 4  # CPU tiny shape
    assert data["value"] > 0
