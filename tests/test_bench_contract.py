"""bench.py driver-contract tests (CPU).

The round driver runs `python bench.py --gpus N --steps K --warmup W`
(N>1 under torch.distributed.run) and parses ONE JSON line from rank 0.
These tests execute that exact contract on CPU (gloo, scaled-down smoke
mode) so a contract regression is caught before it can cost a GPU run.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def _last_json_line(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout[-2000:]}")


@pytest.mark.timeout(300)
def test_bench_single_process_contract():
    p = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=280)
    assert p.returncode == 0, p.stderr[-2000:]
    rec = _last_json_line(p.stdout)
    for k in REQUIRED:
        assert k in rec, k
    assert rec["n_gpus"] == 1 and rec["steps"] == 2
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    assert rec["dtype"] == "fp32" and rec["data"] == "synthetic"


@pytest.mark.timeout(600)
def test_bench_torchrun_two_rank_contract():
    """The exact multi-rank launch line the driver uses, world_size=2 on
    CPU/gloo; rank 0 must print the one JSON line with MAX-over-ranks
    timing."""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    p = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29581", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=560)
    assert p.returncode == 0, (p.stdout[-1500:], p.stderr[-2000:])
    rec = _last_json_line(p.stdout)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "client-sharded dp2"
    assert rec["value"] > 0
