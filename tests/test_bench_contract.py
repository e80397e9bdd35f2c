"""bench.py must print exactly one valid JSON line with the driver's keys."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "2", "--height", "32", "--width", "64", "--seq-len", "4",
         "--encoder", "vgg", "--kernels", "eager", "--no-decode"],
        cwd=REPO, capture_output=True, text=True, timeout=570,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.strip()]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    for key in ["metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"]:
        assert key in rec, key
    assert rec["n_gpus"] == 1 and rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["value"] > 0 and rec["higher_is_better"] is True
    assert rec["scaling"] == "weak" and rec["data"] == "synthetic"
    assert rec["config"]["parallelism"] == "dp1"


@pytest.mark.timeout(600)
def test_bench_world2_contract():
    """The exact launch pattern the driver uses for N>1 (one rank per
    device over torch.distributed.run) must produce one JSON line with
    whole-job aggregation — gloo/CPU here, RCCL on the GPU node."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29613", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "2", "--height", "32",
         "--width", "64", "--seq-len", "4", "--encoder", "vgg",
         "--kernels", "eager", "--no-decode"],
        cwd=REPO, capture_output=True, text=True, timeout=570,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.strip().startswith("{")]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 4    # whole-job aggregate
    assert rec["value"] > 0
