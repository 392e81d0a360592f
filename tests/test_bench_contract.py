"""The driver's bench contract: torchrun multi-rank launch emits ONE JSON
line from rank 0 with the whole-job aggregate (weak scaling, MAX-over-ranks
timing). Runs the real bench.py exactly as the driver does (gloo on CPU)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(420)
def test_torchrun_world2_bench_json():
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29741", "bench.py", "--gpus", "2", "--steps",
         "2", "--warmup", "1", "--cpu"],
        cwd=REPO, capture_output=True, text=True, timeout=400)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, "exactly one JSON line from rank 0"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2 and d["scaling"] == "weak"
    assert d["config"]["parallelism"] == "local_sgd_dp2"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    for key in ("metric", "unit", "steps", "warmup", "higher_is_better",
                "vs_baseline", "dtype", "data"):
        assert key in d


def test_single_process_bench_json():
    r = subprocess.run(
        [sys.executable, "bench.py", "--cpu", "--steps", "2", "--warmup",
         "1"], cwd=REPO, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    d = json.loads([ln for ln in r.stdout.splitlines()
                    if ln.startswith("{")][0])
    assert d["n_gpus"] == 1
