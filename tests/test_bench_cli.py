"""CPU guards for bench.py: the CLI parses, and on a GPU-less box it exits
with a clear JSON error instead of a traceback (the real bench runs only on
MI355X boxes)."""

import json
import os
import subprocess
import sys

import torch
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")


def test_help_parses():
    r = subprocess.run([sys.executable, BENCH, "--help"], capture_output=True, text=True)
    assert r.returncode == 0
    for flag in ("--gpus", "--steps", "--warmup", "--kernel"):
        assert flag in r.stdout


@pytest.mark.timeout(420)
def test_distributed_plumbing_rehearsal_world2():
    """Runs the EXACT multi-rank bench path the driver uses (torchrun,
    world_size 2, barriers, per-step source all-gather, max-over-ranks
    timing, one JSON line from rank 0) over gloo with a stub compute."""
    env = dict(os.environ, SKELLY_BENCH_REHEARSAL="gloo", MASTER_ADDR="127.0.0.1")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29581", BENCH,
         "--gpus", "2", "--steps", "2", "--warmup", "1", "--npoints", "10000",
         "--skip-cpu-baseline"],
        capture_output=True, text=True, timeout=360, env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout  # exactly one JSON line (rank 0)
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2 and out["steps"] == 2
    assert out["metric"] == "rehearsal (no compute)"
    assert out["config"]["n_src"] == 10000


@pytest.mark.skipif(torch.cuda.is_available(), reason="GPU present")
def test_no_gpu_is_a_clean_json_error():
    r = subprocess.run([sys.executable, BENCH, "--steps", "1", "--warmup", "0"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 1
    out = json.loads(r.stdout.strip().splitlines()[-1])
    assert "error" in out
