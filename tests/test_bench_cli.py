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


@pytest.mark.skipif(torch.cuda.is_available(), reason="GPU present")
def test_no_gpu_is_a_clean_json_error():
    r = subprocess.run([sys.executable, BENCH, "--steps", "1", "--warmup", "0"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 1
    out = json.loads(r.stdout.strip().splitlines()[-1])
    assert "error" in out
