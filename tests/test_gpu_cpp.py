"""GPU test of the C++ drop-in harness: the reference kernel-parity recipe
(kernel_test.cpp) driven from C++ through include/skelly_evaluator.hpp over
the C-ABI."""

import os
import subprocess

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_cpp_dropin_parity(hip_lib_path):
    exe = os.path.join(REPO, "examples", "cpp_dropin")
    if not os.path.exists(exe):
        subprocess.run(["make", "-C", os.path.join(REPO, "examples")],
                       check=True, capture_output=True)
    r = subprocess.run([exe], capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, f"cpp_dropin failed: {r.stdout} {r.stderr}"
    assert "frobenius err" in r.stdout
