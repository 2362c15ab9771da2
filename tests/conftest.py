import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def oracle_mod():
    import oracle
    oracle.lib()  # builds liboracle_cpu.so if needed
    return oracle


@pytest.fixture(scope="session")
def golden_dir():
    return os.path.join(REPO, "tests", "golden")


@pytest.fixture(scope="session")
def hip_lib_path():
    """Ensure the HIP extension exists (builds it if hipcc is available)."""
    path = os.path.join(REPO, "skellysim_amd", "libskellyhip.so")
    if not os.path.exists(path):
        subprocess.run(["make", "-C", os.path.join(REPO, "skellysim_amd", "csrc")],
                       check=True, capture_output=True)
    return path
