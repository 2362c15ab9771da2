"""CPU tests of the C-ABI boundary: the in-tree .so builds for gfx950, loads,
and exports every symbol include/skelly_hip.h declares. No compute calls
(no GPU in the CPU CI container)."""

import ctypes
import os
import re

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(REPO, "include", "skelly_hip.h")


def declared_functions():
    """Function names declared in include/skelly_hip.h."""
    src = open(HEADER).read()
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)
    names = re.findall(r"^\s*(?:const\s+char\s*\*|int|void)\s+(\w+)\s*\(", src, re.M)
    return [n for n in names if n != "SKELLY_HIP_H"]


def test_header_declares_dropin_entry_points():
    names = declared_functions()
    # the reference seam (include/kernels.hpp:17-20)
    assert "stokeslet_direct_gpu_impl" in names
    assert "stresslet_direct_gpu_impl" in names
    assert len(names) >= 12


def test_lib_loads_and_exports_all_symbols(hip_lib_path):
    lib = ctypes.CDLL(hip_lib_path)
    missing = [n for n in declared_functions() if not hasattr(lib, n)]
    assert not missing, f"symbols missing from libskellyhip.so: {missing}"


def test_version_string(hip_lib_path):
    lib = ctypes.CDLL(hip_lib_path)
    lib.skelly_hip_version.restype = ctypes.c_char_p
    v = lib.skelly_hip_version().decode()
    assert "gfx950" in v


def test_last_error_initially_empty(hip_lib_path):
    lib = ctypes.CDLL(hip_lib_path)
    lib.skelly_hip_last_error.restype = ctypes.c_char_p
    assert lib.skelly_hip_last_error().decode() == ""


def test_code_object_targets_gfx950(hip_lib_path):
    """The embedded HSA code object is for gfx950 (MI355X) only."""
    blob = open(hip_lib_path, "rb").read()
    assert b"gfx950" in blob
    for other in (b"gfx90a", b"gfx942", b"sm_80", b"sm_90"):
        assert other not in blob


def test_native_module_fails_loudly_when_lib_missing(monkeypatch):
    from skellysim_amd import _native
    monkeypatch.setattr(_native, "_LIB_PATH", "/nonexistent/libskellyhip.so")
    monkeypatch.setattr(_native, "_lib", None)
    with pytest.raises(RuntimeError, match="no CPU fallback"):
        _native.lib()


def test_set_evaluator_strings(hip_lib_path):
    import skellysim_amd as ska
    st, dl = ska.set_evaluator("HIP")
    assert callable(st) and callable(dl)
    with pytest.raises(NotImplementedError):
        ska.set_evaluator("CPU")
    with pytest.raises(NotImplementedError):
        ska.set_evaluator("FMM")
    with pytest.raises(ValueError):
        ska.set_evaluator("bogus")
