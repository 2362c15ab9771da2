"""ctypes binding to the in-tree HIP extension (libskellyhip.so).

The extension is built in-tree by `make -C skellysim_amd/csrc` (driven by
__graft_entry__.build()) so the .so travels with the repo snapshot. If it is
missing, every entry point raises RuntimeError — the product path never falls
back to a CPU implementation.
"""

import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "libskellyhip.so")

_lib = None
_load_error = None

_DP = ctypes.POINTER(ctypes.c_double)
_LL = ctypes.c_longlong


def _bind(lib):
    lib.skelly_hip_version.restype = ctypes.c_char_p
    lib.skelly_hip_last_error.restype = ctypes.c_char_p
    lib.skelly_hip_device_count.restype = ctypes.c_int
    lib.skelly_hip_set_device.argtypes = [ctypes.c_int]
    lib.stokeslet_direct_gpu_impl.argtypes = [_DP, _DP, ctypes.c_int, _DP, _DP, ctypes.c_int]
    lib.stokeslet_direct_gpu_impl.restype = None
    lib.stresslet_direct_gpu_impl.argtypes = [_DP, _DP, ctypes.c_int, _DP, _DP, ctypes.c_int]
    lib.stresslet_direct_gpu_impl.restype = None
    lib.skelly_stokeslet_host.argtypes = [_DP, _DP, _LL, _DP, _DP, _LL, ctypes.c_double]
    lib.skelly_stresslet_host.argtypes = [_DP, _DP, _LL, _DP, _DP, _LL, ctypes.c_double]
    lib.skelly_oseen_contract_host.argtypes = [_DP, _DP, _DP, _DP, _LL, _LL,
                                               ctypes.c_double, ctypes.c_double, ctypes.c_double]
    lib.skelly_rotlet_host.argtypes = [_DP, _DP, _DP, _DP, _LL, _LL,
                                       ctypes.c_double, ctypes.c_double, ctypes.c_double]
    pv = ctypes.c_void_p
    lib.skelly_stokeslet_device.argtypes = [pv, pv, _LL, pv, pv, _LL, ctypes.c_double, pv]
    lib.skelly_stresslet_device.argtypes = [pv, pv, _LL, pv, pv, _LL, ctypes.c_double, pv]
    lib.skelly_oseen_contract_device.argtypes = [pv, pv, pv, pv, _LL, _LL,
                                                 ctypes.c_double, ctypes.c_double,
                                                 ctypes.c_double, pv]
    lib.skelly_rotlet_device.argtypes = [pv, pv, pv, pv, _LL, _LL,
                                         ctypes.c_double, ctypes.c_double, ctypes.c_double, pv]
    lib.skelly_stresslet_normal_density_host.argtypes = [_DP, _DP, _DP, _DP, _LL,
                                                          ctypes.c_double, ctypes.c_double]
    lib.skelly_stresslet_normal_density_device.argtypes = [pv, pv, pv, pv, _LL, _LL,
                                                            ctypes.c_double, ctypes.c_double, pv]
    lib.skelly_stresslet_times_normal_device.argtypes = [pv, pv, pv, _LL, ctypes.c_double,
                                                          ctypes.c_double, pv]
    lib.skelly_oseen_tensor_batched_device.argtypes = [pv, pv, _LL, _LL, ctypes.c_double,
                                                        ctypes.c_double, ctypes.c_double, pv]
    lib.skelly_fp64_peak_tflops.argtypes = [ctypes.POINTER(ctypes.c_double)]
    lib.skelly_set_persistent_ws.argtypes = [ctypes.c_int]
    lib.skelly_set_persistent_ws.restype = None


def lib():
    """The loaded extension; raises loudly if it is missing/unloadable."""
    global _lib, _load_error
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            raise RuntimeError(
                f"skellysim_amd HIP extension not found at {_LIB_PATH}. "
                "Build it with `make -C skellysim_amd/csrc` (or run "
                "__graft_entry__.build()). The product path has no CPU fallback.")
        try:
            _lib = ctypes.CDLL(_LIB_PATH)
        except OSError as e:
            _load_error = e
            raise RuntimeError(f"failed to load {_LIB_PATH}: {e}") from e
        _bind(_lib)
    return _lib


def check(rc, what):
    if rc != 0:
        err = lib().skelly_hip_last_error().decode()
        raise RuntimeError(f"{what} failed (rc={rc}): {err}")
