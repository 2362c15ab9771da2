"""Host-side mirror of SkellySim's evaluator interface.

Mirrors kernels::Evaluator (reference include/kernels.hpp:14-15):

    Evaluator(r_sl, r_dl, r_trg, f_sl, f_dl, eta) -> u [n_trg x 3]

Stokeslet consumers pass single-layer sources (r_sl, f_sl) with r_dl/f_dl
empty; stresslet consumers the reverse (see reference
tests/core/kernel_test.cpp:40,65). Results are divided by eta, exactly as the
reference host wrappers do (src/core/kernels.cpp:358,365).

Array convention: an (n, 3) C-order numpy array is byte-identical to the
reference's col-major 3 x n Eigen matrix (per-point xyz contiguous); (3, n)
inputs are accepted and transposed. Stresslet strengths are (n, 9).

Backend selection mirrors FiberContainerBase::set_evaluator
(src/core/fiber_container_base.cpp:20-33): set_evaluator("HIP") returns the
callables below; "CPU"/"GPU"/"FMM" raise — this engine IS the GPU backend and
ships no FMM or CPU compute path.
"""

import ctypes

import numpy as np

from . import _native

_DP = ctypes.POINTER(ctypes.c_double)


def _rows(a, dim, name):
    """Normalize to a C-contiguous (n, dim) fp64 array (accepts (dim, n))."""
    if a is None:
        return np.empty((0, dim))
    a = np.asarray(a, dtype=np.float64)
    if a.size == 0:
        return np.empty((0, dim))
    if a.ndim != 2:
        raise ValueError(f"{name}: expected 2-D array, got shape {a.shape}")
    if a.shape[1] != dim:
        if a.shape[0] == dim:
            a = a.T
        else:
            raise ValueError(f"{name}: expected (n, {dim}) or ({dim}, n), got {a.shape}")
    return np.ascontiguousarray(a)


def _ptr(a):
    return a.ctypes.data_as(_DP)


def stokeslet_direct_gpu(r_sl, r_dl, r_trg, f_sl, f_dl, eta):
    """Evaluator-signature Stokeslet (mirrors kernels::stokeslet_direct_gpu,
    src/core/kernels.cpp:361-366): uses (r_sl, f_sl), returns u/eta."""
    r_src = _rows(r_sl, 3, "r_sl")
    f_src = _rows(f_sl, 3, "f_sl")
    trg = _rows(r_trg, 3, "r_trg")
    u = np.zeros((len(trg), 3))
    rc = _native.lib().skelly_stokeslet_host(_ptr(r_src), _ptr(f_src), len(r_src),
                                             _ptr(trg), _ptr(u), len(trg), float(eta))
    _native.check(rc, "stokeslet")
    return u


def stresslet_direct_gpu(r_sl, r_dl, r_trg, f_sl, f_dl, eta):
    """Evaluator-signature stresslet (mirrors kernels::stresslet_direct_gpu,
    src/core/kernels.cpp:354-359): uses (r_dl, f_dl), returns u/eta."""
    r_src = _rows(r_dl, 3, "r_dl")
    f_src = _rows(f_dl, 9, "f_dl")
    trg = _rows(r_trg, 3, "r_trg")
    u = np.zeros((len(trg), 3))
    rc = _native.lib().skelly_stresslet_host(_ptr(r_src), _ptr(f_src), len(r_src),
                                             _ptr(trg), _ptr(u), len(trg), float(eta))
    _native.check(rc, "stresslet")
    return u


def oseen_contract_direct_gpu(r_src, r_trg, density, eta=1.0, reg=5e-3, epsilon_distance=1e-5):
    """Mirrors kernels::oseen_tensor_contract_direct (kernels.cpp:85-131;
    defaults kernels.hpp:34-35)."""
    src = _rows(r_src, 3, "r_src")
    trg = _rows(r_trg, 3, "r_trg")
    rho = _rows(density, 3, "density")
    u = np.zeros((len(trg), 3))
    rc = _native.lib().skelly_oseen_contract_host(_ptr(src), _ptr(trg), _ptr(rho), _ptr(u),
                                                  len(src), len(trg), float(eta), float(reg),
                                                  float(epsilon_distance))
    _native.check(rc, "oseen_contract")
    return u


def rotlet_gpu(r_src, r_trg, density, eta=1.0, reg=5e-3, epsilon_distance=1e-5):
    """Mirrors kernels::rotlet (kernels.cpp:206-242; defaults kernels.hpp:44-45)."""
    src = _rows(r_src, 3, "r_src")
    trg = _rows(r_trg, 3, "r_trg")
    rho = _rows(density, 3, "density")
    u = np.zeros((len(trg), 3))
    rc = _native.lib().skelly_rotlet_host(_ptr(src), _ptr(trg), _ptr(rho), _ptr(u),
                                          len(src), len(trg), float(eta), float(reg),
                                          float(epsilon_distance))
    _native.check(rc, "rotlet")
    return u


class Evaluator:
    """Callable mirroring kernels::Evaluator for one kernel type."""

    def __init__(self, fn):
        self._fn = fn

    def __call__(self, r_sl, r_dl, r_trg, f_sl, f_dl, eta):
        return self._fn(r_sl, r_dl, r_trg, f_sl, f_dl, eta)


def set_evaluator(name):
    """Mirror of the reference's string-keyed backend selection
    (fiber_container_base.cpp:20-33, periphery.cpp:337-352). Returns
    (stokeslet_evaluator, stresslet_evaluator) for name == "HIP"."""
    if name == "HIP":
        return Evaluator(stokeslet_direct_gpu), Evaluator(stresslet_direct_gpu)
    if name in ("CPU", "GPU", "FMM"):
        raise NotImplementedError(
            f'evaluator "{name}": skellysim_amd is the MI355X-native GPU backend; '
            'use "HIP". The reference CPU/FMM paths are out of scope '
            "(SURVEY.md §8) and the CPU restatement under oracle/ is test "
            "infrastructure only.")
    raise ValueError(f"unknown evaluator {name!r}")


def stresslet_times_normal_times_density(r_src, normals, density, reg=5e-3,
                                         epsilon_distance=1e-5):
    """Mirrors kernels::stresslet_times_normal_times_density
    (kernels.cpp:307-334; defaults kernels.hpp:50-51). Returns (n, 3);
    no eta dependence (factor -3/(4 pi))."""
    src = _rows(r_src, 3, "r_src")
    nrm = _rows(normals, 3, "normals")
    rho = _rows(density, 3, "density")
    out = np.zeros((len(src), 3))
    rc = _native.lib().skelly_stresslet_normal_density_host(
        _ptr(src), _ptr(nrm), _ptr(rho), _ptr(out), len(src),
        float(reg), float(epsilon_distance))
    _native.check(rc, "stresslet_normal_density")
    return out


# ---------------------------------------------------------------------------
# torch device-tensor API (async on the current torch stream)
# ---------------------------------------------------------------------------

def _dev_rows(t, dim, name):
    import torch
    if not (isinstance(t, torch.Tensor) and t.is_cuda):
        raise TypeError(f"{name}: expected a CUDA torch tensor (no CPU fallback)")
    if t.dtype != torch.float64:
        raise TypeError(f"{name}: expected float64, got {t.dtype}")
    if t.dim() != 2 or t.shape[1] != dim:
        raise ValueError(f"{name}: expected (n, {dim}), got {tuple(t.shape)}")
    return t.contiguous()


def _stream_ptr():
    import torch
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def stokeslet_device(r_src, f_src, r_trg, eta, out=None):
    """Stokeslet on device tensors (n,3); returns (n_trg,3) velocities
    (fully scaled 1/(8 pi eta)). Async on torch's current stream."""
    import torch
    r_src = _dev_rows(r_src, 3, "r_src")
    f_src = _dev_rows(f_src, 3, "f_src")
    r_trg = _dev_rows(r_trg, 3, "r_trg")
    if out is None:
        out = torch.empty_like(r_trg)
    rc = _native.lib().skelly_stokeslet_device(
        ctypes.c_void_p(r_src.data_ptr()), ctypes.c_void_p(f_src.data_ptr()), len(r_src),
        ctypes.c_void_p(r_trg.data_ptr()), ctypes.c_void_p(out.data_ptr()), len(r_trg),
        float(eta), _stream_ptr())
    _native.check(rc, "stokeslet_device")
    return out


def stresslet_device(r_src, f_src, r_trg, eta, out=None):
    import torch
    r_src = _dev_rows(r_src, 3, "r_src")
    f_src = _dev_rows(f_src, 9, "f_src")
    r_trg = _dev_rows(r_trg, 3, "r_trg")
    if out is None:
        out = torch.empty_like(r_trg)
    rc = _native.lib().skelly_stresslet_device(
        ctypes.c_void_p(r_src.data_ptr()), ctypes.c_void_p(f_src.data_ptr()), len(r_src),
        ctypes.c_void_p(r_trg.data_ptr()), ctypes.c_void_p(out.data_ptr()), len(r_trg),
        float(eta), _stream_ptr())
    _native.check(rc, "stresslet_device")
    return out


def oseen_contract_device(r_src, r_trg, density, eta=1.0, reg=5e-3, epsilon_distance=1e-5,
                          out=None):
    import torch
    r_src = _dev_rows(r_src, 3, "r_src")
    r_trg = _dev_rows(r_trg, 3, "r_trg")
    density = _dev_rows(density, 3, "density")
    if out is None:
        out = torch.empty_like(r_trg)
    rc = _native.lib().skelly_oseen_contract_device(
        ctypes.c_void_p(r_src.data_ptr()), ctypes.c_void_p(r_trg.data_ptr()),
        ctypes.c_void_p(density.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        len(r_src), len(r_trg), float(eta), float(reg), float(epsilon_distance), _stream_ptr())
    _native.check(rc, "oseen_contract_device")
    return out


def stresslet_normal_density_device(r_src, normals, density, r_trg=None, reg=5e-3,
                                    epsilon_distance=1e-5, out=None):
    """Device form of stresslet_times_normal_times_density; r_trg defaults to
    r_src (the reference's self form)."""
    import torch
    r_src = _dev_rows(r_src, 3, "r_src")
    normals = _dev_rows(normals, 3, "normals")
    density = _dev_rows(density, 3, "density")
    r_trg = r_src if r_trg is None else _dev_rows(r_trg, 3, "r_trg")
    nd = torch.cat([normals, density], dim=1).contiguous()
    if out is None:
        out = torch.empty_like(r_trg)
    rc = _native.lib().skelly_stresslet_normal_density_device(
        ctypes.c_void_p(r_src.data_ptr()), ctypes.c_void_p(nd.data_ptr()),
        ctypes.c_void_p(r_trg.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        len(r_src), len(r_trg), float(reg), float(epsilon_distance), _stream_ptr())
    _native.check(rc, "stresslet_normal_density_device")
    return out


def stresslet_times_normal_device(pts, normals, reg=5e-3, epsilon_distance=1e-5, out=None):
    """Dense stresslet_times_normal builder (kernels.cpp:264-287).
    pts, normals: (n, 3) CUDA fp64 -> (3n, 3n)."""
    import torch
    pts = _dev_rows(pts, 3, "pts")
    normals = _dev_rows(normals, 3, "normals")
    n = len(pts)
    if out is None:
        out = torch.empty((3 * n, 3 * n), dtype=torch.float64, device=pts.device)
    rc = _native.lib().skelly_stresslet_times_normal_device(
        ctypes.c_void_p(pts.data_ptr()), ctypes.c_void_p(normals.data_ptr()),
        ctypes.c_void_p(out.data_ptr()), n, float(reg), float(epsilon_distance), _stream_ptr())
    _native.check(rc, "stresslet_times_normal_device")
    return out


def oseen_tensor_batched_device(pts, eta=1.0, reg=5e-3, epsilon_distance=1e-5, out=None):
    """Batched dense self-Oseen-tensor builder (kernels.cpp:146-195; the
    per-fiber self-stokeslet, fiber_finite_difference.cpp:56).
    pts: (nf, n, 3) CUDA fp64 -> G (nf, 3n, 3n)."""
    import torch
    if not (isinstance(pts, torch.Tensor) and pts.is_cuda and pts.dtype == torch.float64
            and pts.dim() == 3 and pts.shape[2] == 3):
        raise TypeError("pts: expected (nf, n, 3) fp64 CUDA tensor")
    pts = pts.contiguous()
    nf, n = pts.shape[0], pts.shape[1]
    if out is None:
        out = torch.empty((nf, 3 * n, 3 * n), dtype=torch.float64, device=pts.device)
    rc = _native.lib().skelly_oseen_tensor_batched_device(
        ctypes.c_void_p(pts.data_ptr()), ctypes.c_void_p(out.data_ptr()), nf, n,
        float(eta), float(reg), float(epsilon_distance), _stream_ptr())
    _native.check(rc, "oseen_tensor_batched_device")
    return out


def rotlet_device(r_src, r_trg, density, eta=1.0, reg=5e-3, epsilon_distance=1e-5, out=None):
    import torch
    r_src = _dev_rows(r_src, 3, "r_src")
    r_trg = _dev_rows(r_trg, 3, "r_trg")
    density = _dev_rows(density, 3, "density")
    if out is None:
        out = torch.empty_like(r_trg)
    rc = _native.lib().skelly_rotlet_device(
        ctypes.c_void_p(r_src.data_ptr()), ctypes.c_void_p(r_trg.data_ptr()),
        ctypes.c_void_p(density.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        len(r_src), len(r_trg), float(eta), float(reg), float(epsilon_distance), _stream_ptr())
    _native.check(rc, "rotlet_device")
    return out
