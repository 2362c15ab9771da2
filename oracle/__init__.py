"""oracle — CPU parity oracle for the SkellySim hot-path pair kernels.

TEST INFRASTRUCTURE ONLY. Only tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg may import this package. The product path (skellysim_amd/)
must never route through it; skellysim_amd fails loudly when its HIP extension
is missing.

Two independent statements of the reference math live here:

  * C restatement (kernels_ref.c, compiled to liboracle_cpu.so) — follows the
    reference C++/CUDA sources line by line (citations in kernels_ref.c) with
    the reference's OpenMP static target-chunking. This is the oracle used by
    parity tests and the timed `cpu_baseline` (kind="port").
  * numpy restatement (this file) — a vectorized second statement of the same
    formulas, used to cross-check the C oracle and to pin it against the
    reference's own Python statement of the math
    (/root/reference/src/skelly_sim/kernels.py, importable in the build
    container with a numba shim; see make_golden.py).

Layouts match the reference Evaluator (include/kernels.hpp:14-15): col-major
3 x n (xyz contiguous per point); stresslet strengths 9 x n.
All functions here take/return numpy arrays shaped (n, 3) (or (n, 9)) in C
order, which is byte-identical to the reference's 3 x n col-major layout.
"""

import ctypes
import os
import subprocess

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "liboracle_cpu.so")
_lib = None


def _build_lib():
    subprocess.run(["make", "-C", _HERE], check=True, capture_output=True)


def lib():
    """Load (building if necessary) the C oracle library."""
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            _build_lib()
        _lib = ctypes.CDLL(_LIB_PATH)
        d = ctypes.POINTER(ctypes.c_double)
        _lib.oracle_stokeslet.argtypes = [d, d, ctypes.c_long, d, d, ctypes.c_long, ctypes.c_double]
        _lib.oracle_stresslet.argtypes = [d, d, ctypes.c_long, d, d, ctypes.c_long, ctypes.c_double]
        _lib.oracle_oseen_contract.argtypes = [d, d, d, d, ctypes.c_long, ctypes.c_long,
                                               ctypes.c_double, ctypes.c_double, ctypes.c_double]
        _lib.oracle_rotlet.argtypes = [d, d, d, d, ctypes.c_long, ctypes.c_long,
                                       ctypes.c_double, ctypes.c_double, ctypes.c_double]
        _lib.oracle_stresslet_times_normal_times_density.argtypes = [
            d, d, d, d, ctypes.c_long, ctypes.c_double, ctypes.c_double]
        _lib.oracle_oseen_tensor.argtypes = [d, d, ctypes.c_long, ctypes.c_double,
                                             ctypes.c_double, ctypes.c_double]
        _lib.oracle_num_threads.restype = ctypes.c_int
    return _lib


def _asbuf(a):
    a = np.ascontiguousarray(a, dtype=np.float64)
    return a, a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))


def num_threads():
    return lib().oracle_num_threads()


def stokeslet(r_src, f_src, r_trg, eta=1.0):
    """C-oracle Stokeslet (kernels.cu:62-76 + 1/(8 pi) + /eta). (n,3) arrays."""
    r_src, ps = _asbuf(r_src)
    f_src, pf = _asbuf(f_src)
    r_trg, pt = _asbuf(r_trg)
    u = np.zeros((len(r_trg), 3))
    _, pu = _asbuf(u)
    lib().oracle_stokeslet(ps, pf, len(r_src), pt, pu, len(r_trg), eta)
    return u


def stresslet(r_src, f_src, r_trg, eta=1.0):
    """C-oracle stresslet double-layer (kernels.cu:29-54). f_src is (n,9)."""
    r_src, ps = _asbuf(r_src)
    f_src, pf = _asbuf(f_src)
    r_trg, pt = _asbuf(r_trg)
    u = np.zeros((len(r_trg), 3))
    _, pu = _asbuf(u)
    lib().oracle_stresslet(ps, pf, len(r_src), pt, pu, len(r_trg), eta)
    return u


def oseen_contract(r_src, r_trg, density, eta=1.0, reg=5e-3, eps=1e-5):
    """C-oracle regularized Oseen contraction (kernels.cpp:85-131)."""
    r_src, ps = _asbuf(r_src)
    r_trg, pt = _asbuf(r_trg)
    density, pd = _asbuf(density)
    u = np.zeros((len(r_trg), 3))
    _, pu = _asbuf(u)
    lib().oracle_oseen_contract(ps, pt, pd, pu, len(r_src), len(r_trg), eta, reg, eps)
    return u


def rotlet(r_src, r_trg, density, eta=1.0, reg=5e-3, eps=1e-5):
    """C-oracle rotlet (kernels.cpp:206-242)."""
    r_src, ps = _asbuf(r_src)
    r_trg, pt = _asbuf(r_trg)
    density, pd = _asbuf(density)
    u = np.zeros((len(r_trg), 3))
    _, pu = _asbuf(u)
    lib().oracle_rotlet(ps, pt, pd, pu, len(r_src), len(r_trg), eta, reg, eps)
    return u


def stresslet_times_normal_times_density(r_src, normals, density, reg=5e-3, eps=1e-5):
    """C-oracle restatement of kernels.cpp:307-334 (no eta; factor -3/4pi)."""
    r_src, ps = _asbuf(r_src)
    normals, pn = _asbuf(normals)
    density, pd = _asbuf(density)
    out = np.zeros((len(r_src), 3))
    _, po = _asbuf(out)
    lib().oracle_stresslet_times_normal_times_density(ps, pn, pd, po, len(r_src), reg, eps)
    return out


def oseen_tensor(pts, eta=1.0, reg=5e-3, eps=1e-5):
    """C-oracle restatement of the square oseen_tensor_direct builder
    (kernels.cpp:146-195): (n,3) -> (3n, 3n)."""
    pts, pp = _asbuf(pts)
    n = len(pts)
    G = np.zeros((3 * n, 3 * n))
    _, pg = _asbuf(G)
    lib().oracle_oseen_tensor(pp, pg, n, eta, reg, eps)
    return G


# ---------------------------------------------------------------------------
# numpy restatements (second independent statement; cross-checks the C oracle)
# ---------------------------------------------------------------------------

def np_stokeslet(r_src, f_src, r_trg, eta=1.0):
    """u(t) = 1/(8 pi eta) sum_s (1/r)(f + rhat (f.rhat)), r = t - s, r=0 -> 0.
    Restates src/core/kernels.cu:62-76."""
    r_src = np.asarray(r_src, float)
    f_src = np.asarray(f_src, float)
    r_trg = np.asarray(r_trg, float)
    d = r_trg[:, None, :] - r_src[None, :, :]          # (T, S, 3)
    r2 = np.einsum("tsi,tsi->ts", d, d)
    with np.errstate(divide="ignore"):
        rinv = np.where(r2 == 0.0, 0.0, 1.0 / np.sqrt(r2))
    inner = np.einsum("si,tsi->ts", f_src, d) * rinv * rinv
    u = np.einsum("ts,si->ti", rinv, f_src) + np.einsum("ts,tsi->ti", rinv * inner, d)
    return u / (8.0 * np.pi) / eta


def np_stresslet(r_src, f_src, r_trg, eta=1.0):
    """u(t) = -3/(8 pi eta) sum_s (d^T S d)/r^5 d, d = t - s, r=0 -> 0.
    Restates src/core/kernels.cu:29-54; S is the (n,9) row-major 3x3 tensor."""
    r_src = np.asarray(r_src, float)
    S = np.asarray(f_src, float).reshape(-1, 3, 3)
    r_trg = np.asarray(r_trg, float)
    d = r_trg[:, None, :] - r_src[None, :, :]
    r2 = np.einsum("tsi,tsi->ts", d, d)
    with np.errstate(divide="ignore"):
        rinv5 = np.where(r2 == 0.0, 0.0, r2 ** -2.5)
    dSd = np.einsum("tsi,sij,tsj->ts", d, S, d)
    u = np.einsum("ts,tsi->ti", -3.0 * dSd * rinv5, d)
    return u / (8.0 * np.pi) / eta


def np_oseen_contract(r_src, r_trg, density, eta=1.0, reg=5e-3, eps=1e-5):
    """Regularized Oseen contraction; restates src/core/kernels.cpp:85-131:
    dr==0 skipped; dr>eps -> 1/dr forms; else 1/sqrt(dr^2+reg^2) forms."""
    r_src = np.asarray(r_src, float)
    r_trg = np.asarray(r_trg, float)
    rho = np.asarray(density, float)
    factor = 1.0 / (8.0 * np.pi * eta)
    d = r_src[None, :, :] - r_trg[:, None, :]          # src - trg (kernels.cpp:99-101)
    dr2 = np.einsum("tsi,tsi->ts", d, d)
    dr = np.sqrt(dr2)
    near = dr <= eps                                   # reference: NOT (dr > eps)
    denom = np.where(near, np.sqrt(dr2 + reg * reg), dr)
    with np.errstate(divide="ignore"):
        fr = np.where(dr == 0.0, 0.0, factor / denom)
        gr = np.where(dr == 0.0, 0.0, factor / denom ** 3)
    ddotrho = np.einsum("tsi,si->ts", d, rho)
    u = np.einsum("ts,si->ti", fr, rho) + np.einsum("ts,tsi->ti", gr * ddotrho, d)
    return u


def np_stresslet_times_normal_times_density(r, normals, density, reg=5e-3, eps=1e-5):
    """Restates kernels.cpp:307-334: Sdn_i = -3/(4 pi) sum_{j!=i}
    (d.rho_j)(d.n_j)/r^5 d, d = r_i - r_j, r<eps regularized."""
    r = np.asarray(r, float)
    n = len(r)
    d = r[:, None, :] - r[None, :, :]
    dr2 = np.einsum("ijk,ijk->ij", d, d)
    rn = np.sqrt(dr2)
    rn = np.where(rn < eps, np.sqrt(dr2 + reg * reg), rn)
    with np.errstate(divide="ignore"):
        rinv5 = 1.0 / rn ** 5
    np.fill_diagonal(rinv5, 0.0)  # i == j skip
    f0 = (np.einsum("ijk,jk->ij", d, density) * np.einsum("ijk,jk->ij", d, normals) * rinv5)
    return -3.0 / (4.0 * np.pi) * np.einsum("ij,ijk->ik", f0, d)


def np_oseen_tensor(pts, eta=1.0, reg=5e-3, eps=1e-5):
    """Restates kernels.cpp:146-195 (square self form)."""
    pts = np.asarray(pts, float)
    n = len(pts)
    factor = 1.0 / (8.0 * np.pi * eta)
    d = pts[None, :, :] - pts[:, None, :]     # d[t, s] = src - trg
    dr2 = np.einsum("tsk,tsk->ts", d, d)
    dr = np.sqrt(dr2)
    near = dr <= eps
    denom = np.where(near, np.sqrt(dr2 + reg * reg), dr)
    with np.errstate(divide="ignore"):
        fr = np.where(dr2 == 0.0, 0.0, factor / denom)
        gr = np.where(dr2 == 0.0, 0.0, factor / denom ** 3)
    G = gr[:, :, None, None] * np.einsum("tsa,tsb->tsab", d, d)
    G += fr[:, :, None, None] * np.eye(3)[None, None]
    G[dr2 == 0.0] = 0.0
    return G.transpose(0, 2, 1, 3).reshape(3 * n, 3 * n)


def np_stresslet_times_normal(r, normals, reg=5e-3, eps=1e-5):
    """Restates kernels.cpp:264-287: Snormal block (i,j) =
    -3/(4 pi) (d.n_j)/r^5 d d^T, d = r_i - r_j; diagonal blocks zero."""
    r = np.asarray(r, float)
    n = len(r)
    d = r[:, None, :] - r[None, :, :]
    dr2 = np.einsum("ijk,ijk->ij", d, d)
    rn = np.sqrt(dr2)
    rn = np.where(rn < eps, np.sqrt(dr2 + reg * reg), rn)
    with np.errstate(divide="ignore"):
        rinv5 = 1.0 / rn ** 5
    np.fill_diagonal(rinv5, 0.0)
    c = -3.0 / (4.0 * np.pi) * np.einsum("ijk,jk->ij", d, normals) * rinv5
    S = c[:, :, None, None] * np.einsum("ija,ijb->ijab", d, d)
    return S.transpose(0, 2, 1, 3).reshape(3 * n, 3 * n)


def np_rotlet(r_src, r_trg, density, eta=1.0, reg=5e-3, eps=1e-5):
    """Rotlet; restates src/core/kernels.cpp:206-242 (no dr==0 skip;
    dr2 < eps^2 regularized)."""
    r_src = np.asarray(r_src, float)
    r_trg = np.asarray(r_trg, float)
    rho = np.asarray(density, float)
    factor = 1.0 / (8.0 * np.pi * eta)
    d = r_trg[:, None, :] - r_src[None, :, :]          # trg - src (kernels.cpp:220-222)
    dr2 = np.einsum("tsi,tsi->ts", d, d)
    dr = np.where(dr2 < eps * eps, np.sqrt(reg * reg + dr2), np.sqrt(dr2))
    fr = 1.0 / dr ** 3
    # u_x += fr*(dz*rho_y - dy*rho_z), etc. (kernels.cpp:228-236)
    cross = np.empty_like(d)
    cross[..., 0] = d[..., 2] * rho[None, :, 1] - d[..., 1] * rho[None, :, 2]
    cross[..., 1] = -d[..., 2] * rho[None, :, 0] + d[..., 0] * rho[None, :, 2]
    cross[..., 2] = d[..., 1] * rho[None, :, 0] - d[..., 0] * rho[None, :, 1]
    return factor * np.einsum("ts,tsi->ti", fr, cross)
