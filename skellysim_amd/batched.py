"""Batched per-fiber dense algebra on device.

The reference preconditioner applies one LU solve per fiber per GMRES
iteration on the host (A_LU_.solve, fiber_container_finite_difference.cpp:
331-339, factorized once per timestep at fiber_finite_difference.cpp:340),
and the self-interaction correction is one small dense GEMV per fiber
(f_c_fd.cpp:203-210 — covered by flows.fiber_flow). Thousands of independent
4n x 4n (n <= 64) problems are the ideal batched-GPU shape: here they map to
rocSOLVER batched getrf/getrs via torch.linalg.lu_factor / lu_solve with all
factors resident in HBM across the solve.
"""

import numpy as np
import torch

_magma_latched = False


def _lu_factor(mats):
    """torch.linalg.lu_factor with a magma fallback: ROCm 7.2's
    hipblasDgetrfBatched fails with HIPBLAS_STATUS_ALLOC_FAILED for m > 128
    (measured on MI355X, profiles/components_r01.md); magma handles all the
    fiber sizes (4n up to 256) at per-timestep-negligible cost."""
    global _magma_latched
    if not mats.is_cuda:
        return torch.linalg.lu_factor(mats)
    try:
        return torch.linalg.lu_factor(mats)
    except RuntimeError as e:
        if "HIPBLAS" not in str(e) or _magma_latched:
            raise
        torch.backends.cuda.preferred_linalg_library("magma")
        _magma_latched = True
        return torch.linalg.lu_factor(mats)


_huge_inv_done = False


def _host_inv(A):
    import scipy.linalg as scla
    return torch.from_numpy(scla.inv(A.cpu().numpy())).to(A.device)


def _inv_probe_ok(A, X, rtol=1e-6):
    """Cheap correctness probe for an inverse: A (X v) ~= v for a fixed
    probe vector (two GEMVs). ROCm 7.2's hipBLAS trsm at ~18000^2 can
    SILENTLY return corrupt results on some nodes (the same failure that
    elsewhere raises ALLOC_FAILED or memory-faults) — a wrong M_inv then
    poisons entire solves, so every inverse is verified before use."""
    import math
    n = A.shape[-1]
    v = torch.linspace(-1.0, 1.0, n, dtype=A.dtype, device=A.device)
    r = A @ (X @ v) - v
    rel = float(torch.linalg.vector_norm(r) / torch.linalg.vector_norm(v))
    return math.isfinite(rel) and rel < rtol


def robust_inv(A):
    """torch.linalg.inv hardened against ROCm 7.2 linalg failure modes
    measured on MI355X (all at fp64):
      - hipBLAS (default backend) inverts 24576^2 fine in a fresh process
        but ABORTS (uncatchable) on a repeat in the same process, and hits
        a catchable HIPBLAS_STATUS_ALLOC_FAILED at 18000^2;
      - magma inverts 18000^2 fine but SEGFAULTS at 24576^2.
    Strategy: release cached blocks first; for huge matrices (>= 20000)
    force the default backend, allow one in-process GPU inversion and route
    repeats through host LAPACK (~15 s once per geometry — setup-time);
    below that, default backend with the magma fallback on the catchable
    ALLOC failure."""
    global _magma_latched, _huge_inv_done
    if not A.is_cuda:
        return torch.linalg.inv(A)
    n = A.shape[-1]
    torch.cuda.empty_cache()
    if n >= 20000:
        if _huge_inv_done:
            return _host_inv(A)
        if _magma_latched:
            torch.backends.cuda.preferred_linalg_library("default")
        try:
            out = torch.linalg.inv(A)
            if not _inv_probe_ok(A, out):
                out = _host_inv(A)
        except RuntimeError:
            out = _host_inv(A)
        finally:
            if _magma_latched:
                torch.backends.cuda.preferred_linalg_library("magma")
        _huge_inv_done = True
        return out
    try:
        out = torch.linalg.inv(A)
        if _inv_probe_ok(A, out):
            return out
        import sys
        print(f"robust_inv: probe failed for {n}x{n} inverse on the default "
              "backend (silent hipBLAS corruption); retrying via magma",
              file=sys.stderr, flush=True)
    except RuntimeError as e:
        if "HIPBLAS" not in str(e) or _magma_latched:
            raise
    if not _magma_latched:
        torch.backends.cuda.preferred_linalg_library("magma")
        _magma_latched = True
        out = torch.linalg.inv(A)
        if _inv_probe_ok(A, out):
            return out
    return _host_inv(A)


class BatchedLU:
    """Factor once per timestep, solve per GMRES iteration.

    mats: (n_fibers, m, m) fp64 tensor (uniform fiber discretization — the
    reference's per-fiber A_ are all 4n x 4n for the configured n_nodes).
    """

    def __init__(self, mats):
        if mats.dim() != 3 or mats.shape[1] != mats.shape[2]:
            raise ValueError(f"expected (batch, m, m), got {tuple(mats.shape)}")
        if mats.dtype != torch.float64:
            raise TypeError("BatchedLU expects fp64")
        self._host_lus = None
        self._Ainv = None
        if mats.is_cuda and self._mode() == "inv":
            # explicit batched inverse, applied per iteration as ONE
            # rocBLAS batched GEMM (torch.bmm) — no magma and no
            # triangular-solve kernels inside the GMRES iteration at all.
            # Safe for (right-)preconditioning: M^-1 is a FIXED linear
            # operator whatever its rounding, GMRES converges on the true
            # residual regardless; the factor-time inversion happens once
            # per timestep under a shallow queue (prep syncs), where the
            # deep-queue magma hazard (profiles/cadence_matrix_r02.md)
            # does not apply. Probe-verified like every inverse here.
            # force the default (hipSOLVER) backend for the batched
            # inversion: once robust_inv's 18000^2 probe failure has latched
            # magma globally, magma's getri LOOPS over the 4000-matrix batch
            # (~16 s/step measured at config 5 vs 8 ms batched —
            # profiles/cadence_matrix_r02.md addendum)
            global _magma_latched
            if _magma_latched:
                torch.backends.cuda.preferred_linalg_library("default")
            try:
                self._Ainv = torch.linalg.inv(mats)
            except RuntimeError:
                self._Ainv = None
            finally:
                if _magma_latched:
                    torch.backends.cuda.preferred_linalg_library("magma")
            # probe at 1e-4: the probe exists to catch CORRUPTION (O(1)
            # garbage from the hipBLAS/hipSOLVER failure modes), not the
            # cond*eps forward error of a legitimate explicit inverse —
            # fiber operators at 4n=256 carry cond ~1e10 and a correct
            # inverse legitimately probes ~1e-6..1e-5 (which as a fixed
            # right-preconditioner costs nothing: GMRES converges on the
            # true residual regardless)
            if self._Ainv is not None and self._probe_ok(mats, rtol=1e-4):
                return
            self._Ainv = None  # corrupt or failed: fall through to LU
        self.LU, self.pivots = _lu_factor(mats)
        if mats.is_cuda and self._mode() == "inv":
            # inverse-from-LU: one shallow-queue magma lu_solve against I
            # per timestep (safe — the deep-queue hazard is per-ITERATION
            # calls), then bmm per iteration like the direct-inv path
            m = mats.shape[-1]
            eye = torch.eye(m, dtype=mats.dtype, device=mats.device) \
                .expand_as(mats).contiguous()
            try:
                self._Ainv = torch.linalg.lu_solve(self.LU, self.pivots, eye)
                torch.cuda.synchronize()
            except RuntimeError:
                self._Ainv = None
            if self._Ainv is not None and self._probe_ok(mats, rtol=1e-4):
                return
            self._Ainv = None
            import sys
            print("BatchedLU: explicit-inverse probes failed (direct and "
                  "from-LU); falling back to per-iteration triangular "
                  "solves", file=sys.stderr, flush=True)
        if mats.is_cuda and not self._probe_ok(mats):
            # same defensive posture as robust_inv: a silently corrupt
            # device factorization must not poison solves
            import sys
            print("BatchedLU: factor probe failed on the device backend; "
                  "falling back to host LAPACK factors", file=sys.stderr,
                  flush=True)
            import scipy.linalg as scla
            m_np = mats.cpu().numpy()
            self._host_lus = [scla.lu_factor(a) for a in m_np]
            self._dev = mats.device

    def _probe_ok(self, mats, rtol=1e-6):
        import math
        m = mats.shape[-1]
        v = torch.linspace(1.0, 2.0, m, dtype=mats.dtype, device=mats.device)
        rhs = mats @ v
        x = self.solve(rhs)
        rel = float(torch.linalg.vector_norm(x - v) /
                    torch.linalg.vector_norm(v) / max(1, mats.shape[0]) ** 0.5)
        return math.isfinite(rel) and rel < rtol

    @staticmethod
    def _mode():
        """Per-iteration solve strategy (SKELLY_LU_MODE):
          "inv"   (default) — explicit batched inverse applied with one
                  torch.bmm per apply; fastest and keeps magma AND
                  triangular kernels out of the GMRES iteration entirely.
          "trsm"  — batched solve_triangular (correct under deep queues,
                  measured bitwise-stable at cadence 8).
          "magma" — torch.linalg.lu_solve: CORRUPTS under deep stream
                  queues (sync cadence > 1) — magma's batched trsv builds
                  per-call device pointer arrays, a use-after-free that
                  the round-2 experiment matrix isolated
                  (profiles/cadence_matrix_r02.md). Keep only for
                  retesting future ROCm/magma.
        The legacy SKELLY_LU_TRSM=1/0 env maps to trsm/magma when
        SKELLY_LU_MODE is unset."""
        import os
        mode = os.environ.get("SKELLY_LU_MODE")
        if mode in ("inv", "trsm", "magma"):
            return mode
        legacy = os.environ.get("SKELLY_LU_TRSM")
        if legacy == "1":
            return "trsm"
        if legacy == "0":
            return "magma"
        return "inv"

    @classmethod
    def _use_trsm(cls):
        return cls._mode() != "magma"

    def _perm_from_pivots(self):
        if getattr(self, "_perm", None) is None:
            piv = self.pivots.cpu().numpy() - 1      # LAPACK 1-based
            nb, m = piv.shape
            perm = np.tile(np.arange(m), (nb, 1))
            for b in range(nb):
                for i in range(m):
                    j = piv[b, i]
                    perm[b, [i, j]] = perm[b, [j, i]]
            self._perm = torch.from_numpy(perm).to(self.LU.device)
        return self._perm

    def solve(self, rhs):
        """rhs: (n_fibers, m) or (n_fibers, m, k) -> same shape solution."""
        if self._Ainv is not None:
            if rhs.dim() == 2:
                return torch.bmm(self._Ainv, rhs.unsqueeze(-1)).squeeze(-1)
            return torch.bmm(self._Ainv, rhs)
        if self._host_lus is not None:
            import numpy as _np
            import scipy.linalg as scla
            r = rhs.cpu().numpy()
            out = _np.stack([scla.lu_solve(lu, b)
                             for lu, b in zip(self._host_lus, r)])
            return torch.from_numpy(out).to(self._dev)
        squeeze = rhs.dim() == 2
        if squeeze:
            rhs = rhs.unsqueeze(-1)
        if self._use_trsm():
            perm = self._perm_from_pivots()          # (nb, m)
            b = torch.gather(rhs, 1,
                             perm.unsqueeze(-1).expand(-1, -1, rhs.shape[-1]))
            y = torch.linalg.solve_triangular(self.LU, b, upper=False,
                                              unitriangular=True)
            x = torch.linalg.solve_triangular(self.LU, y, upper=True)
        else:
            x = torch.linalg.lu_solve(self.LU, self.pivots, rhs)
        return x.squeeze(-1) if squeeze else x


def batched_matvec(mats, vecs):
    """(n_fibers, m, m) @ (n_fibers, m) -> (n_fibers, m): the per-fiber
    dense matvecs (A_ @ x, force_operator_ @ x) as one rocBLAS batched GEMV."""
    return torch.bmm(mats, vecs.unsqueeze(-1)).squeeze(-1)
