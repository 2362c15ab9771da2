"""Right-preconditioned restarted GMRES with ICGS orthogonalization, on torch
tensors (CPU or GPU) — the solver seam of the hot path (SURVEY.md §8f
next-row 3).

Mirrors the reference solve configuration (src/core/solver_hydro.cpp:64-87):
Belos PseudoBlockGmres, RIGHT preconditioner (problem.setRightPrec, line 66),
"ICGS" orthogonalization (iterated classical Gram-Schmidt, two passes,
line 72), relative convergence tolerance params.gmres_tol (default 1e-10,
src/core/params.cpp:14). Operators are callables (the reference's
A_fiber_hydro::apply / P_inv_hydro::apply just forward to System::
apply_matvec / apply_preconditioner, solver_hydro.cpp:23-29,42-48).

The Krylov basis is one (restart+1, n) matrix — each basis vector a
contiguous row, so both ICGS GEMVs (h = V_k w; w -= V_k^T h) take rocBLAS's
coalesced paths — and each ICGS pass is one fused dot-block instead of O(k)
scalar ops (scalar torch CPU ops pay a fork-join on many-core hosts).

Distributed: vectors may be rank-local slices of a block-row-distributed
global vector (the reference's Tpetra map, solver_hydro.cpp:17-20); the
dot-block and norms then all-reduce across ranks (Tpetra's distributed
dots). The matvec/precond callables own whatever gathers they need.
"""

import numpy as np
import torch


def _make_reduce(distributed, group=None):
    if not distributed:
        return lambda t: t

    import torch.distributed as dist

    def allreduce(t):
        dist.all_reduce(t, group=group)
        return t

    return allreduce


def _arnoldi_body(matvec, precond, V, Hstage, idx_k, idx_k1):
    """One shape-invariant Arnoldi iteration against the FULL basis: rows
    of V beyond the current k are zero, so their ICGS contributions vanish
    exactly; idx_k/idx_k1 are 1-element device index tensors whose VALUES
    are read at replay time."""
    vk = V.index_select(0, idx_k).reshape(-1)
    w = matvec(precond(vk))
    h1 = V @ w
    w = w - V.T @ h1
    h2 = V @ w
    w = w - V.T @ h2
    hk1 = torch.sqrt(torch.dot(w, w))
    V.index_copy_(0, idx_k1, (w / hk1).unsqueeze(0))
    Hstage.index_copy_(0, idx_k,
                       torch.cat([h1 + h2, hk1.reshape(1)]).unsqueeze(0))


def _capture_arnoldi(matvec, precond, V, Hstage, idx_k, idx_k1):
    """Warm up (materializes rocBLAS/extension workspaces — including the
    persistent split-K buffers, which must pre-exist because capture may
    not allocate), then record one iteration as a hipGraph. The warmup
    corrupts V[1]/Hstage[0]; the CALLER must re-zero V[1:]/Hstage after."""
    from . import _native
    try:
        _native.lib().skelly_set_persistent_ws(1)
    except (RuntimeError, AttributeError):
        pass  # CPU-only environment; capture will fail loudly if reached
    g = torch.cuda.CUDAGraph()
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(2):
            _arnoldi_body(matvec, precond, V, Hstage, idx_k, idx_k1)
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    # capture on the SAME stream the warmup ran on: the extension's
    # persistent split-K workspace is cached per stream, so a different
    # capture stream would miss the cache and hipMalloc inside capture
    # (-> hipErrorStreamCaptureInvalidated)
    with torch.cuda.graph(g, stream=s):
        _arnoldi_body(matvec, precond, V, Hstage, idx_k, idx_k1)
    return g


def gmres(matvec, b, precond=None, tol=1e-10, maxiter=200, restart=30,
          x0=None, distributed=False, group=None, callback=None,
          sync_cadence=None, use_graph=None):
    """Solve A x = b with right-preconditioned GMRES(restart).

    matvec(v) -> A v ; precond(v) -> M^-1 v (right preconditioner: solves
    A M^-1 u = b, x = M^-1 u, as the reference does). Convergence: implicit
    residual ||r|| / ||b|| <= tol. Returns (x, info) with info = dict(
    converged, iters, residuals).

    sync_cadence: device->host scalar transfers (each a full-stream sync)
    are BATCHED every this many iterations — the basis build is fully
    device-side, so up to sync_cadence Arnoldi steps stay queued on the
    device stream while the host processes earlier Hessenberg columns.
    The host bookkeeping is identical for any cadence (deferred, not
    changed), and CPU tensors are bitwise-identical across cadences
    (tests/test_gmres.py). DEFAULT 8 (~20% faster per iteration at
    config-5 scale): safe since the round-2 experiment matrix
    (profiles/cadence_matrix_r02.md) isolated the deep-queue corruption
    that cadence>1 used to trigger to magma's torch.linalg.lu_solve in
    the preconditioner — with magma out of the iteration (batched.py
    SKELLY_LU_MODE=inv default: explicit probed inverse, one bmm per
    apply) deep cadences are stable; the trsm mode was measured bitwise
    equal to cadence 1 at config-5 scale. Override via the
    SKELLY_GMRES_SYNC_CADENCE env var.

    use_graph: capture ONE Arnoldi iteration (matvec∘precond + two-pass
    ICGS + normalize + Hessenberg staging) as a hipGraph and replay it
    per iteration. Shape-invariance comes from orthogonalizing against
    the FULL (m+1)-row basis every iteration (rows beyond the current k
    are zero, so the extra terms vanish exactly) with device-index
    tensors selecting the read/write rows — the cost is (m+1)·n GEMV
    traffic per iteration instead of (k+1)·n. MEASURED (round 2,
    tools/prof_iter.py at config-4 shape): 7.31 vs 6.78 ms/iteration —
    the sync-cadence-8 batching already keeps the host far enough ahead
    that launch overhead is hidden, so the graph only adds the padded
    GEMV traffic; it stays OPT-IN (SKELLY_HIPGRAPH=1; requires CUDA
    tensors, non-distributed), validated for correctness on GPU
    (tests/test_gpu_graph.py: same solution and iteration count as the
    eager path). Results are ULP-equivalent, not bitwise (GEMV reduction
    order differs with the padded rows).
    """
    if sync_cadence is None:
        import os
        sync_cadence = int(os.environ.get("SKELLY_GMRES_SYNC_CADENCE", "8"))
    if use_graph is None:
        import os
        use_graph = os.environ.get("SKELLY_HIPGRAPH", "0") == "1"
    use_graph = bool(use_graph) and b.is_cuda and not distributed
    if precond is None:
        precond = lambda v: v
    reduce_ = _make_reduce(distributed, group)

    def norm(v):
        return torch.sqrt(reduce_(torch.dot(v, v)))

    b = b.reshape(-1)
    n = b.shape[0]
    x = torch.zeros_like(b) if x0 is None else x0.clone().reshape(-1)
    bnorm = norm(b)
    if float(bnorm) == 0.0:
        return x, {"converged": True, "iters": 0, "residuals": [0.0]}

    residuals = []
    total_iters = 0
    converged = False
    true_resid = None

    gstate = None
    if use_graph:
        m_fix = min(restart, maxiter)
        V_g = torch.zeros((m_fix + 1, n), dtype=b.dtype, device=b.device)
        Hstage = torch.zeros((m_fix, m_fix + 2), dtype=b.dtype,
                             device=b.device)
        idx_k = torch.zeros(1, dtype=torch.long, device=b.device)
        idx_k1 = torch.ones(1, dtype=torch.long, device=b.device)

    while total_iters < maxiter and not converged:
        r = b - matvec(x)
        beta = norm(r)
        residuals.append(float(beta / bnorm))
        if residuals[-1] <= tol:
            converged = True
            true_resid = residuals[-1]  # this IS the true residual
            break

        m = min(restart, maxiter - total_iters)
        # Krylov basis stored (m+1, n): each basis vector is a CONTIGUOUS
        # row, so both ICGS GEMVs (V_k w and V_k^T h) hit rocBLAS's
        # coalesced paths — the (n, m+1) layout made V_k h a strided
        # column-slice gemv that dominated whole solves (53% of config-5
        # GPU time at 278 GB/s; profiles/components_r01.md)
        if use_graph:
            V = V_g
            V.zero_()
            V[0] = r / beta
            if gstate is None:
                idx_k.fill_(0)
                idx_k1.fill_(1)
                gstate = _capture_arnoldi(matvec, precond, V, Hstage,
                                          idx_k, idx_k1)
                V[1:].zero_()  # warmup corrupted V[1]
            Hstage.zero_()
        else:
            V = torch.zeros((m + 1, n), dtype=b.dtype, device=b.device)
            V[0] = r / beta
        # small dense Hessenberg/Givens state in numpy: scalar torch-CPU ops
        # cost a fork-join on many-core hosts
        H = np.zeros((m + 1, m))
        g = np.zeros(m + 1)
        g[0] = float(beta)
        cs = np.zeros(m)
        sn = np.zeros(m)
        k_done = 0

        pending = []  # deferred (device-side) Hessenberg columns
        stop = False
        for k in range(m):
            if use_graph:
                idx_k.fill_(k)
                idx_k1.fill_(k + 1)
                gstate.replay()
                pending.append(k)  # marker; columns live in Hstage
            else:
                w = matvec(precond(V[k]))
                # ICGS: two classical Gram-Schmidt passes (Belos "ICGS",
                # solver_hydro.cpp:72), each as one fused dot-block +
                # update. Scalars stay on device; host transfers
                # (full-stream syncs) are batched every sync_cadence
                # iterations.
                Vk = V[: k + 1]
                hcol_dev = None
                for _ in range(2):
                    h = reduce_(Vk @ w)
                    w = w - Vk.T @ h
                    hcol_dev = h if hcol_dev is None else hcol_dev + h
                hk1_dev = torch.sqrt(reduce_(torch.dot(w, w)))
                # device-side normalize without a host read; if hk1 == 0
                # the row is never consumed (processing below discards
                # past it)
                V[k + 1] = w / hk1_dev
                pending.append(torch.cat([hcol_dev.reshape(-1),
                                          hk1_dev.reshape(1)]))
            if len(pending) < max(1, sync_cadence) and k != m - 1:
                continue

            # flush: ONE host transfer, then sequential Givens bookkeeping
            # for the batched columns
            k0 = k + 1 - len(pending)
            if use_graph:
                rows = Hstage[k0: k + 1].cpu().numpy()
                scalars = np.concatenate(
                    [np.concatenate([rows[i][: k0 + i + 2 - 1],
                                     rows[i][m_fix + 1: m_fix + 2]])
                     for i in range(len(rows))])
            else:
                scalars = torch.cat(pending).cpu().numpy()
            off = 0
            for kk in range(k0, k + 1):
                col_scalars = scalars[off: off + kk + 2]
                off += kk + 2
                H[: kk + 1, kk] = col_scalars[:-1]
                hk1 = col_scalars[-1]
                H[kk + 1, kk] = hk1

                # Givens rotations on the new column
                if kk:
                    col = H[: kk + 1, kk]
                    for j in range(kk):
                        t = cs[j] * col[j] + sn[j] * col[j + 1]
                        col[j + 1] = -sn[j] * col[j] + cs[j] * col[j + 1]
                        col[j] = t
                denom = np.sqrt(H[kk, kk] ** 2 + H[kk + 1, kk] ** 2)
                if denom == 0.0 or not np.isfinite(denom):
                    stop = True
                    break
                cs[kk] = H[kk, kk] / denom
                sn[kk] = H[kk + 1, kk] / denom
                H[kk, kk] = denom
                H[kk + 1, kk] = 0.0
                g[kk + 1] = -sn[kk] * g[kk]
                g[kk] = cs[kk] * g[kk]

                total_iters += 1
                k_done = kk + 1
                resid = float(abs(g[kk + 1]) / bnorm)
                residuals.append(resid)
                if callback is not None:
                    callback(total_iters, resid)
                if resid <= tol or hk1 == 0.0:
                    converged = resid <= tol
                    stop = True
                    break
            pending = []
            if stop:
                break

        if k_done > 0:
            import scipy.linalg as _scla
            y = _scla.solve_triangular(H[:k_done, :k_done], g[:k_done])
            yt = torch.from_numpy(y).to(dtype=b.dtype, device=b.device)
            update = V[:k_done].T @ yt
            x = x + precond(update)
        else:
            break

        # verify against the TRUE residual whenever the implicit (Givens)
        # residual claims convergence: the two can diverge when the operator
        # pipeline misbehaves (measured on MI355X: ROCm 7.2 dense-linalg
        # calls occasionally return silently corrupt results — the same
        # failure robust_inv probes for). A failed verification DEMOTES the
        # claim and the solve CONTINUES from the current x with the
        # remaining iteration budget, which re-expands the Krylov space from
        # the true residual and recovers from a poisoned cycle.
        if converged:
            true_resid = float(norm(b - matvec(x)) / bnorm)
            residuals.append(true_resid)
            if true_resid > max(100.0 * tol, 1e-12):
                converged = False  # poisoned cycle: keep iterating

    return x, {"converged": converged, "iters": total_iters,
               "residuals": residuals,
               "true_residual": true_resid if converged else None}
