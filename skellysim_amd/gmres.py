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

Distributed: vectors may be rank-local slices of a block-row-distributed
global vector (the reference's Tpetra map, solver_hydro.cpp:17-20); inner
products then all-reduce across ranks (Tpetra's distributed dots). The
matvec/precond callables own whatever gathers they need (e.g.
ShellOperator's source all-gather).

Small dense Hessenberg algebra (size ~ iteration count) runs on the host —
it is O(restart^2) scalars per iteration.
"""

import torch


def _make_dot(distributed, group=None):
    if not distributed:
        return lambda a, b: torch.dot(a, b)

    import torch.distributed as dist

    def dot(a, b):
        s = torch.dot(a, b)
        dist.all_reduce(s, group=group)
        return s

    return dot


def gmres(matvec, b, precond=None, tol=1e-10, maxiter=200, restart=30,
          x0=None, distributed=False, group=None, callback=None):
    """Solve A x = b with right-preconditioned GMRES(restart).

    matvec(v) -> A v ; precond(v) -> M^-1 v (right preconditioner: solves
    A M^-1 u = b, x = M^-1 u, as the reference does). Convergence: implicit
    residual ||r|| / ||b|| <= tol. Returns (x, info) with info = dict(
    converged, iters, residuals).
    """
    if precond is None:
        precond = lambda v: v
    dot = _make_dot(distributed, group)
    norm = lambda v: torch.sqrt(dot(v, v))

    b = b.reshape(-1)
    x = torch.zeros_like(b) if x0 is None else x0.clone().reshape(-1)
    bnorm = norm(b)
    if float(bnorm) == 0.0:
        return x, {"converged": True, "iters": 0, "residuals": [0.0]}

    residuals = []
    total_iters = 0
    converged = False

    while total_iters < maxiter and not converged:
        r = b - matvec(x)
        beta = norm(r)
        residuals.append(float(beta / bnorm))
        if residuals[-1] <= tol:
            converged = True
            break

        m = min(restart, maxiter - total_iters)
        V = [r / beta]
        H = torch.zeros((m + 1, m), dtype=torch.float64)
        g = torch.zeros(m + 1, dtype=torch.float64)
        g[0] = float(beta)
        cs = torch.zeros(m, dtype=torch.float64)
        sn = torch.zeros(m, dtype=torch.float64)
        k_done = 0

        for k in range(m):
            w = matvec(precond(V[k]))
            # ICGS: two classical Gram-Schmidt passes (Belos "ICGS",
            # solver_hydro.cpp:72)
            for _ in range(2):
                for j in range(k + 1):
                    hjk = dot(V[j], w)
                    H[j, k] += float(hjk)
                    w = w - hjk * V[j]
            hk1 = norm(w)
            H[k + 1, k] = float(hk1)

            # Givens rotations on the new column
            for j in range(k):
                t = cs[j] * H[j, k] + sn[j] * H[j + 1, k]
                H[j + 1, k] = -sn[j] * H[j, k] + cs[j] * H[j + 1, k]
                H[j, k] = t
            denom = torch.sqrt(H[k, k] ** 2 + H[k + 1, k] ** 2)
            if float(denom) == 0.0:
                k_done = k
                break
            cs[k] = H[k, k] / denom
            sn[k] = H[k + 1, k] / denom
            H[k, k] = denom
            H[k + 1, k] = 0.0
            g[k + 1] = -sn[k] * g[k]
            g[k] = cs[k] * g[k]

            total_iters += 1
            k_done = k + 1
            resid = float(abs(g[k + 1]) / bnorm)
            residuals.append(resid)
            if callback is not None:
                callback(total_iters, resid)
            if resid <= tol or float(hk1) == 0.0:
                converged = resid <= tol
                break
            V.append(w / hk1)

        if k_done > 0:
            y = torch.linalg.solve_triangular(H[:k_done, :k_done],
                                              g[:k_done].reshape(-1, 1),
                                              upper=True).reshape(-1)
            basis = torch.stack(V[:k_done], dim=1)  # (n_local, k_done)
            update = basis @ y.to(basis.device)
            x = x + precond(update)
        else:
            break

    # final check against the true residual
    if converged:
        true_resid = float(norm(b - matvec(x)) / bnorm)
        residuals.append(true_resid)
    return x, {"converged": converged, "iters": total_iters, "residuals": residuals}
