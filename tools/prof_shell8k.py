#!/usr/bin/env python3
"""Config-4-scale periphery on GPU: assemble the 8192-node ellipsoid shell
operator on device, invert, GMRES-solve a uniform-background problem and
check interior cancellation. Prints timings + physics residual."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from skellysim_amd.periphery_precompute import assemble_shell_operator
from skellysim_amd.flows import ShellOperator, periphery_flow
from skellysim_amd.gmres import gmres


def main():
    fix = np.load(os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                               "tests", "golden", "ellipsoid_8192_nodes.npz"))
    dev = torch.device("cuda:0")
    nodes = torch.from_numpy(fix["nodes"]).to(dev)
    normals = torch.from_numpy(fix["normals"]).to(dev)
    w = torch.from_numpy(fix["quadrature_weights"]).to(dev)
    N = len(nodes)

    torch.cuda.synchronize()
    t0 = time.perf_counter()
    A, _ = assemble_shell_operator(nodes, normals, w, want_inverse=False)
    torch.cuda.synchronize()
    t_asm = time.perf_counter() - t0
    t0 = time.perf_counter()
    M_inv = torch.linalg.inv(A)
    torch.cuda.synchronize()
    t_inv = time.perf_counter() - t0
    print(f"N={N}: assemble {t_asm:.2f}s, invert (3N={3*N})^2 {t_inv:.2f}s")

    op = ShellOperator(M_inv, A)
    U = np.array([0.3, -0.2, 0.7])
    rhs = torch.from_numpy(-np.tile(U, N)).to(dev)
    v0 = torch.zeros_like(rhs)
    t0 = time.perf_counter()
    q, info = gmres(lambda x: op.matvec(x, v0), rhs, precond=op.apply_preconditioner,
                    tol=1e-10, maxiter=50, restart=30)
    torch.cuda.synchronize()
    print(f"gmres: iters={info['iters']} converged={info['converged']} "
          f"{time.perf_counter()-t0:.2f}s")

    # interior points well inside the ellipsoid
    rng = np.random.default_rng(5)
    a, b, c = float(fix["a"]), float(fix["b"]), float(fix["c"])
    pts = rng.uniform(-1, 1, (2000, 3)) * np.array([a, b, c])
    lvl = (pts[:, 0] / a) ** 2 + (pts[:, 1] / b) ** 2 + (pts[:, 2] / c) ** 2
    pts = pts[lvl < 0.5][:200]
    u = periphery_flow(nodes, normals, q.reshape(N, 3),
                       torch.from_numpy(pts).to(dev), 1.0)
    torch.cuda.synchronize()
    resid = np.abs(u.cpu().numpy() + U[None, :]).max()
    print(f"interior |U + D[q]| max over {len(pts)} pts = {resid:.3e}")


if __name__ == "__main__":
    main()
