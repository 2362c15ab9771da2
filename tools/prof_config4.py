#!/usr/bin/env python3
"""BASELINE config 4 at full shape on one MI355X: 512 fibers x 64 nodes,
minus-clamped at an 8192-node ellipsoidal periphery, full GMRES solve with
the HIP matvec. Prints setup/solve timings and solver stats.

Fiber placement is synthetic (inward-pointing from evenly sampled shell
nodes; the reference's generator seeds random insertions) — the SHAPE and
operator sizes match BASELINE config 4: solution size 512*4*64 + 3*8192 =
155,648."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from skellysim_amd.fiber_fd import FiberFD
from skellysim_amd.system_fd import SystemFD, HipBackend, Shell
from skellysim_amd.periphery_precompute import assemble_shell_operator


def main(n_fibers=512, n_nodes=64):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    fix = np.load(os.path.join(repo, "tests", "golden", "ellipsoid_8192_nodes.npz"))
    dev = torch.device("cuda:0")

    t0 = time.perf_counter()
    nodes_t = torch.from_numpy(fix["nodes"]).to(dev)
    normals_t = torch.from_numpy(fix["normals"]).to(dev)
    w_t = torch.from_numpy(fix["quadrature_weights"]).to(dev)
    A, M_inv = assemble_shell_operator(nodes_t, normals_t, w_t)
    torch.cuda.synchronize()
    t_shell = time.perf_counter() - t0
    shell = Shell(fix["nodes"], fix["normals"], A, M_inv)  # device-resident
    print(f"shell operator (8192 nodes): {t_shell:.2f}s (on-GPU assembly + inversion)")

    # fibers: length 1.0, rigidity 2.5e-3 (survey config-4 params), minus end
    # at the shell pointing inward, minus_clamped
    length, E = 1.0, 2.5e-3
    sel = np.linspace(0, len(fix["nodes"]) - 1, n_fibers).astype(int)
    fibers = []
    for i in sel:
        p = fix["nodes"][i]
        n = fix["normals"][i] / np.linalg.norm(fix["normals"][i])  # inward
        s = np.linspace(0.02, 0.02 + length, n_nodes)
        x = p[None, :] + s[:, None] * n[None, :]
        fibers.append(FiberFD(x, length=length, bending_rigidity=E, eta=1.0,
                              minus_clamped=True, force_scale=-0.05))

    backend = HipBackend()
    sys_ = SystemFD(fibers, eta=1.0, dt=0.025, shell=shell, backend=backend)
    print(f"solution size: {sys_.fiber_sol_size + sys_.shell_sol_size}")

    t0 = time.perf_counter()
    rhs = sys_.prep_state_for_solver()
    t_prep = time.perf_counter() - t0
    print(f"prep_state_for_solver: {t_prep:.2f}s (rhs norm {np.linalg.norm(rhs):.3e})")

    iters = []
    t0 = time.perf_counter()
    info = sys_.solve(tol=1e-10, maxiter=300, restart=150)
    t_solve = time.perf_counter() - t0
    print(f"solve: converged={info['converged']} iters={info['iters']} "
          f"{t_solve:.2f}s ({t_solve/max(1,info['iters'])*1e3:.0f} ms/iteration)")

    t0 = time.perf_counter()
    info2 = sys_.step(tol=1e-10, maxiter=300)
    t_step = time.perf_counter() - t0
    print(f"full timestep (prep+solve+adopt): {t_step:.2f}s, iters={info2['iters']}, "
          f"converged={info2['converged']}")
    xs = np.concatenate([f.x.reshape(-1) for f in fibers])
    print(f"fiber positions finite: {np.isfinite(xs).all()}; "
          f"max |x|: {np.abs(xs).max():.3f}")


if __name__ == "__main__":
    main()
