#!/usr/bin/env python3
"""MTOC/aster scenario at reference scale on one MI355X: a 600-node
spherical body (the reference's default body discretization,
skelly_config.py:737-738) with N fibers attached at its nucleation sites,
stepped through the coupled solve — the round-2 measurement target for the
body path (it currently runs on the host matvec loop; the pair-kernel legs
and the body's dense algebra are device work)."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from skellysim_amd.body import SphericalBody
from skellysim_amd.fiber_fd import FiberFD
from skellysim_amd.system_fd import SystemFD, HipBackend


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--fibers", type=int, default=50)
    ap.add_argument("--nodes", type=int, default=32)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--length", type=float, default=1.0)
    args = ap.parse_args()

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    g = np.load(os.path.join(repo, "tests", "golden", "body_sphere_600.npz"))
    sites = g["nucleation_sites"]
    n_fib = min(args.fibers, len(sites))
    body = SphericalBody(g["nodes"], g["normals"],
                         g["quadrature_weights"].reshape(-1),
                         float(g["radius"]), nucleation_sites_ref=sites)

    fibers = []
    s = np.linspace(0, args.length, args.nodes)
    for i in range(n_fib):
        site = body.nucleation_sites[i]
        u = site / np.linalg.norm(site)
        x = site[None, :] + s[:, None] * u[None, :]
        f = FiberFD(x, length=args.length, bending_rigidity=2.5e-3, eta=1.0,
                    minus_clamped=True, force_scale=-0.05)
        f.binding_site = (0, i)
        fibers.append(f)

    sys_ = SystemFD(fibers, eta=1.0, dt=0.025, bodies=[body],
                    backend=HipBackend())
    print(f"aster: {n_fib} fibers x {args.nodes} nodes on a "
          f"{body.n_nodes}-node body; solution size "
          f"{sys_.fiber_sol_size + sys_.body_sol_size}", flush=True)

    t0 = time.perf_counter()
    iters = []
    for _ in range(args.steps):
        info = sys_.step(tol=1e-10, maxiter=300, restart=150)
        assert info["converged"], info
        iters.append(info["iters"])
    wall = time.perf_counter() - t0
    print(f"{args.steps} steps in {wall:.2f}s = {wall/args.steps:.2f} s/step; "
          f"iters {iters}; body |U| {np.linalg.norm(body.velocity):.3e}, "
          f"|w| {np.linalg.norm(body.angular_velocity):.3e}", flush=True)


if __name__ == "__main__":
    main()
