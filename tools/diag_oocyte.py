#!/usr/bin/env python3
"""Diagnose the oocyte (surface-of-revolution) config-5 GMRES stall:
assemble the corrected-geometry shell operator ONCE, then sweep solver
parameters (dt, restart, tol) and fiber-placement styles, printing the
iteration count and residual tail for one timestep each.

Context: with the sphere periphery (radius 15.6, dilute) config-5 solves
in 71 iterations; with the reference's ACTUAL oocyte envelope (half-length
~3.9, dense fiber packing) the residual stalls near 1e-5 at tol=1e-10.
The reference's own oocyte example runs at gmres_tol=1e-8 (skelly_config
.py:421 — the Python default written into every generated toml) and
dt_initial = dt_max = 1e-2 with 3000 fibers at >= 0.1 separation
(examples/oocyte/gen_config.py)."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from skellysim_amd.fiber_fd import FiberFD
from skellysim_amd.system_fd import SystemFD, HipBackend, Shell
from skellysim_amd.periphery_precompute import assemble_shell_operator
from skellysim_amd.precompute import surface_of_revolution_normals_weights


def envelope_inside(fix, pts, margin=0.02):
    T, p1, p2, L = (float(fix[k]) for k in
                    ("envelope_T", "envelope_p1", "envelope_p2",
                     "envelope_length"))
    q = pts / float(fix["scale_factor"])
    xq = np.clip(2 * q[:, 0] / L, -0.999, 0.999)
    h = 0.5 * T * (1 + xq) ** p1 * (1 - xq) ** p2 * L
    return np.all(q[:, 1] ** 2 + q[:, 2] ** 2 < (h - margin) ** 2) \
        and np.all(np.abs(q[:, 0]) < L / 2 - 0.05)


def place_fibers(fix, normals, n_fibers, n_nodes, ds_min=0.0, length=1.0,
                 E=2.5e-3, seed=0, clearance=0.0):
    """Fibers seeded at shell nodes pointing inward; ds_min > 0 applies the
    reference's move_fibers_to_surface minimum separation (minus ends
    only). clearance > 0 additionally requires the WHOLE candidate fiber
    to stay that far from every accepted fiber's nodes — inward fibers
    from a concave surface FOCUS toward each other, and a near-contact
    pair (separation below the node spacing) mis-integrates the stokeslet
    quadrature into huge spurious velocities that reject timesteps at any
    dt (the reference's placement checks minus ends only,
    skelly_config.py:692-697, and suffers the same focusing)."""
    from scipy.spatial import cKDTree
    fibers = []
    placed = []
    cloud = None
    cloud_n = 0
    order = np.random.default_rng(seed).permutation(len(fix["nodes"]))
    for i in order:
        if len(fibers) == n_fibers:
            break
        p = fix["nodes"][i]
        if ds_min > 0 and placed and \
                np.min(np.linalg.norm(np.asarray(placed) - p, axis=1)) < ds_min:
            continue
        n = normals[i] / np.linalg.norm(normals[i])
        n = -n if np.dot(n, p) > 0 else n
        s = np.linspace(0.05, 0.05 + length, n_nodes)
        x = p[None, :] + s[:, None] * n[None, :]
        if not envelope_inside(fix, x):
            continue
        if clearance > 0 and fibers:
            if cloud is None or len(fibers) - cloud_n >= 64:
                cloud = cKDTree(np.concatenate([f.x.T for f in fibers]))
                cloud_n = len(fibers)
            d, _ = cloud.query(x, k=1)
            if np.min(d) < clearance:
                continue
            recent = [f.x.T for f in fibers[cloud_n:]]
            if recent:
                dr = np.linalg.norm(
                    np.concatenate(recent)[None, :, :] - x[:, None, :],
                    axis=2)
                if dr.min() < clearance:
                    continue
        placed.append(p)
        fibers.append(FiberFD(x, length=length, bending_rigidity=E, eta=1.0,
                              minus_clamped=True, force_scale=-0.05))
    return fibers


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=1)
    args = ap.parse_args()

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    fix = np.load(os.path.join(repo, "tests", "golden", "oocyte_nodes.npz"))
    dev = torch.device("cuda:0")

    own = surface_of_revolution_normals_weights(
        fix["nodes"], float(fix["envelope_T"]), float(fix["envelope_p1"]),
        float(fix["envelope_p2"]), float(fix["envelope_length"]),
        scale_factor=float(fix["scale_factor"]))
    normals, weights = own["normals"], own["weights"]

    t0 = time.perf_counter()
    A, M_inv = assemble_shell_operator(
        torch.from_numpy(fix["nodes"]).to(dev),
        torch.from_numpy(np.ascontiguousarray(normals)).to(dev),
        torch.from_numpy(np.ascontiguousarray(weights)).to(dev))
    torch.cuda.synchronize()
    print(f"shell operator ({len(fix['nodes'])} nodes): "
          f"{time.perf_counter()-t0:.1f}s", flush=True)
    shell = Shell(fix["nodes"], normals, A, M_inv)

    configs = [
        dict(label="baseline dt=.025 r=150 tol=1e-8", dt=0.025, restart=150,
             tol=1e-8, nf=4000, ds=0.0),
        dict(label="ref dt=.01 r=150 tol=1e-8", dt=0.01, restart=150,
             tol=1e-8, nf=4000, ds=0.0),
        dict(label="ref dt=.01 r=300 tol=1e-8", dt=0.01, restart=300,
             tol=1e-8, nf=4000, ds=0.0),
        dict(label="dt=.005 r=150 tol=1e-8", dt=0.005, restart=150,
             tol=1e-8, nf=4000, ds=0.0),
        dict(label="ref-placement 3000@0.1 dt=.01 r=300 tol=1e-8", dt=0.01,
             restart=300, tol=1e-8, nf=3000, ds=0.1),
    ]
    for cfg in configs:
        fibers = place_fibers(fix, normals, cfg["nf"], 32, ds_min=cfg["ds"])
        s = SystemFD(fibers, eta=1.0, dt=cfg["dt"], shell=shell,
                     backend=HipBackend())
        t0 = time.perf_counter()
        for k in range(args.steps):
            info = s.step(tol=cfg["tol"], maxiter=400, restart=cfg["restart"])
            r = info["residuals"]
            print(f"[{cfg['label']}] n_fib={len(fibers)} step {k}: "
                  f"iters={info['iters']} conv={info['converged']} "
                  f"resid[-3:]={[f'{x:.2e}' for x in r[-3:]]} "
                  f"true={info.get('true_residual')}", flush=True)
        print(f"[{cfg['label']}] {time.perf_counter()-t0:.1f}s "
              f"for {args.steps} step(s)", flush=True)
        del s
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
