#!/usr/bin/env python3
"""Oocyte-stall diagnosis, phase 2: separate PACKING DENSITY from
GEOMETRY QUALITY, and test eventual convergence.

Cases:
  control — same fiber count/length on a DENSE SPHERE of matching surface
            area (tests/golden/sphere_dense_6000.npz, radius 2.6, RBF
            quadrature — reference-grade geometry). If this stalls too,
            the stall is physics/preconditioning (dense hydrodynamic
            coupling), not the SOR quadrature.
  count   — oocyte with 1000/2000/4000 fibers: where does the stall set in?
  long    — oocyte 4000 fibers, restart 500, maxiter 2000: does GMRES
            eventually converge (slow cluster) or truly stagnate?
"""

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
from diag_oocyte import place_fibers


def build_shell(nodes, normals, weights, dev):
    t0 = time.perf_counter()
    A, M_inv = assemble_shell_operator(
        torch.from_numpy(np.ascontiguousarray(nodes)).to(dev),
        torch.from_numpy(np.ascontiguousarray(normals)).to(dev),
        torch.from_numpy(np.ascontiguousarray(weights)).to(dev))
    torch.cuda.synchronize()
    print(f"shell operator ({len(nodes)} nodes): "
          f"{time.perf_counter()-t0:.1f}s", flush=True)
    return Shell(nodes, normals, A, M_inv)


def sphere_fibers(fix, n_fibers, n_nodes=32, length=1.0, E=2.5e-3, seed=0):
    fibers = []
    order = np.random.default_rng(seed).permutation(len(fix["nodes"]))
    R = float(fix["radius"])
    for i in order:
        if len(fibers) == n_fibers:
            break
        p = fix["nodes"][i]
        n = fix["normals"][i] / np.linalg.norm(fix["normals"][i])
        n = -n if np.dot(n, p) > 0 else n
        s = np.linspace(0.05, 0.05 + length, n_nodes)
        x = p[None, :] + s[:, None] * n[None, :]
        if not np.all(np.linalg.norm(x, axis=1) < R - 0.02):
            continue
        fibers.append(FiberFD(x, length=length, bending_rigidity=E, eta=1.0,
                              minus_clamped=True, force_scale=-0.05))
    return fibers


def one_step(label, fibers, shell, dt=0.01, tol=1e-8, maxiter=400,
             restart=300):
    s = SystemFD(fibers, eta=1.0, dt=dt, shell=shell, backend=HipBackend())
    t0 = time.perf_counter()
    info = s.step(tol=tol, maxiter=maxiter, restart=restart)
    r = info["residuals"]
    print(f"[{label}] n_fib={len(fibers)}: iters={info['iters']} "
          f"conv={info['converged']} resid tail={[f'{x:.2e}' for x in r[-3:]]} "
          f"true={info.get('true_residual')} "
          f"({time.perf_counter()-t0:.1f}s)", flush=True)
    del s
    torch.cuda.empty_cache()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cases", default="control,count,long")
    args = ap.parse_args()
    cases = args.cases.split(",")

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    dev = torch.device("cuda:0")

    if "control" in cases:
        fx = np.load(os.path.join(repo, "tests", "golden",
                                  "sphere_dense_6000.npz"))
        shell = build_shell(fx["nodes"], fx["normals"],
                            fx["quadrature_weights"].reshape(-1), dev)
        for nf in (2000, 4000):
            one_step("dense-sphere", sphere_fibers(fx, nf), shell)
        del shell
        torch.cuda.empty_cache()

    oocyte_cases = {"count", "long", "refenv", "longrestart", "protocol",
                    "refproto", "sep1500"}
    if oocyte_cases & set(cases):
        fx = np.load(os.path.join(repo, "tests", "golden", "oocyte_nodes.npz"))
        own = surface_of_revolution_normals_weights(
            fx["nodes"], float(fx["envelope_T"]), float(fx["envelope_p1"]),
            float(fx["envelope_p2"]), float(fx["envelope_length"]),
            scale_factor=float(fx["scale_factor"]))
        shell = build_shell(fx["nodes"], own["normals"], own["weights"], dev)
        if "count" in cases:
            for nf in (1000, 2000):
                one_step("oocyte", place_fibers(fx, own["normals"], nf, 32),
                         shell)
        if "long" in cases:
            one_step("oocyte-long", place_fibers(fx, own["normals"], 4000, 32),
                     shell, maxiter=2000, restart=500)
        if "sep1500" in cases:
            # why does the reference-style 3000@0.1 placement fail where
            # 4000-at-nodes converges at full restart? residual trace:
            one_step("oocyte-3000@0.1-r1500",
                     place_fibers(fx, own["normals"], 3000, 32, ds_min=0.1),
                     shell, maxiter=1500, restart=1500)
        if "refenv" in cases:
            # the reference's EXACT solver envelope (Belos defaults:
            # restart=Num Blocks=300, maxiter=1000, tol from config=1e-8)
            # on the reference example's fiber protocol (3000 fibers,
            # >= 0.1 separation, dt=1e-2)
            one_step("oocyte-refenv",
                     place_fibers(fx, own["normals"], 3000, 32, ds_min=0.1),
                     shell, maxiter=1000, restart=300)
        if "longrestart" in cases:
            # restart stagnation test: no restart inside the budget
            one_step("oocyte-r1500",
                     place_fibers(fx, own["normals"], 4000, 32),
                     shell, maxiter=1500, restart=1500)
        if "protocol" in cases:
            # multi-step protocol with WARM STARTS and NO in-budget restart
            # (restart stagnation kills the dense-packing solve: restart
            # 300 stalls forever, restart >= ~800 converges in ~732 iters)
            os.environ["SKELLY_WARM_START"] = "1"
            s = SystemFD(place_fibers(fx, own["normals"], 4000, 32),
                         eta=1.0, dt=0.01, shell=shell, backend=HipBackend())
            for k in range(5):
                t0 = time.perf_counter()
                info = s.step(tol=1e-8, maxiter=1500, restart=1500)
                print(f"[oocyte-protocol-warm] step {k}: "
                      f"iters={info['iters']} conv={info['converged']} "
                      f"({time.perf_counter()-t0:.1f}s)", flush=True)
            os.environ.pop("SKELLY_WARM_START")
        if "refproto" in cases:
            # the reference example's OWN protocol: 3000 fibers at >= 0.1
            # separation (move_fibers_to_surface ds_min), dt 1e-2, the
            # reference ADAPTIVE loop (rejection + dt shrink on
            # non-convergence), engine solver envelope (full restart)
            s = SystemFD(place_fibers(fx, own["normals"], 3000, 32,
                                      ds_min=0.1),
                         eta=1.0, dt=0.01, shell=shell, backend=HipBackend())
            t0 = time.perf_counter()
            accepted = 0
            attempt = 0
            while s.time < 0.05 and attempt < 20:
                attempt += 1
                s.backup()
                ta = time.perf_counter()
                info = s.step(tol=1e-8, maxiter=1500, restart=1500)
                err = s.fiber_error()
                ok = info["converged"] and err <= 0.1
                print(f"[oocyte-refproto] attempt {attempt}: dt={s.dt:.5f} "
                      f"iters={info['iters']} conv={info['converged']} "
                      f"fiber_err={err:.3e} -> "
                      f"{'ACCEPT' if ok else 'REJECT'} "
                      f"({time.perf_counter()-ta:.1f}s)", flush=True)
                if ok:
                    accepted += 1
                    s.time += s.dt
                    if err <= 0.09:
                        s.dt = min(0.01, s.dt * 1.2)
                else:
                    s.dt *= 0.5
                    s.restore()
                    if s.dt < 1e-4:
                        print("[oocyte-refproto] dt underflow", flush=True)
                        break
            wall = time.perf_counter() - t0
            print(f"[oocyte-refproto] {accepted} accepted / {attempt} "
                  f"attempts in {wall:.1f}s", flush=True)


if __name__ == "__main__":
    main()
