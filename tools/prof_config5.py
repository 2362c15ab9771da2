#!/usr/bin/env python3
"""BASELINE config-5 shape on one MI355X: ~4k fibers x 32 nodes,
minus-clamped inside a 6000-node spherical periphery (oocyte-scale), N
backward-Euler timesteps with the device-resident GMRES, plus a
velocity-field cross-check of the final state against the CPU oracle
(config 5's "velocity-field tolerance vs CPU").

The 8-GPU leg of config 5 is round-2 scope; this measures the single-GPU
timestep throughput of the full pipeline."""

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
from skellysim_amd.flows import velocity_at_targets


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--fibers", type=int, default=4000)
    ap.add_argument("--nodes", type=int, default=32)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--geometry", default="sphere", choices=["sphere", "oocyte"],
                    help="oocyte = the reference example's surface of revolution "
                         "(tests/golden/oocyte_nodes.npz)")
    ap.add_argument("--tol", type=float, default=1e-10)
    ap.add_argument("--maxiter", type=int, default=300)
    ap.add_argument("--restart", type=int, default=150)
    ap.add_argument("--warm", action="store_true",
                    help="warm-start each solve from the previous solution")
    ap.add_argument("--dt", type=float, default=0.025)
    args = ap.parse_args()
    if args.warm:
        os.environ["SKELLY_WARM_START"] = "1"

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    geom = "sphere_6000_nodes.npz" if args.geometry == "sphere" else "oocyte_nodes.npz"
    fix = np.load(os.path.join(repo, "tests", "golden", geom))
    dev = torch.device("cuda:0")

    shell_normals = fix["normals"]
    shell_weights = fix["quadrature_weights"]
    if args.geometry == "oocyte":
        # the reference-pipeline fixture's normals/weights degrade the
        # operator near the caps; use the engine's analytic SOR precompute
        # on the same nodes (tools/check_oocyte_geometry.py)
        from skellysim_amd.precompute import surface_of_revolution_normals_weights
        own = surface_of_revolution_normals_weights(
            fix["nodes"], float(fix["envelope_T"]), float(fix["envelope_p1"]),
            float(fix["envelope_p2"]), float(fix["envelope_length"]),
            scale_factor=float(fix["scale_factor"]))
        shell_normals, shell_weights = own["normals"], own["weights"]

    t0 = time.perf_counter()
    A, M_inv = assemble_shell_operator(torch.from_numpy(fix["nodes"]).to(dev),
                                       torch.from_numpy(np.ascontiguousarray(shell_normals)).to(dev),
                                       torch.from_numpy(np.ascontiguousarray(shell_weights)).to(dev))
    torch.cuda.synchronize()
    
    print(f"shell operator ({len(fix['nodes'])} nodes): "
          f"{time.perf_counter()-t0:.2f}s")
    shell = Shell(fix["nodes"], shell_normals, A, M_inv)

    length, E = 1.0, 2.5e-3

    def inside(pts):
        """All pts strictly inside the periphery (with margin)."""
        if args.geometry == "sphere":
            return np.all(np.linalg.norm(pts, axis=1) < float(fix["radius"]) - 0.02)
        T, p1, p2, L = (float(fix[k]) for k in
                        ("envelope_T", "envelope_p1", "envelope_p2", "envelope_length"))
        q = pts / float(fix["scale_factor"])  # envelope is pre-scale
        xq = np.clip(2 * q[:, 0] / L, -0.999, 0.999)
        h = 0.5 * T * (1 + xq) ** p1 * (1 - xq) ** p2 * L
        return np.all(q[:, 1] ** 2 + q[:, 2] ** 2 < (h - 0.02) ** 2) \
            and np.all(np.abs(q[:, 0]) < L / 2 - 0.05)

    fibers = []
    order = np.random.default_rng(0).permutation(len(fix["nodes"]))
    for i in order:
        if len(fibers) == args.fibers:
            break
        p = fix["nodes"][i]
        n = fix["normals"][i] / np.linalg.norm(fix["normals"][i])
        n = -n if np.dot(n, p) > 0 else n  # point inward
        s = np.linspace(0.05, 0.05 + length, args.nodes)
        x = p[None, :] + s[:, None] * n[None, :]
        if not inside(x):
            continue  # tip regions: an inward fiber would pierce the far wall
        fibers.append(FiberFD(x, length=length, bending_rigidity=E, eta=1.0,
                              minus_clamped=True, force_scale=-0.05))
    print(f"placed {len(fibers)} fibers")

    sys_ = SystemFD(fibers, eta=1.0, dt=args.dt, shell=shell, backend=HipBackend())
    print(f"solution size: {sys_.fiber_sol_size + sys_.shell_sol_size}")

    t0 = time.perf_counter()
    iters = []
    for k in range(args.steps):
        info = sys_.step(tol=args.tol, maxiter=args.maxiter,
                         restart=args.restart)
        iters.append(info["iters"])
        assert info["converged"], info
    dt_total = time.perf_counter() - t0
    print(f"{args.steps} timesteps: {dt_total:.2f}s = {dt_total/args.steps:.2f} s/step "
          f"({args.steps/dt_total:.3f} steps/s); GMRES iters per step: {iters}")

    # velocity-field cross-check vs the CPU oracle on the FINAL state
    import oracle
    rng = np.random.default_rng(2)
    if args.geometry == "sphere":
        pts = rng.uniform(-0.5, 0.5, (64, 3)) * float(fix["radius"])
    else:
        pts = 0.35 * fix["nodes"][rng.integers(0, len(fix["nodes"]), 64)]
    r_fib = sys_.fiber_nodes()
    w = np.concatenate([f.quadrature_weights() for f in fibers])
    fw = np.zeros_like(r_fib)
    off = 0
    for f in fibers:  # motor forces as the field source (force_scale * xs)
        fw[off: off + f.n_nodes] = (f.force_scale * f.xs).T
        off += f.n_nodes
    dens = sys_.solution[sys_.fiber_sol_size:].reshape(-1, 3)

    T = lambda a: torch.from_numpy(np.ascontiguousarray(a)).to(dev)
    u_gpu = velocity_at_targets(
        T(pts), 1.0,
        fiber=dict(r_src=T(r_fib), forces=T(fw), weights=T(w)),
        shell=dict(node_pos=T(fix["nodes"]),
                   node_normal=T(np.ascontiguousarray(shell_normals)),
                   density=T(dens)))
    torch.cuda.synchronize()
    f_dl = 2.0 * np.einsum("ni,nj->nij", shell_normals, dens).reshape(-1, 9)
    u_cpu = (oracle.stokeslet(r_fib, fw * w[:, None], pts, 1.0)
             + oracle.stresslet(fix["nodes"], f_dl, pts, 1.0))
    rel = np.linalg.norm(u_gpu.cpu().numpy() - u_cpu) / np.linalg.norm(u_cpu)
    print(f"velocity field at 64 interior targets: GPU vs CPU oracle rel = {rel:.3e}")


if __name__ == "__main__":
    main()
