#!/usr/bin/env python3
"""Per-iteration cost breakdown of the device-resident GMRES at config-4
shape (512 fibers x 64 nodes + 8k-node shell, ~139k unknowns) and config-5
shape (4000 x 32 + 6000-node shell, 530k): time the matvec apply, the
preconditioner apply, and the full solve; the residual component is ICGS +
Python/launch overhead — the hipGraph-capture candidate (VERDICT r1
next-step 7)."""

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


def t_ms(fn, iters=50, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def build(nf, n, shell_fix):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    fix = np.load(os.path.join(repo, "tests", "golden", shell_fix))
    dev = torch.device("cuda:0")
    A, M_inv = assemble_shell_operator(
        torch.from_numpy(fix["nodes"]).to(dev),
        torch.from_numpy(fix["normals"]).to(dev),
        torch.from_numpy(fix["quadrature_weights"]).to(dev))
    torch.cuda.synchronize()
    shell = Shell(fix["nodes"], fix["normals"], A, M_inv)
    if "radius" in fix:
        R = float(fix["radius"])
        inside = lambda x: np.all(np.linalg.norm(x, axis=1) < R - 0.02)
    else:
        abc = np.array([float(fix["a"]), float(fix["b"]), float(fix["c"])])
        inside = lambda x: np.all(((x / (abc - 0.02)) ** 2).sum(axis=1) < 1.0)
    fibers = []
    order = np.random.default_rng(0).permutation(len(fix["nodes"]))
    length = 1.0
    for i in order:
        if len(fibers) == nf:
            break
        p = fix["nodes"][i]
        nrm = fix["normals"][i] / np.linalg.norm(fix["normals"][i])
        nrm = -nrm if np.dot(nrm, p) > 0 else nrm
        s = np.linspace(0.05, 0.05 + length, n)
        x = p[None, :] + s[:, None] * nrm[None, :]
        if not inside(x):
            continue
        fibers.append(FiberFD(x, length=length, bending_rigidity=2.5e-3,
                              eta=1.0, minus_clamped=True, force_scale=-0.05))
    return SystemFD(fibers, eta=1.0, dt=0.025, shell=shell,
                    backend=HipBackend())


def profile(label, s, restart=150):
    rhs = s.prep_state_for_solver()
    s._build_device_operators()
    b = s.backend._t(rhs)
    x = torch.randn_like(b)

    mv = t_ms(lambda: s._apply_matvec_device(x))
    pc = t_ms(lambda: s._apply_precond_device(x))
    # preconditioner sub-parts
    d = s._dev
    nf, n = d["nf"], d["n"]
    x_fib = x[: 4 * nf * n].reshape(nf, 4 * n).contiguous()
    fib_ms = t_ms(lambda: d["lu"].solve(x_fib))
    if s.shell:
        sh = x[4 * nf * n: 4 * nf * n + s.shell_sol_size].contiguous()
        shell_ms = t_ms(lambda: torch.mv(d["sh_Minv"], sh))
        shell_t_ms = t_ms(lambda: torch.mv(d["sh_Minv"].t(), sh))
    else:
        shell_ms = shell_t_ms = 0.0
    print(f"  precond parts: fiber-solve {fib_ms:.2f} ms, "
          f"shell M_inv mv {shell_ms:.2f} ms (transposed-layout mv "
          f"{shell_t_ms:.2f} ms)", flush=True)

    t0 = time.perf_counter()
    info = s.solve(tol=1e-10, maxiter=300, restart=restart)
    solve_s = time.perf_counter() - t0
    iters = max(1, info["iters"])
    per = solve_s / iters * 1e3
    print(f"[{label}] n={len(b)}: matvec {mv:.2f} ms, precond {pc:.2f} ms, "
          f"solve {solve_s:.2f}s / {iters} iters = {per:.2f} ms/iter "
          f"-> ICGS+overhead ~{per - mv - pc:.2f} ms/iter "
          f"(conv={info['converged']})", flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="4", choices=["4", "5", "both"])
    args = ap.parse_args()
    if args.config in ("4", "both"):
        profile("config4", build(512, 64, "ellipsoid_8192_nodes.npz"))
    if args.config in ("5", "both"):
        profile("config5", build(4000, 32, "sphere_6000_nodes.npz"))


if __name__ == "__main__":
    main()
