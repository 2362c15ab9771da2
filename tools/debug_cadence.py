#!/usr/bin/env python3
"""Diagnose the config-5 convergence bimodality: run the same 2-step
config-5-shaped problem with GMRES sync_cadence 1 and 8 side by side,
comparing solutions bitwise after every step."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from skellysim_amd.fiber_fd import FiberFD
from skellysim_amd.system_fd import SystemFD, HipBackend, Shell
from skellysim_amd.periphery_precompute import assemble_shell_operator


def build(nf=4000, n=32):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    fix = np.load(os.path.join(repo, "tests", "golden", "sphere_6000_nodes.npz"))
    dev = torch.device("cuda:0")
    A, M_inv = assemble_shell_operator(
        torch.from_numpy(fix["nodes"]).to(dev),
        torch.from_numpy(fix["normals"]).to(dev),
        torch.from_numpy(fix["quadrature_weights"]).to(dev))
    torch.cuda.synchronize()
    shell = Shell(fix["nodes"], fix["normals"], A, M_inv)
    length, E = 1.0, 2.5e-3
    fibers = []
    order = np.random.default_rng(0).permutation(len(fix["nodes"]))
    for i in order:
        if len(fibers) == nf:
            break
        p = fix["nodes"][i]
        nrm = fix["normals"][i] / np.linalg.norm(fix["normals"][i])
        nrm = -nrm if np.dot(nrm, p) > 0 else nrm
        s = np.linspace(0.05, 0.05 + length, n)
        x = p[None, :] + s[:, None] * nrm[None, :]
        if not np.all(np.linalg.norm(x, axis=1) < float(fix["radius"]) - 0.02):
            continue
        fibers.append(FiberFD(x, length=length, bending_rigidity=E, eta=1.0,
                              minus_clamped=True, force_scale=-0.05))
    return SystemFD(fibers, eta=1.0, dt=0.025, shell=shell, backend=HipBackend())


def main():
    sols = {}
    for cad in (1, 8):
        os.environ["SKELLY_GMRES_SYNC_CADENCE"] = str(cad)
        s = build()
        sols[cad] = []
        for k in range(2):
            info = s.step(tol=1e-10, maxiter=300, restart=150)
            print(f"cadence={cad} step={k}: iters={info['iters']} "
                  f"converged={info['converged']} "
                  f"true_resid={info.get('true_residual')}", flush=True)
            sols[cad].append(s.solution.copy())
        del s
        torch.cuda.empty_cache()
    for k in range(2):
        a, b = sols[1][k], sols[8][k]
        same = np.array_equal(a, b)
        rel = np.linalg.norm(a - b) / max(np.linalg.norm(b), 1e-300)
        print(f"step {k}: cadence1 vs cadence8 bitwise={same} rel={rel:.3e}",
              flush=True)


if __name__ == "__main__":
    main()
