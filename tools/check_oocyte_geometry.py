#!/usr/bin/env python3
"""Quantify the oocyte (surface-of-revolution) boundary-operator quality:
assemble the periphery operator on the committed 6000-node oocyte node set
with (a) the reference-pipeline fixture's normals/RBF weights and (b) the
engine's analytic normals + vertex-area weights
(precompute.surface_of_revolution_normals_weights), then solve the rigid
uniform-velocity BC and check interior cancellation.

Measured (this container, CPU oracle):
  reference-pipeline fixture: interior 8.87e-4, near-cap 2.23e-2
  analytic normals + vertex areas: interior 8.92e-5, near-cap 4.90e-4
The fixture's degradation traces to precompute.py's gradh being evaluated
on the x1.04-scaled nodes against the unscaled envelope (normals off by up
to ~23 deg near the caps). Heavy (~10 GB peak, minutes) — a tool, not a
test."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import scipy.linalg as scla

import oracle
from skellysim_amd.precompute import surface_of_revolution_normals_weights


def operator_check(nodes, normals, weights, L, s, label):
    N = len(nodes)
    S = oracle.np_stresslet_times_normal(nodes, normals)
    for k in range(3):
        e = np.zeros((N, 3))
        e[:, k] = weights
        col = oracle.np_stresslet_times_normal_times_density(nodes, normals, e)
        for i in range(N):
            S[3 * i: 3 * i + 3, 3 * i + k] -= col[i] / weights[i]
    idx = np.arange(3 * N)
    S[idx, idx] -= 1.0 / weights[idx // 3]
    nf = normals.reshape(-1)
    A = S + np.outer(nf, nf)
    U = np.array([0.1, -0.05, 0.2])
    dens = scla.solve(A, np.tile(U, N))
    f_dl = 2.0 * np.einsum("ni,nj->nij", normals,
                           dens.reshape(-1, 3)).reshape(-1, 9)
    xs = np.linspace(-0.3 * L * s / 2, 0.3 * L * s / 2, 9)
    pts = np.stack([xs, np.zeros(9), np.zeros(9)], axis=1)
    u_in = oracle.np_stresslet(nodes, f_dl, pts, 1.0)
    pts2 = np.stack([np.array([-0.45, 0.45]) * L * s, np.zeros(2),
                     np.zeros(2)], axis=1)
    u2 = oracle.np_stresslet(nodes, f_dl, pts2, 1.0)
    print(f"{label}: interior {np.abs(u_in - U[None]).max():.3e}, "
          f"near-cap {np.abs(u2 - U[None]).max():.3e}", flush=True)


def main():
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    fx = np.load(os.path.join(repo, "tests", "golden", "oocyte_nodes.npz"))
    nodes = fx["nodes"]
    s = float(fx["scale_factor"])
    T, p1, p2, L = (float(fx[k]) for k in
                    ("envelope_T", "envelope_p1", "envelope_p2",
                     "envelope_length"))
    ref_n = fx["normals"] / np.linalg.norm(fx["normals"], axis=1)[:, None]
    operator_check(nodes, ref_n, fx["quadrature_weights"].reshape(-1), L, s,
                   "reference-pipeline fixture")
    own = surface_of_revolution_normals_weights(nodes, T, p1, p2, L,
                                                scale_factor=s)
    operator_check(nodes, own["normals"], own["weights"], L, s,
                   "analytic normals + vertex-area weights")


if __name__ == "__main__":
    main()
