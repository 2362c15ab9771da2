#!/usr/bin/env python3
"""Max elementwise relative error of the HIP kernels vs the CPU oracle —
used to qualify rsqrt refinement variants on real hardware."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import oracle
import skellysim_amd as ska


def maxrel(a, b):
    denom = np.maximum(np.abs(b), np.abs(b).max() * 1e-6)
    return float(np.abs(a - b).max() / np.abs(b).max()), float((np.abs(a - b) / denom).max())


def main():
    rng = np.random.default_rng(100)
    n_src, n_trg, eta = 20000, 4096, 1.3
    r_src = rng.uniform(-1, 1, (n_src, 3))
    f3 = rng.uniform(-1, 1, (n_src, 3))
    f9 = rng.uniform(-1, 1, (n_src, 9))
    rho = rng.uniform(-1, 1, (n_src, 3))
    r_trg = rng.uniform(-1, 1, (n_trg, 3))
    # include near-coincident pairs to stress small r^2
    r_trg[:64] = r_src[:64] + rng.uniform(-1, 1, (64, 3)) * 1e-4
    r_trg[64:96] = r_src[64:96]  # exact coincident

    checks = [
        ("stokeslet", ska.stokeslet_direct_gpu(r_src, None, r_trg, f3, None, eta),
         oracle.stokeslet(r_src, f3, r_trg, eta)),
        ("stresslet", ska.stresslet_direct_gpu(None, r_src, r_trg, None, f9, eta),
         oracle.stresslet(r_src, f9, r_trg, eta)),
        ("oseen", ska.oseen_contract_direct_gpu(r_src, r_trg, rho, eta),
         oracle.oseen_contract(r_src, r_trg, rho, eta)),
        ("rotlet", ska.rotlet_gpu(r_src[:500], r_trg, rho[:500], eta),
         oracle.rotlet(r_src[:500], r_trg, rho[:500], eta)),
    ]
    for name, u, ref in checks:
        norm_rel = np.linalg.norm(u - ref) / np.linalg.norm(ref)
        mr_scaled, mr_elem = maxrel(u, ref)
        print(f"{name}: norm_rel={norm_rel:.3e} max_err/max|u|={mr_scaled:.3e} "
              f"max_elem_rel={mr_elem:.3e}")


if __name__ == "__main__":
    main()
