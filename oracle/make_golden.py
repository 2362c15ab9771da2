"""Generate golden parity fixtures for the hot-path kernels (tests/golden/*.npz).

Runs ONLY in the build container, where /root/reference is mounted: it imports
the reference's own Python statement of the kernel math
(/root/reference/src/skelly_sim/kernels.py, with a no-op numba shim), checks the
C oracle (oracle/kernels_ref.c) against it, and then freezes C-oracle outputs
as golden fixtures at the reference's own kernel-parity recipe sizes
(tests/core/kernel_test.cpp:25-27: n_src=1229, n_trg=743, eta=1.3).

The GPU box never runs this script and never reads /root/reference: GPU parity
tests compare the HIP kernels against the committed fixtures and against the
C oracle shipped in-tree.

Cross-check notes (documented mismatches between the Python statement and the
C++ hot path, which the C oracle follows):
  * python oseen/stresslet/rotlet regularize pairs with 0 < r < eps while the
    C++ stokeslet/stresslet evaluators only zero exact r==0 (kernels.cu:39,70)
    — cross-check clouds therefore keep min separation >> eps (asserted);
  * python oseen has no r==0 skip while kernels.cpp:105-106 skips — coincident
    pairs are excluded from the python cross-check and covered by
    C-vs-numpy-oracle tests instead (tests/test_oracle.py).

Usage: python3 oracle/make_golden.py   (from the repo root)
"""

import sys
import os
import types

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
import oracle

REF_KERNELS = "/root/reference/src/skelly_sim/kernels.py"
GOLDEN_DIR = os.path.join(REPO, "tests", "golden")


def import_reference_kernels():
    """Import the reference kernels.py with a no-op numba shim."""
    numba = types.ModuleType("numba")

    def njit(*args, **kwargs):
        if args and callable(args[0]):
            return args[0]
        return lambda f: f

    numba.njit = njit
    numba.prange = range
    sys.modules.setdefault("numba", numba)

    import importlib.util
    spec = importlib.util.spec_from_file_location("ref_kernels", REF_KERNELS)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def rel_err(a, b):
    return np.linalg.norm(a - b) / np.linalg.norm(b)


def main():
    ref = import_reference_kernels()
    rng = np.random.default_rng(100)

    # --- small clouds for the (slow, pure-python) reference cross-check ---
    S, T = 257, 141
    eta = 1.3
    r_src = rng.uniform(-1, 1, (S, 3))
    r_trg = rng.uniform(-1, 1, (T, 3))
    f3 = rng.uniform(-1, 1, (S, 3))
    normal = rng.uniform(-1, 1, (S, 3))
    rho = rng.uniform(-1, 1, (S, 3))

    allpts = np.vstack([r_src, r_trg])
    dmin = np.inf
    for i in range(len(allpts)):
        d = np.linalg.norm(allpts[i + 1:] - allpts[i], axis=1)
        if len(d):
            dmin = min(dmin, d.min())
    assert dmin > 1e-3, f"cloud min separation {dmin} too small for python cross-check"

    checks = {}

    # Stokeslet: singular Stokeslet == reference python oseen kernel for
    # separated points (kernels.py:272, fr=factor/r, gr=factor/r^3).
    u_c = oracle.stokeslet(r_src, f3, r_trg, eta)
    u_py = ref.oseen_kernel_source_target_numba(
        r_src.flatten(), r_trg.flatten(), f3.flatten(), eta=eta).reshape(-1, 3)
    checks["stokeslet_vs_refpy"] = rel_err(u_c, u_py)

    # Regularized Oseen contraction: same reference python function IS the
    # regularized oseen statement (identical branches for r > 0).
    u_c = oracle.oseen_contract(r_src, r_trg, f3, eta)
    checks["oseen_vs_refpy"] = rel_err(u_c, u_py)

    # Near-branch check: targets displaced ~5e-6 (< eps=1e-5) from sources.
    near_trg = r_src[:40] + rng.uniform(-1, 1, (40, 3)) * 3e-6
    u_c = oracle.oseen_contract(r_src[:40], near_trg, f3[:40], eta)
    u_py = ref.oseen_kernel_source_target_numba(
        r_src[:40].flatten(), near_trg.flatten(), f3[:40].flatten(), eta=eta).reshape(-1, 3)
    checks["oseen_near_vs_refpy"] = rel_err(u_c, u_py)

    # Stresslet: C++ evaluator consumes f_dl(i*3+j) = 2*eta*n_i*rho_j
    # (periphery.cpp:68-71); that equals the reference python
    # stresslet_kernel_source_target_numba(r_src, r_trg, n, rho) exactly.
    f9 = 2.0 * eta * np.einsum("si,sj->sij", normal, rho).reshape(S, 9)
    u_c = oracle.stresslet(r_src, f9, r_trg, eta)
    u_py = ref.stresslet_kernel_source_target_numba(
        r_src.flatten(), r_trg.flatten(), normal.flatten(), rho.flatten(),
        eta=eta).reshape(-1, 3)
    checks["stresslet_vs_refpy"] = rel_err(u_c, u_py)

    # Rotlet: direct statement match (kernels.py:336 vs kernels.cpp:206-242).
    u_c = oracle.rotlet(r_src, r_trg, rho, eta)
    u_py = ref.rotlet_kernel_source_target_numba(
        r_src.flatten(), r_trg.flatten(), rho.flatten(), eta=eta).reshape(-1, 3)
    checks["rotlet_vs_refpy"] = rel_err(u_c, u_py)

    for k, v in checks.items():
        print(f"{k}: rel err = {v:.3e}")
        assert v < 1e-12, f"{k} FAILED: {v}"

    os.makedirs(GOLDEN_DIR, exist_ok=True)
    np.savez_compressed(
        os.path.join(GOLDEN_DIR, "refpy_small.npz"),
        r_src=r_src, r_trg=r_trg, f3=f3, normal=normal, rho=rho, f9=f9,
        eta=eta, near_trg=near_trg,
        u_stokeslet=oracle.stokeslet(r_src, f3, r_trg, eta),
        u_stresslet=oracle.stresslet(r_src, f9, r_trg, eta),
        u_oseen=oracle.oseen_contract(r_src, r_trg, f3, eta),
        u_oseen_near=oracle.oseen_contract(r_src[:40], near_trg, f3[:40], eta),
        u_rotlet=oracle.rotlet(r_src, r_trg, rho, eta),
    )

    # --- kernel_test.cpp recipe-size fixtures (C oracle, now pinned) ---
    # n_src=1229, n_trg=743, eta=1.3 (tests/core/kernel_test.cpp:25-27); the
    # reference uses MatrixXd::Random clouds — we use the same distribution
    # (uniform [-1,1]) from a fixed numpy seed.
    rng = np.random.default_rng(100)
    S, T = 1229, 743
    r_src = rng.uniform(-1, 1, (S, 3))
    r_trg = rng.uniform(-1, 1, (T, 3))
    f3 = rng.uniform(-1, 1, (S, 3))
    f9 = rng.uniform(-1, 1, (S, 9))
    np.savez_compressed(
        os.path.join(GOLDEN_DIR, "kernel_test_1229x743.npz"),
        r_src=r_src, r_trg=r_trg, f3=f3, f9=f9, eta=eta,
        u_stokeslet=oracle.stokeslet(r_src, f3, r_trg, eta),
        u_stresslet=oracle.stresslet(r_src, f9, r_trg, eta),
        u_oseen=oracle.oseen_contract(r_src, r_trg, f3, eta),
        u_rotlet=oracle.rotlet(r_src[:37], r_trg, f3[:37], eta),
    )

    # --- edge-case fixture: coincident points, src==trg self-interaction ---
    rng = np.random.default_rng(7)
    r = rng.uniform(-1, 1, (128, 3))
    r_dup = np.vstack([r, r[:17]])          # duplicated points
    f3d = rng.uniform(-1, 1, (len(r_dup), 3))
    f9d = rng.uniform(-1, 1, (len(r_dup), 9))
    np.savez_compressed(
        os.path.join(GOLDEN_DIR, "edge_selfdup.npz"),
        r=r_dup, f3=f3d, f9=f9d, eta=0.7,
        u_stokeslet=oracle.stokeslet(r_dup, f3d, r_dup, 0.7),
        u_stresslet=oracle.stresslet(r_dup, f9d, r_dup, 0.7),
        u_oseen=oracle.oseen_contract(r_dup, r_dup, f3d, 0.7),
        u_rotlet=oracle.rotlet(r_dup, r_dup, f3d, 0.7),
    )
    print("golden fixtures written to", GOLDEN_DIR)


if __name__ == "__main__":
    main()
