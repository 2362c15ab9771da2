#!/usr/bin/env python3
"""Minimal repros for the ROCm dense-linalg failure modes that
skellysim_amd/batched.py works around (measured on MI355X, ROCm 7.2,
torch 2.10.0+rocm7.0). Re-run on every ROCm/torch upgrade; when a case
reports OK the corresponding workaround in batched.py can be retired.

Cases (all fp64, --case all by default):
  inv-repeat   torch.linalg.inv at 18000^2 on the default (hipSOLVER/
               hipBLAS) backend: first call in a fresh process is fine,
               an in-process REPEAT has been measured returning a
               SILENTLY corrupt inverse (residual probe ~1e-3 instead of
               ~1e-12) or raising HIPBLAS_STATUS_ALLOC_FAILED.
  inv-huge     torch.linalg.inv at 24576^2: fine once per process on the
               default backend; a repeat ABORTS (uncatchable hipBLAS
               abort) -> this script runs the repeat in a SUBPROCESS and
               reports its exit code. magma at this size SEGFAULTS.
  getrf-batched torch.linalg.lu_factor of a (4000, 132, 132) batch:
               hipblasDgetrfBatched fails with ALLOC_FAILED for m > 128.

Exit code 0 = every requested case behaved correctly (workarounds can be
retired); 1 = at least one still misbehaves.
"""

import argparse
import math
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def make_spd(n, dev, seed=7):
    import torch
    g = torch.Generator(device="cpu").manual_seed(seed)
    A = torch.randn(n, n, generator=g, dtype=torch.float64)
    A = A.to(dev)
    # diagonally dominant: well-conditioned, so probe failures mean
    # corruption, not conditioning
    A += torch.eye(n, dtype=torch.float64, device=dev) * (2.0 * n)
    return A


def probe_rel(A, X):
    import torch
    n = A.shape[-1]
    v = torch.linspace(-1.0, 1.0, n, dtype=A.dtype, device=A.device)
    r = A @ (X @ v) - v
    return float(torch.linalg.vector_norm(r) / torch.linalg.vector_norm(v))


def case_inv_repeat(n=18000, repeats=3):
    import torch
    dev = torch.device("cuda:0")
    A = make_spd(n, dev)
    bad = False
    for i in range(repeats):
        try:
            X = torch.linalg.inv(A)
            rel = probe_rel(A, X)
            ok = math.isfinite(rel) and rel < 1e-8
            print(f"inv-repeat[{i}] n={n}: probe rel={rel:.3e} "
                  f"{'OK' if ok else 'CORRUPT'}", flush=True)
            bad |= not ok
            del X
        except RuntimeError as e:
            print(f"inv-repeat[{i}] n={n}: RAISED {e}", flush=True)
            bad = True
        torch.cuda.empty_cache()
    return not bad


def case_inv_huge():
    # the failing repeat is uncatchable -> isolate it in a subprocess
    code = (
        "import torch, sys; sys.path.insert(0, %r)\n"
        "from tools.repro_rocm_linalg import make_spd, probe_rel\n"
        "dev = torch.device('cuda:0')\n"
        "A = make_spd(24576, dev)\n"
        "for i in range(2):\n"
        "    X = torch.linalg.inv(A)\n"
        "    rel = probe_rel(A, X)\n"
        "    print(f'inv-huge[{i}]: probe rel={rel:.3e}', flush=True)\n"
        "    assert rel < 1e-8, rel\n"
        "    del X; torch.cuda.empty_cache()\n"
    ) % os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run([sys.executable, "-c", code], timeout=900)
    print(f"inv-huge subprocess exit code: {r.returncode} "
          f"{'OK' if r.returncode == 0 else 'STILL ABORTS/CORRUPTS'}",
          flush=True)
    return r.returncode == 0


def case_getrf_batched(nb=4000, m=132):
    import torch
    dev = torch.device("cuda:0")
    g = torch.Generator(device="cpu").manual_seed(3)
    A = torch.randn(nb, m, m, generator=g, dtype=torch.float64).to(dev)
    A += torch.eye(m, dtype=torch.float64, device=dev) * (2.0 * m)
    try:
        LU, piv = torch.linalg.lu_factor(A)
        x = torch.linalg.lu_solve(LU, piv, A[:, :, :1])
        rel = float(torch.linalg.vector_norm(A @ x - A[:, :, :1]) /
                    torch.linalg.vector_norm(A[:, :, :1]))
        ok = math.isfinite(rel) and rel < 1e-8
        print(f"getrf-batched nb={nb} m={m}: solve rel={rel:.3e} "
              f"{'OK' if ok else 'CORRUPT'}", flush=True)
        return ok
    except RuntimeError as e:
        print(f"getrf-batched nb={nb} m={m}: RAISED {e}", flush=True)
        return False


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--case", default="all",
                    choices=["all", "inv-repeat", "inv-huge", "getrf-batched"])
    args = ap.parse_args()
    results = {}
    if args.case in ("all", "inv-repeat"):
        results["inv-repeat"] = case_inv_repeat()
    if args.case in ("all", "getrf-batched"):
        results["getrf-batched"] = case_getrf_batched()
    if args.case in ("all", "inv-huge"):
        results["inv-huge"] = case_inv_huge()
    print("summary:", {k: ("OK" if v else "FAIL") for k, v in results.items()},
          flush=True)
    sys.exit(0 if all(results.values()) else 1)


if __name__ == "__main__":
    main()
