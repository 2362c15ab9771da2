#!/usr/bin/env python3
"""Config-4-scale component timings on one MI355X: the periphery dense GEMVs
(8k-node shell), batched per-fiber LU, batched self-stokeslet tensor build,
and the GMRES-sized flow evaluations. Prints one line per component."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import skellysim_amd as ska
from skellysim_amd.flows import ShellOperator
from skellysim_amd.batched import BatchedLU


def timeit(fn, iters=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dev = torch.device("cuda:0")
    rng = np.random.default_rng(1)

    # Periphery dense GEMVs at the config-4 scale: 8192-node shell -> 24576^2
    N = 8192
    M = torch.randn(3 * N, 3 * N, dtype=torch.float64, device=dev)
    x = torch.randn(3 * N, dtype=torch.float64, device=dev)
    v = torch.randn(3 * N, dtype=torch.float64, device=dev)
    op = ShellOperator(M, M)
    t = timeit(lambda: op.apply_preconditioner(x))
    gb = (3 * N) * (3 * N) * 8 / 1e9
    print(f"shell_precond_gemv N={N}: {t*1e3:.3f} ms  ({gb/t:.0f} GB/s of {gb:.1f} GB matrix)")
    t = timeit(lambda: op.matvec(x, v))
    print(f"shell_matvec_gemv  N={N}: {t*1e3:.3f} ms  ({gb/t:.0f} GB/s)")

    # Batched per-fiber LU at config-4 scale: 512 fibers x 64 nodes -> 256^2
    nf, m = 512, 256
    A = torch.randn(nf, m, m, dtype=torch.float64, device=dev) + 8 * torch.eye(
        m, dtype=torch.float64, device=dev)
    t_factor = timeit(lambda: BatchedLU(A), iters=5, warmup=1)
    lu = BatchedLU(A)
    b = torch.randn(nf, m, dtype=torch.float64, device=dev)
    t_solve = timeit(lambda: lu.solve(b))
    print(f"batched_lu nf={nf} m={m}: factor {t_factor*1e3:.2f} ms, solve {t_solve*1e3:.3f} ms")

    # Batched self-stokeslet build: 512 fibers x 64 nodes
    pts = torch.from_numpy(rng.uniform(-1, 1, (nf, 64, 3))).to(dev)
    t = timeit(lambda: ska.oseen_tensor_batched_device(pts))
    print(f"oseen_tensor_batched nf={nf} n=64: {t*1e3:.3f} ms")

    # Dense stresslet_times_normal at a body/shell assembly scale
    n = 8192
    p8 = torch.from_numpy(rng.uniform(-1, 1, (n, 3))).to(dev)
    n8 = torch.from_numpy(rng.uniform(-1, 1, (n, 3))).to(dev)
    t = timeit(lambda: ska.stresslet_times_normal_device(p8, n8), iters=5, warmup=1)
    print(f"stresslet_times_normal_dense n={n}: {t*1e3:.2f} ms "
          f"({(3*n)**2*8/1e9/t:.0f} GB/s write)")

    # GMRES-iteration-sized pair evaluations (config 4: ~33k fiber nodes +
    # 8k shell nodes): fiber-stokeslet all->all and shell-stresslet
    n_fib = 512 * 64
    n_shell = 8192
    n_all = n_fib + n_shell
    r_all = torch.from_numpy(rng.uniform(-1, 1, (n_all, 3))).to(dev)
    f3 = torch.from_numpy(rng.uniform(-1, 1, (n_fib, 3))).to(dev)
    f9 = torch.from_numpy(rng.uniform(-1, 1, (n_shell, 9))).to(dev)
    t = timeit(lambda: ska.stokeslet_device(r_all[:n_fib], f3, r_all, 1.0))
    print(f"fiber_stokeslet {n_fib}x{n_all}: {t*1e3:.3f} ms "
          f"({n_fib*n_all/t/1e12:.3f}e12 pairs/s)")
    t = timeit(lambda: ska.stresslet_device(r_all[n_fib:], f9, r_all, 1.0))
    print(f"shell_stresslet {n_shell}x{n_all}: {t*1e3:.3f} ms "
          f"({n_shell*n_all/t/1e12:.3f}e12 pairs/s)")


if __name__ == "__main__":
    main()
