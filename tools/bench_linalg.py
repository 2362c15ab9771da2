#!/usr/bin/env python3
"""Microbenchmark the batched dense-linalg candidates for the per-iteration
fiber-block preconditioner at config-5 shape (batch 4000, m = 128, fp64):
factor/invert once per timestep, solve once per GMRES iteration.
Places the cost of each option so the default SKELLY_LU_MODE is chosen on
measurement, not guesswork (round-2: 'inv' mode showed 18.7 s/step at
config 5 — this isolates whether torch.linalg.inv-batched or bmm is the
regression)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def t_ms(fn, iters=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    dev = torch.device("cuda:0")
    nb, m = 4000, 128
    g = torch.Generator().manual_seed(0)
    A = torch.randn(nb, m, m, generator=g, dtype=torch.float64).to(dev)
    A += torch.eye(m, dtype=torch.float64, device=dev) * (2.0 * m)
    x = torch.randn(nb, m, 1, generator=g, dtype=torch.float64).to(dev)

    # ---- once-per-timestep candidates ----
    print(f"lu_factor batched:        {t_ms(lambda: torch.linalg.lu_factor(A), 5):9.2f} ms")
    print(f"linalg.inv batched:       {t_ms(lambda: torch.linalg.inv(A), 3, 1):9.2f} ms")
    LU, piv = torch.linalg.lu_factor(A)
    eye = torch.eye(m, dtype=torch.float64, device=dev).expand(nb, m, m).contiguous()
    print(f"inv via lu_solve(I):      {t_ms(lambda: torch.linalg.lu_solve(LU, piv, eye), 3, 1):9.2f} ms")

    # ---- per-iteration candidates ----
    Ainv = torch.linalg.inv(A)
    print(f"bmm (Ainv@x, k=1):        {t_ms(lambda: torch.bmm(Ainv, x)):9.3f} ms")
    print(f"matmul (Ainv@x):          {t_ms(lambda: Ainv @ x):9.3f} ms")
    print(f"mul+sum (Ainv*x).sum:     {t_ms(lambda: (Ainv * x.transpose(1, 2)).sum(-1)):9.3f} ms")
    print(f"lu_solve (magma):         {t_ms(lambda: torch.linalg.lu_solve(LU, piv, x)):9.3f} ms")
    tri = lambda: torch.linalg.solve_triangular(
        LU, torch.linalg.solve_triangular(LU, x, upper=False, unitriangular=True),
        upper=True)
    print(f"2x solve_triangular:      {t_ms(tri):9.3f} ms")

    xf = x.reshape(nb, m)
    print(f"einsum bij,bj->bi:        {t_ms(lambda: torch.einsum('bij,bj->bi', Ainv, xf)):9.3f} ms")
    # k=32 block solve (the cadence-8 regime batches up to 8 RHS anyway? no
    # — RHS arrive one per iteration; k=1 is the real shape. k=32 shown for
    # context on how much of bmm's k=1 cost is launch/shape overhead)
    x32 = torch.randn(nb, m, 32, generator=g, dtype=torch.float64).to(dev)
    print(f"bmm k=32 (context):       {t_ms(lambda: torch.bmm(Ainv, x32)):9.3f} ms")


if __name__ == "__main__":
    main()
