#!/usr/bin/env python3
"""Hunt the config-5 solve corruption: exercise the pair kernels at the
exact failing sizes (stokeslet 128k x 134k split-8, stresslet 6k x 128k
split-3) checking (a) bitwise determinism across repeated calls with
allocator churn in between, and (b) 64-target subset parity against a
torch-composed direct sum."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import skellysim_amd as ska


def torch_stokeslet(r_src, f_src, r_trg, eta):
    dr = r_trg[:, None, :] - r_src[None, :, :]
    r2 = (dr * dr).sum(-1)
    rinv = torch.where(r2 > 0, r2.rsqrt(), torch.zeros_like(r2))
    fdotr = (dr * f_src[None, :, :]).sum(-1)
    u = (f_src[None, :, :] * rinv[:, :, None]
         + dr * (fdotr * rinv ** 3)[:, :, None]).sum(1)
    return u / (8 * np.pi * eta)


def torch_stresslet(r_src, f9, r_trg, eta):
    dr = r_trg[:, None, :] - r_src[None, :, :]
    r2 = (dr * dr).sum(-1)
    rinv = torch.where(r2 > 0, r2.rsqrt(), torch.zeros_like(r2))
    S = f9.reshape(-1, 3, 3)
    sv = torch.einsum("tsj,sjk,tsk->ts", dr, S, dr)
    u = (dr * (sv * rinv ** 5)[:, :, None]).sum(1)
    return -3.0 / (8 * np.pi * eta) * u


def churn(i):
    """Vary allocator layout between calls."""
    sizes = [1 << (14 + (i % 8)), 12345 + 777 * i, 3 * ((i * 997) % 100000 + 1)]
    keep = [torch.empty(s, dtype=torch.float64, device="cuda:0") for s in sizes]
    del keep
    if i % 3 == 0:
        torch.cuda.empty_cache()


def run_case(name, fn, ref_fn, r_src, f_src, n_trg):
    r_trg = r_src[:n_trg] if n_trg <= len(r_src) else torch.cat(
        [r_src, r_src[: n_trg - len(r_src)]])
    u0 = fn(r_src, f_src, r_trg, 1.0).clone()
    torch.cuda.synchronize()
    ok = True
    for i in range(12):
        churn(i)
        u = fn(r_src, f_src, r_trg, 1.0)
        torch.cuda.synchronize()
        if not torch.equal(u, u0):
            nbad = int((u != u0).any(dim=1).sum())
            print(f"{name}: NONDETERMINISTIC at call {i}: {nbad} rows differ, "
                  f"max abs diff {float((u - u0).abs().max()):.3e}", flush=True)
            ok = False
            break
    if ok:
        print(f"{name}: deterministic over 12 churned calls", flush=True)
    ref = ref_fn(r_src, f_src, r_trg[:64], 1.0)
    rel = float(torch.norm(u0[:64] - ref) / torch.norm(ref))
    print(f"{name}: 64-target parity vs torch direct rel={rel:.3e}", flush=True)
    return ok and rel < 1e-10


def main():
    dev = torch.device("cuda:0")
    g = torch.Generator(device="cpu").manual_seed(7)
    mk = lambda *s: torch.from_numpy(
        np.random.default_rng(hash(s) % 2**31).uniform(-1, 1, s)).to(dev)

    r_fib = mk(128000, 3)
    f_fib = mk(128000, 3)
    ok1 = run_case("stokeslet 128k->134k(split8)", ska.stokeslet_device,
                   torch_stokeslet, r_fib, f_fib, 134000)

    r_sh = mk(6000, 3)
    f9 = mk(6000, 9)
    ok2 = run_case("stresslet 6k->128k(split3)", ska.stresslet_device,
                   torch_stresslet, r_sh, f9, 128000)

    # G bmm + oseen builder determinism at config-5 shape
    pts = mk(4000, 32, 3)
    G0 = ska.oseen_tensor_batched_device(pts, eta=1.0).clone()
    torch.cuda.synchronize()
    for i in range(6):
        churn(i)
        G = ska.oseen_tensor_batched_device(pts, eta=1.0)
        torch.cuda.synchronize()
        assert torch.equal(G, G0), f"oseen builder nondeterministic at {i}"
    print("oseen_tensor_batched: deterministic over 6 churned calls", flush=True)
    print("ALL OK" if (ok1 and ok2) else "PROBLEM FOUND", flush=True)


if __name__ == "__main__":
    main()
