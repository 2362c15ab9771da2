#!/usr/bin/env python3
"""Minimal profiling driver: repeated hot-path kernel launches on cuda:0,
nothing else timed/launched. Used under rocprofv3 (kernel-trace/stats in one
run; PMC counters in separate runs)."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import skellysim_amd as ska


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--kernel", default="stokeslet",
                   choices=["stokeslet", "stresslet", "oseen", "rotlet"])
    p.add_argument("--n-src", type=int, default=1_000_000)
    p.add_argument("--n-trg", type=int, default=1_000_000)
    p.add_argument("--iters", type=int, default=3)
    p.add_argument("--seed", type=int, default=100)
    args = p.parse_args()

    rng = np.random.default_rng(args.seed)
    dev = torch.device("cuda:0")
    r_src = torch.from_numpy(rng.uniform(-1, 1, (args.n_src, 3))).to(dev)
    r_trg = torch.from_numpy(rng.uniform(-1, 1, (args.n_trg, 3))).to(dev)
    f3 = torch.from_numpy(rng.uniform(-1, 1, (args.n_src, 3))).to(dev)
    f9 = torch.from_numpy(rng.uniform(-1, 1, (args.n_src, 9))).to(dev)
    out = torch.empty_like(r_trg)
    torch.cuda.synchronize()

    def run():
        if args.kernel == "stokeslet":
            ska.stokeslet_device(r_src, f3, r_trg, 1.0, out=out)
        elif args.kernel == "stresslet":
            ska.stresslet_device(r_src, f9, r_trg, 1.0, out=out)
        elif args.kernel == "oseen":
            ska.oseen_contract_device(r_src, r_trg, f3, 1.0, out=out)
        else:
            ska.rotlet_device(r_src, r_trg, f3, 1.0, out=out)

    t0 = time.perf_counter()
    for _ in range(args.iters):
        run()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    pairs = float(args.n_src) * args.n_trg
    print(f"{args.kernel} n_src={args.n_src} n_trg={args.n_trg} iters={args.iters} "
          f"avg={dt*1e3:.2f} ms  pairs/s={pairs/dt:.4e}")


if __name__ == "__main__":
    main()
