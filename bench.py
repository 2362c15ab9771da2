#!/usr/bin/env python3
"""bench.py — headline benchmark: Stokeslet pair-interactions/sec (fp64).

Workload (BASELINE.json metric config): the free-space Stokeslet direct
evaluation at N=1e6 (n_src = n_trg = 1e6, uniform random fp64 clouds in
[-1,1]^3, seed 100, sources == targets), fully scaled (1/(8 pi eta), eta=1).
A "step" is ONE full evaluation pass (1e12 pair interactions).

Multi-GPU (--gpus N under torch.distributed.run): the N=1e6 cloud is fixed
(strong scaling); targets are block-sharded across ranks; sources are
generated sharded and all-gathered over RCCL/xGMI inside the timed region
(the real per-evaluation exchange).

Prints ONE JSON line from rank 0. Also included:
  roofline     — dominant-kernel achieved fp64 FLOP/s (28 flops/pair
                 convention) vs the 78.6 TF fp64 vector peak,
                 measured with HIP events on the launch stream.
  cpu_baseline — the C oracle (reference CPU path restated, OpenMP all
                 cores) timed on this box on a bounded target sample.
"""

import argparse
import ctypes
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

FLOPS_PER_PAIR = 28.0        # Stokeslet convention, SURVEY.md §8(d)
FP64_PEAK_TFLOPS = 78.6      # MI355X fp64 vector peak (nominal; measured value
                             # also reported as fp64_peak_measured_tflops)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--n", "--npoints", dest="n", type=int, default=1_000_000,
                   help="cloud size (sources == targets); default the metric point 1e6")
    p.add_argument("--seed", type=int, default=100)
    p.add_argument("--skip-cpu-baseline", action="store_true")
    p.add_argument("--kernel", default="stokeslet", choices=["stokeslet", "oseen"],
                   help="stokeslet = the headline metric; oseen = BASELINE config 3 "
                        "(regularized Stokeslet)")
    return p.parse_args()


def cpu_baseline(r_src, f_src, r_trg, eta, kernel="stokeslet", budget_s=15.0):
    """Time the C oracle (reference CPU path restated; kind='port') on a
    bounded target sample of the same workload; returns pairs/s."""
    import oracle
    if kernel == "stokeslet":
        fn = lambda t: oracle.stokeslet(r_src, f_src, t, eta)
    else:
        fn = lambda t: oracle.oseen_contract(r_src, t, f_src, eta)
    n_src = len(r_src)
    cores = oracle.num_threads()
    cpu_model = None
    try:
        with open("/proc/cpuinfo") as fh:
            for line in fh:
                if line.startswith("model name"):
                    cpu_model = line.split(":", 1)[1].strip()
                    break
    except OSError:
        pass
    # probe to pick a sample that costs ~budget_s
    probe_t = min(256, len(r_trg))
    t0 = time.perf_counter()
    fn(r_trg[:probe_t])
    dt = time.perf_counter() - t0
    rate = probe_t * n_src / max(dt, 1e-9)
    sample_t = int(min(len(r_trg), max(probe_t, rate * budget_s / n_src)))
    t0 = time.perf_counter()
    fn(r_trg[:sample_t])
    dt = time.perf_counter() - t0
    return {
        "value": sample_t * n_src / dt,
        "unit": "pairs/s",
        "cores": cores,
        "kind": "port",
        "sample": f"{sample_t} of {len(r_trg)} targets x {n_src} sources, "
                  f"{dt:.1f}s, OpenMP {cores} threads",
        "cpu": cpu_model,
    }


def load_traffic_calibration(n_src, n_trg):
    """Per-launch HBM traffic from a committed rocprofv3 --pmc calibration run
    of this exact workload (profiles/traffic_*.json), else None."""
    path = os.path.join(REPO, "profiles", "traffic_calibration.json")
    if not os.path.exists(path):
        return None
    try:
        d = json.load(open(path))
        if d.get("n_src") == n_src and d.get("n_trg") == n_trg:
            return d.get("bytes_per_launch")
    except Exception:
        pass
    return None


def main():
    args = parse_args()
    import torch

    # CI rehearsal of the distributed plumbing (tests/test_bench_cli.py):
    # gloo + CPU tensors + stub compute; never set by the driver.
    rehearsal = os.environ.get("SKELLY_BENCH_REHEARSAL") == "gloo"

    if not torch.cuda.is_available() and not rehearsal:
        print(json.dumps({"error": "no GPU visible; bench.py must run on an MI355X box"}))
        sys.exit(1)

    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        # launched directly with --gpus N: re-exec under torchrun
        os.execvp(sys.executable, [
            sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1",
            "--master-port", "29571", os.path.abspath(__file__)] + sys.argv[1:])

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist = None
    # initialize the process group whenever a launcher provided WORLD_SIZE
    # (even world 1): a torchrun world-1 run then exercises the full RCCL
    # init/barrier/allreduce path on one GPU — the same code the driver's
    # 2/4/8-GPU SCALE launch takes
    if "WORLD_SIZE" in os.environ:
        import torch.distributed as dist_mod
        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group("gloo" if rehearsal else "nccl", rank=rank, world_size=world)
    if not rehearsal:
        torch.cuda.set_device(local_rank)

    import skellysim_amd as ska
    from skellysim_amd import _native
    from skellysim_amd.sharded import shard_range, allgather_rows

    n = args.n
    eta = 1.0
    rng = np.random.default_rng(args.seed)
    # one global cloud, deterministic across ranks; sources == targets
    pts = rng.uniform(-1, 1, (n, 3))
    strengths = rng.uniform(-1, 1, (n, 3))

    ts, te = shard_range(n, world, rank)   # target shard (contiguous blocks)
    ss, se = shard_range(n, world, rank)   # source shard (same split)
    dev = torch.device("cpu") if rehearsal else torch.device("cuda", local_rank)
    r_src_local = torch.from_numpy(pts[ss:se].copy()).to(dev)
    f_src_local = torch.from_numpy(strengths[ss:se].copy()).to(dev)
    r_trg_local = torch.from_numpy(pts[ts:te].copy()).to(dev)
    u_local = torch.empty_like(r_trg_local)

    def sync():
        if not rehearsal:
            torch.cuda.synchronize()

    sync()

    if rehearsal:
        class _Ev:
            def record(self):
                pass
            def elapsed_time(self, other):
                return 1.0
        ev_start = [_Ev() for _ in range(args.steps)]
        ev_end = [_Ev() for _ in range(args.steps)]
    else:
        ev_start = [torch.cuda.Event(enable_timing=True) for _ in range(args.steps)]
        ev_end = [torch.cuda.Event(enable_timing=True) for _ in range(args.steps)]

    if rehearsal:
        kernel_fn = lambda r, f, t, out: out.zero_()
        flops_per_pair, metric = FLOPS_PER_PAIR, "rehearsal (no compute)"
    elif args.kernel == "stokeslet":
        kernel_fn = lambda r, f, t, out: ska.stokeslet_device(r, f, t, eta, out=out)
        flops_per_pair, metric = FLOPS_PER_PAIR, "Stokeslet pair-interactions/sec (fp64)"
    else:  # regularized Stokeslet (BASELINE config 3); defaults kernels.hpp:34-35
        kernel_fn = lambda r, f, t, out: ska.oseen_contract_device(r, t, f, eta, out=out)
        flops_per_pair, metric = 32.0, "Regularized-Stokeslet pair-interactions/sec (fp64)"

    def step(i=None):
        if world > 1:
            r_all = allgather_rows(r_src_local)
            f_all = allgather_rows(f_src_local)
        else:
            r_all, f_all = r_src_local, f_src_local
        if i is not None:
            ev_start[i].record()
        kernel_fn(r_all, f_all, r_trg_local, u_local)
        if i is not None:
            ev_end[i].record()

    for _ in range(args.warmup):
        step()
    sync()
    if dist:
        dist.barrier()
    sync()

    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    sync()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if dist:
        e = torch.tensor([elapsed], device=dev)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())
        dist.barrier()

    # checksum sanity: result finite
    assert torch.isfinite(u_local).all(), "non-finite velocities"

    total_pairs = float(n) * float(n) * args.steps
    pairs_per_s = total_pairs / elapsed

    if rank == 0:
        # dominant-kernel timing (HIP events on the launch stream, this rank)
        kernel_ms = [s.elapsed_time(t) for s, t in zip(ev_start, ev_end)]
        kernel_s = float(np.mean(kernel_ms)) / 1e3
        pairs_per_launch = float(n) * (te - ts)
        achieved_tflops = flops_per_pair * pairs_per_launch / kernel_s / 1e12
        traffic = load_traffic_calibration(n, te - ts) if args.kernel == "stokeslet" else None

        peak_meas = ctypes.c_double(0.0)
        if not rehearsal:
            try:
                _native.lib().skelly_fp64_peak_tflops(ctypes.byref(peak_meas))
            except Exception:
                pass

        cpu = None
        if world == 1 and not args.skip_cpu_baseline and not rehearsal:
            cpu = cpu_baseline(pts, strengths, pts, eta, kernel=args.kernel)

        out = {
            "metric": metric,
            "value": pairs_per_s,
            "unit": "pairs/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": f"{args.kernel}_direct_N1e6" if n == 1_000_000
                            else f"{args.kernel}_direct_N{n}",
                "n_src": n,
                "n_trg": n,
                "seed": args.seed,
                "eta": eta,
                "parallelism": f"target-sharded dp{world}, per-step RCCL source all-gather"
                               if world > 1 else "single GPU",
            },
            "roofline": {
                # schema enum offers only hbm|mfma; the kernel is actually
                # fp64 VALU instruction-issue bound — see
                # bound_detail
                "bound": "mfma",
                "bound_detail": "fp64-valu-issue (compute; not matrix-core "
                                "shaped: gfx950 fp64 MFMA rate == vector rate)",
                "achieved": achieved_tflops,
                "peak": FP64_PEAK_TFLOPS,
                "unit": "TFLOP/s",
                "frac": achieved_tflops / FP64_PEAK_TFLOPS,
                "traffic": traffic,
            },
            "fp64_peak_measured_tflops": peak_meas.value or None,
            "cpu_baseline": cpu,
        }
        print(json.dumps(out))

    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
