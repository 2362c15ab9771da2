"""Multi-GPU target-sharded pair evaluation (RCCL over xGMI).

Decomposition (mirroring the reference's contiguous block
distribution of targets across MPI ranks — fibers:
fiber_container_finite_difference.cpp:102-121, shell nodes:
periphery.cpp:388-406): each rank owns a contiguous block of targets and a
contiguous block of sources; per evaluation the sources are all-gathered
(one collective over xGMI), then each rank computes its target block against
ALL sources with zero further communication. Velocities stay sharded.

The reference's own GPU path is strictly single-rank (system.cpp:618-623
hard-errors "CPU"/"GPU" with >1 MPI rank) — this module is the new capability
that lifts that restriction.

One process per GPU via torch.distributed; backend "nccl" IS RCCL on ROCm.
The collective logic is backend-agnostic so the world_size>1 path is covered
by gloo CPU tests (tests/test_sharding.py); compute on the product path is
the HIP extension (compute_fn defaults to the stokeslet device kernel and
fails loudly without a GPU).
"""

import numpy as np


def shard_sizes(n, world):
    """Contiguous block distribution: first n % world ranks get one extra
    (mirrors get_chunk_start_and_size, reference src/core/kernels.cpp:42-51)."""
    base = n // world
    rem = n % world
    return [base + 1 if r < rem else base for r in range(world)]


def shard_range(n, world, rank):
    sizes = shard_sizes(n, world)
    start = sum(sizes[:rank])
    return start, start + sizes[rank]


def allgather_rows(local, group=None):
    """All-gather a (n_local, d) fp64 tensor with per-rank-varying n_local,
    preserving rank order (rank 0's rows first). Returns a (n_total, d)
    tensor on the same device. Works over nccl (GPU) and gloo (CPU)."""
    import torch
    import torch.distributed as dist

    world = dist.get_world_size(group)
    if world == 1:
        return local
    d = local.shape[1]
    n_local = torch.tensor([local.shape[0]], dtype=torch.int64, device=local.device)
    counts = torch.empty(world, dtype=torch.int64, device=local.device)
    dist.all_gather_into_tensor(counts, n_local, group=group)
    counts = counts.tolist()
    n_max = max(counts)
    padded = torch.zeros((n_max, d), dtype=local.dtype, device=local.device)
    padded[: local.shape[0]] = local
    gathered = torch.empty((world * n_max, d), dtype=local.dtype, device=local.device)
    dist.all_gather_into_tensor(gathered, padded, group=group)
    if all(c == n_max for c in counts):
        return gathered
    parts = [gathered[r * n_max: r * n_max + counts[r]] for r in range(world)]
    return torch.cat(parts, dim=0)


class ShardedPairEvaluator:
    """Target-sharded pair evaluation over a torch.distributed world.

    compute_fn(r_src_all, f_src_all, r_trg_local, eta) -> u_local.
    The default is the HIP stokeslet device kernel (product path; requires a
    GPU and the in-tree extension — no CPU fallback). Tests may inject an
    oracle-backed compute_fn to exercise the collective logic on CPU.
    """

    def __init__(self, compute_fn=None, kernel="stokeslet", group=None):
        if compute_fn is None:
            from . import evaluator as ev

            if kernel == "stokeslet":
                compute_fn = lambda r, f, t, eta: ev.stokeslet_device(r, f, t, eta)
            elif kernel == "stresslet":
                compute_fn = lambda r, f, t, eta: ev.stresslet_device(r, f, t, eta)
            elif kernel == "oseen":
                compute_fn = lambda r, f, t, eta: ev.oseen_contract_device(r, t, f, eta)
            elif kernel == "rotlet":
                compute_fn = lambda r, f, t, eta: ev.rotlet_device(r, t, f, eta)
            else:
                raise ValueError(f"unknown kernel {kernel!r}")

        self._compute = compute_fn
        self._group = group

    def __call__(self, r_src_local, f_src_local, r_trg_local, eta=1.0):
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            r_all = allgather_rows(r_src_local, self._group)
            f_all = allgather_rows(f_src_local, self._group)
        else:
            r_all, f_all = r_src_local, f_src_local
        return self._compute(r_all, f_all, r_trg_local, eta)


def local_shard(arr, world, rank):
    """numpy helper: this rank's contiguous block of rows."""
    a, b = shard_range(len(arr), world, rank)
    return np.asarray(arr)[a:b]
