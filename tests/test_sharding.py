"""CPU (gloo, world_size=2) tests of the multi-GPU sharding logic: contiguous
target/source block distribution + rank-ordered source all-gather. Compute is
injected (oracle) so the collective path runs without a GPU; the same
ShardedPairEvaluator drives the HIP kernel on GPU boxes."""

import os
import tempfile

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from skellysim_amd.sharded import shard_sizes, shard_range, local_shard

WORLD = 2


def test_shard_sizes_block_distribution():
    # mirrors get_chunk_start_and_size (reference kernels.cpp:42-51)
    assert shard_sizes(10, 4) == [3, 3, 2, 2]
    assert shard_sizes(8, 4) == [2, 2, 2, 2]
    assert shard_sizes(3, 4) == [1, 1, 1, 0]
    assert sum(shard_sizes(1_000_000, 8)) == 1_000_000


def test_shard_range_contiguous_cover():
    n, world = 1013, 8
    spans = [shard_range(n, world, r) for r in range(world)]
    assert spans[0][0] == 0 and spans[-1][1] == n
    for (a0, a1), (b0, b1) in zip(spans, spans[1:]):
        assert a1 == b0


def _worker(rank, world, init_file, n_src, n_trg, q):
    import torch.distributed as dist
    from skellysim_amd.sharded import ShardedPairEvaluator, allgather_rows, local_shard
    import oracle

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        rng = np.random.default_rng(100)
        r_src = rng.uniform(-1, 1, (n_src, 3))
        f_src = rng.uniform(-1, 1, (n_src, 3))
        r_trg = rng.uniform(-1, 1, (n_trg, 3))

        # varlen all-gather preserves rank order
        local = torch.from_numpy(local_shard(r_src, world, rank).copy())
        gathered = allgather_rows(local)
        assert np.array_equal(gathered.numpy(), r_src), "allgather_rows lost rank order"

        def compute(r_all, f_all, trg, eta):
            return torch.from_numpy(
                oracle.stokeslet(r_all.numpy(), f_all.numpy(), trg.numpy(), eta))

        ev = ShardedPairEvaluator(compute_fn=compute)
        u_local = ev(torch.from_numpy(local_shard(r_src, world, rank).copy()),
                     torch.from_numpy(local_shard(f_src, world, rank).copy()),
                     torch.from_numpy(local_shard(r_trg, world, rank).copy()), eta=1.3)
        q.put((rank, u_local.numpy()))
    finally:
        dist.destroy_process_group()


def _shell_worker(rank, world, init_file, N, q):
    import torch.distributed as dist
    from skellysim_amd.flows import ShellOperator
    from skellysim_amd.sharded import shard_range

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        rng = np.random.default_rng(77)
        M = rng.uniform(-1, 1, (3 * N, 3 * N))
        x = rng.uniform(-1, 1, 3 * N)
        v = rng.uniform(-1, 1, 3 * N)
        a, b = shard_range(3 * N, world, rank)  # row-block distribution
        op = ShellOperator(torch.from_numpy(M[a:b].copy()),
                           torch.from_numpy(M[a:b].copy()), distributed=True)
        p = op.apply_preconditioner(torch.from_numpy(x[a:b].copy()))
        m = op.matvec(torch.from_numpy(x[a:b].copy()), torch.from_numpy(v[a:b].copy()))
        q.put((rank, p.numpy(), m.numpy()))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_shell_operator_row_sharded_gloo():
    """Row-sharded periphery GEMVs over gloo (mirrors the reference's
    Allgatherv + row-distributed dense ops, periphery.cpp:21-47,422-442)."""
    N = 37  # 3N = 111 rows, uneven split across 2 ranks
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "pg2")
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_shell_worker, args=(r, WORLD, init_file, N, q))
                 for r in range(WORLD)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(WORLD):
            rank, pre, mv = q.get(timeout=150)
            results[rank] = (pre, mv)
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0
    rng = np.random.default_rng(77)
    M = rng.uniform(-1, 1, (3 * N, 3 * N))
    x = rng.uniform(-1, 1, 3 * N)
    v = rng.uniform(-1, 1, 3 * N)
    pre = np.concatenate([results[r][0] for r in range(WORLD)])
    mv = np.concatenate([results[r][1] for r in range(WORLD)])
    assert np.allclose(pre, M @ x, rtol=1e-13, atol=1e-13)
    assert np.allclose(mv, M @ x + v, rtol=1e-13, atol=1e-13)


@pytest.mark.timeout(180)
def test_sharded_evaluator_matches_single_process_gloo():
    """world_size=2 over gloo: sharded evaluation must equal the
    single-process oracle evaluation bit-for-bit reassembled."""
    n_src, n_trg = 301, 157  # deliberately not divisible by world size
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "pg")
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_worker, args=(r, WORLD, init_file, n_src, n_trg, q))
                 for r in range(WORLD)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(WORLD):
            rank, u = q.get(timeout=150)
            results[rank] = u
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0

    import oracle
    rng = np.random.default_rng(100)
    r_src = rng.uniform(-1, 1, (n_src, 3))
    f_src = rng.uniform(-1, 1, (n_src, 3))
    r_trg = rng.uniform(-1, 1, (n_trg, 3))
    u_ref = oracle.stokeslet(r_src, f_src, r_trg, 1.3)
    u_sharded = np.vstack([results[r] for r in range(WORLD)])
    assert np.array_equal(u_sharded, u_ref)
