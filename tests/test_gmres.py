"""CPU tests of the GMRES solver (torch tensors; same code path runs on GPU).
Checked against direct dense solves and scipy's gmres."""

import os
import tempfile

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from skellysim_amd.gmres import gmres


def make_system(n, seed, cond="easy"):
    rng = np.random.default_rng(seed)
    A = rng.uniform(-1, 1, (n, n))
    A += n * np.eye(n) if cond == "easy" else 2 * np.eye(n)
    b = rng.uniform(-1, 1, n)
    return torch.from_numpy(A), torch.from_numpy(b)


def test_unpreconditioned_matches_direct():
    A, b = make_system(120, 0)
    x, info = gmres(lambda v: A @ v, b, tol=1e-12, maxiter=300, restart=60)
    assert info["converged"], info
    x_ref = torch.linalg.solve(A, b)
    assert torch.norm(x - x_ref) / torch.norm(x_ref) < 1e-9


def test_right_preconditioned_fewer_iters():
    A, b = make_system(200, 1, cond="hard")
    _, info0 = gmres(lambda v: A @ v, b, tol=1e-10, maxiter=400, restart=80)
    # exact inverse as right preconditioner -> converges in O(1) iterations
    Ainv = torch.linalg.inv(A)
    x, info1 = gmres(lambda v: A @ v, b, precond=lambda v: Ainv @ v,
                     tol=1e-10, maxiter=400, restart=80)
    assert info1["converged"]
    assert info1["iters"] < max(3, info0["iters"] // 4)
    x_ref = torch.linalg.solve(A, b)
    assert torch.norm(x - x_ref) / torch.norm(x_ref) < 1e-8


def test_sync_cadence_equivalence():
    """Deferred host bookkeeping (sync_cadence > 1) must produce the exact
    same iterates and residual history as the per-iteration form — the
    device arithmetic is identical, only the transfer schedule differs.
    Surplus basis vectors past convergence are discarded."""
    A, b = make_system(150, 4)
    x1, i1 = gmres(lambda v: A @ v, b, tol=1e-11, maxiter=300, restart=40,
                   sync_cadence=1)
    x8, i8 = gmres(lambda v: A @ v, b, tol=1e-11, maxiter=300, restart=40,
                   sync_cadence=8)
    assert i1["converged"] and i8["converged"]
    assert i1["iters"] == i8["iters"]
    assert torch.equal(x1, x8)
    assert np.allclose(i1["residuals"], i8["residuals"], rtol=0, atol=0)

    # also across a restart boundary with a preconditioner
    A, b = make_system(200, 5, cond="hard")
    M = torch.linalg.inv(A) + 0.05 * torch.from_numpy(
        np.random.default_rng(6).uniform(-1, 1, (200, 200)))
    for cad in (3, 8):
        xc, ic = gmres(lambda v: A @ v, b, precond=lambda v: M @ v,
                       tol=1e-10, maxiter=200, restart=30, sync_cadence=cad)
        x1, i1 = gmres(lambda v: A @ v, b, precond=lambda v: M @ v,
                       tol=1e-10, maxiter=200, restart=30, sync_cadence=1)
        assert ic["iters"] == i1["iters"] and torch.equal(xc, x1)


def test_restart_path():
    rng = np.random.default_rng(2)
    n = 150
    A = torch.from_numpy(rng.uniform(-1, 1, (n, n)) + 10 * np.eye(n))
    b = torch.from_numpy(rng.uniform(-1, 1, n))
    x, info = gmres(lambda v: A @ v, b, tol=1e-10, maxiter=600, restart=25)
    assert info["converged"]
    r = torch.norm(b - A @ x) / torch.norm(b)
    assert float(r) < 1e-9


def test_tolerance_is_relative_to_rhs():
    A, b = make_system(80, 3)
    for scale in (1e-6, 1e6):
        x, info = gmres(lambda v: A @ v, b * scale, tol=1e-10, maxiter=200, restart=40)
        assert info["converged"]
        r = torch.norm(b * scale - A @ x) / torch.norm(b * scale)
        assert float(r) < 1e-9


def test_zero_rhs():
    A, b = make_system(50, 4)
    x, info = gmres(lambda v: A @ v, torch.zeros_like(b))
    assert info["converged"] and torch.all(x == 0)


def test_matches_scipy():
    import scipy.sparse.linalg as spla
    A, b = make_system(100, 5)
    x, info = gmres(lambda v: A @ v, b, tol=1e-11, maxiter=300, restart=50)
    x_sp, code = spla.gmres(A.numpy(), b.numpy(), rtol=1e-11, maxiter=300, restart=50)
    assert code == 0 and info["converged"]
    assert np.linalg.norm(x.numpy() - x_sp) / np.linalg.norm(x_sp) < 1e-7


def _dist_worker(rank, world, init_file, n, q):
    import torch.distributed as dist
    from skellysim_amd.sharded import shard_range, allgather_rows

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        rng = np.random.default_rng(9)
        A = rng.uniform(-1, 1, (n, n)) + n * np.eye(n)
        b = rng.uniform(-1, 1, n)
        a0, a1 = shard_range(n, world, rank)
        A_rows = torch.from_numpy(A[a0:a1].copy())
        b_loc = torch.from_numpy(b[a0:a1].copy())

        def matvec(v_loc):
            v = allgather_rows(v_loc.reshape(-1, 1)).reshape(-1)
            return A_rows @ v

        x_loc, info = gmres(matvec, b_loc, tol=1e-11, maxiter=300, restart=50,
                            distributed=True)
        q.put((rank, x_loc.numpy(), info["converged"]))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_distributed_gmres_gloo():
    """Row-sharded operator + all-reduced dots over gloo (world 2) must match
    the single-process direct solve."""
    n = 121
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "pg3")
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_dist_worker, args=(r, 2, init_file, n, q))
                 for r in range(2)]
        for p in procs:
            p.start()
        res = {}
        for _ in range(2):
            rank, x, conv = q.get(timeout=150)
            assert conv
            res[rank] = x
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0
    rng = np.random.default_rng(9)
    A = rng.uniform(-1, 1, (n, n)) + n * np.eye(n)
    b = rng.uniform(-1, 1, n)
    x = np.concatenate([res[0], res[1]])
    assert np.linalg.norm(A @ x - b) / np.linalg.norm(b) < 1e-9


def test_warm_start_systemfd_matches_cold():
    """SystemFD.solve(warm_start=True) seeds GMRES with the previous
    solution: the converged answer matches the cold-start solve to solver
    tolerance and takes fewer iterations on the second step."""
    import numpy as np
    from skellysim_amd.fiber_fd import FiberFD
    from skellysim_amd.system_fd import SystemFD
    from oracle_backend import OracleBackend

    def build():
        x = np.linspace([0, 0, 0], [0, 0, 1.0], 24)
        f = FiberFD(x, length=1.0, bending_rigidity=2.5e-3, eta=1.0,
                    minus_clamped=True, force_scale=-0.08)
        return SystemFD([f], eta=1.0, dt=0.05, backend=OracleBackend())

    s_cold, s_warm = build(), build()
    # step 1: identical (warm has no previous solution yet)
    i1c = s_cold.step(tol=1e-11, maxiter=200, restart=100)
    s_warm.solve(tol=1e-11, maxiter=200, restart=100, warm_start=True)
    for f, a, b in s_warm._fiber_slices():
        f.step(s_warm.solution[a:b])
    s_warm.repin_to_bodies()
    assert np.array_equal(s_cold.solution, s_warm.solution)

    # step 2: warm start converges to the same solution in fewer iterations
    i2c = s_cold.solve(tol=1e-11, maxiter=200, restart=100)
    i2w = s_warm.solve(tol=1e-11, maxiter=200, restart=100, warm_start=True)
    assert i2c["converged"] and i2w["converged"]
    assert i2w["iters"] <= i2c["iters"], (i2w["iters"], i2c["iters"])
    # the warm start's initial residual is far below the cold start's
    assert i2w["residuals"][0] < 1e-2 * i2c["residuals"][0], \
        (i2w["residuals"][0], i2c["residuals"][0])
    rel = np.linalg.norm(s_warm.solution - s_cold.solution) / \
        np.linalg.norm(s_cold.solution)
    assert rel < 1e-8, rel
