"""gloo world-2 test of the distributed fiber+shell solve
(DistributedSystemFD): block-distributed fibers and shell rows, all-gathered
sources, distributed GMRES dots — reassembled solution must match the
single-process solve."""

import os
import tempfile

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

WORLD = 2


def _make_problem():
    from skellysim_amd.fiber_fd import FiberFD
    here = os.path.dirname(os.path.abspath(__file__))
    fx = np.load(os.path.join(here, "golden", "periphery_sphere_192.npz"))
    rng = np.random.default_rng(77)
    fibers = []
    for k in range(5):
        d = rng.uniform(-1, 1, 3)
        x0 = rng.uniform(-0.25, 0.25, 3)
        d /= np.linalg.norm(d)
        s = np.linspace(0, 0.5, 32)
        x = x0[None, :] + s[:, None] * d[None, :]
        fibers.append(FiberFD(x, length=0.5, bending_rigidity=2.5e-3, eta=1.0,
                              minus_clamped=(k % 2 == 0), force_scale=-0.05))
    U = np.array([0.05, 0.02, -0.04])
    bg = lambda r: np.tile(U, (len(r), 1))
    return fx, fibers, bg


def _dist_worker(rank, world, init_file, q, device_mode=False):
    import torch.distributed as dist
    from skellysim_amd.system_dist import DistributedSystemFD, distribute_fibers
    from skellysim_amd.system_fd import Shell
    from oracle_backend import (OracleBackend, FakeDeviceBackend,
                                patch_device_kernels_with_oracle)
    from skellysim_amd.sharded import shard_range

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        if device_mode:
            patch_device_kernels_with_oracle()
            backend = FakeDeviceBackend()
        else:
            backend = OracleBackend()
        fx, fibers, bg = _make_problem()
        N = len(fx["nodes"])
        a, b = shard_range(N, world, rank)
        A_rows = fx["stresslet_plus_complementary"][3 * a: 3 * b]
        M_rows = fx["M_inv"][3 * a: 3 * b]
        shell = Shell(fx["nodes"], fx["normals"], A_rows, M_rows)
        my_fibers = distribute_fibers(fibers, world, rank)
        sys_ = DistributedSystemFD(my_fibers, eta=1.0, dt=0.05, shell=shell,
                                   shell_rows=(a, b), backend=backend,
                                   background_flow=bg)
        info = sys_.solve(tol=1e-11, maxiter=300, restart=100,
                          device_mode=device_mode or None)
        q.put((rank, sys_.solution, sys_.fiber_sol_size, info["converged"]))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("device_mode", [False, True],
                         ids=["host", "device-layout"])
def test_distributed_solve_matches_single_process(device_mode):
    """device-layout runs the device-resident distributed iteration
    (system_dist._apply_matvec_device) on CPU tensors with oracle-backed
    kernel fakes — the orchestration (gathers, batched blocks, row-block
    GEMVs, distributed GMRES on tensors) is what's under test; the HIP
    kernels themselves are covered by the gpu-marked suites."""
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "pg")
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_dist_worker,
                             args=(r, WORLD, init_file, q, device_mode))
                 for r in range(WORLD)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(WORLD):
            rank, sol, fib_size, conv = q.get(timeout=250)
            assert conv
            results[rank] = (sol, fib_size)
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0

    # single-process reference solve
    from skellysim_amd.system_fd import SystemFD, Shell
    from oracle_backend import OracleBackend
    fx, fibers, bg = _make_problem()
    shell = Shell(fx["nodes"], fx["normals"], fx["stresslet_plus_complementary"],
                  fx["M_inv"])
    sys_ = SystemFD(fibers, eta=1.0, dt=0.05, shell=shell,
                    backend=OracleBackend(), background_flow=bg)
    info = sys_.solve(tol=1e-11, maxiter=300, restart=100)
    assert info["converged"]

    # reassemble: [r0_fib | r1_fib | r0_shell | r1_shell]
    fib_parts, shell_parts = [], []
    for r in range(WORLD):
        sol, fib_size = results[r]
        fib_parts.append(sol[:fib_size])
        shell_parts.append(sol[fib_size:])
    dist_sol = np.concatenate(fib_parts + shell_parts)
    rel = np.linalg.norm(dist_sol - sys_.solution) / np.linalg.norm(sys_.solution)
    assert rel < 1e-7, rel


def _body_problem():
    from skellysim_amd.body import SphericalBody
    from skellysim_amd.fiber_fd import FiberFD
    here = os.path.dirname(os.path.abspath(__file__))
    fx = np.load(os.path.join(here, "golden", "periphery_sphere_192.npz"))
    R = float(fx["radius"])
    sites = np.array([[1.1 * R, 0.0, 0.0], [0.0, 1.1 * R, 0.0],
                      [-1.1 * R, 0.0, 0.0], [0.0, -1.1 * R, 0.0]])
    body = SphericalBody(fx["nodes"], -fx["normals"],
                         fx["quadrature_weights"].reshape(-1), R,
                         nucleation_sites_ref=sites,
                         external_force=(0.0, 0.0, 0.3))
    fibers = []
    s = np.linspace(0, 0.8, 16)
    for i in range(4):
        site = body.nucleation_sites[i]
        u = site / np.linalg.norm(site)
        x = site[None, :] + s[:, None] * u[None, :]
        f = FiberFD(x, length=0.8, bending_rigidity=2.5e-3, eta=1.0,
                    minus_clamped=True, force_scale=-0.05)
        f.binding_site = (0, i)
        fibers.append(f)
    return body, fibers


def _dist_body_worker(rank, world, init_file, q):
    import torch.distributed as dist
    from skellysim_amd.system_dist import DistributedSystemFD, distribute_fibers
    from oracle_backend import OracleBackend

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        body, fibers = _body_problem()
        my_fibers = distribute_fibers(fibers, world, rank)
        sys_ = DistributedSystemFD(my_fibers, eta=1.0, dt=0.05,
                                   backend=OracleBackend(), bodies=[body])
        info = sys_.step(tol=1e-11, maxiter=300, restart=150)
        q.put((rank, sys_.solution, sys_.fiber_sol_size, info["converged"],
               body.position.copy(), body.velocity.copy(),
               body.orientation.copy()))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_distributed_body_solve_matches_single_process():
    """Bodies in the distributed solve: solution block on rank 0, broadcast
    per apply, link forces all-reduced (body_container.hpp:99,
    system.cpp:309, body_container.cpp:131) — the reassembled solution and
    the stepped body state match the single-process solve on every rank."""
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "pg")
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_dist_body_worker,
                             args=(r, WORLD, init_file, q))
                 for r in range(WORLD)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(WORLD):
            rank, sol, fib_size, conv, pos, vel, quat = q.get(timeout=250)
            assert conv
            results[rank] = (sol, fib_size, pos, vel, quat)
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0

    from skellysim_amd.system_fd import SystemFD
    from oracle_backend import OracleBackend
    body, fibers = _body_problem()
    sys_ = SystemFD(fibers, eta=1.0, dt=0.05, bodies=[body],
                    backend=OracleBackend())
    assert sys_.step(tol=1e-11, maxiter=300, restart=150)["converged"]

    # reassemble [r0_fib | r1_fib | body(from r0)]
    fib_parts = [results[r][0][: results[r][1]] for r in range(WORLD)]
    body_part = results[0][0][results[0][1]:]
    assert len(results[1][0]) == results[1][1]  # rank 1 holds no body block
    dist_sol = np.concatenate(fib_parts + [body_part])
    rel = np.linalg.norm(dist_sol - sys_.solution) / np.linalg.norm(sys_.solution)
    assert rel < 1e-7, rel
    for r in range(WORLD):  # body state consistent on every rank
        _, _, pos, vel, quat = results[r]
        assert np.allclose(pos, body.position, atol=1e-10)
        assert np.allclose(vel, body.velocity, atol=1e-10)
        assert np.allclose(quat, body.orientation, atol=1e-12)


def _dist_triple_worker(rank, world, init_file, q):
    import torch.distributed as dist
    from skellysim_amd.system_dist import DistributedSystemFD, distribute_fibers
    from skellysim_amd.system_fd import Shell
    from skellysim_amd.sharded import shard_range
    from oracle_backend import OracleBackend

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        here = os.path.dirname(os.path.abspath(__file__))
        fx = np.load(os.path.join(here, "golden", "periphery_sphere_192.npz"))
        body, fibers = _body_problem()
        N = len(fx["nodes"])
        a, b = shard_range(N, world, rank)
        shell = Shell(fx["nodes"] * 4.0, fx["normals"],
                      fx["stresslet_plus_complementary"][3 * a: 3 * b],
                      fx["M_inv"][3 * a: 3 * b])
        my_fibers = distribute_fibers(fibers, world, rank)
        sys_ = DistributedSystemFD(my_fibers, eta=1.0, dt=0.05, shell=shell,
                                   shell_rows=(a, b),
                                   backend=OracleBackend(), bodies=[body])
        info = sys_.step(tol=1e-11, maxiter=300, restart=150)
        q.put((rank, sys_.solution, sys_.fiber_sol_size,
               sys_.shell_sol_size, info["converged"]))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_distributed_triple_matches_single_process():
    """Fibers + shell rows + a rank-0 body together (covers the dist
    matvec's shell-to-body-target path and the [fib|shell|body] local
    layouts on both ranks)."""
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "pg")
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_dist_triple_worker,
                             args=(r, WORLD, init_file, q))
                 for r in range(WORLD)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(WORLD):
            rank, sol, fib_sz, sh_sz, conv = q.get(timeout=250)
            assert conv
            results[rank] = (sol, fib_sz, sh_sz)
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0

    from skellysim_amd.system_fd import SystemFD, Shell
    from oracle_backend import OracleBackend
    here = os.path.dirname(os.path.abspath(__file__))
    fx = np.load(os.path.join(here, "golden", "periphery_sphere_192.npz"))
    body, fibers = _body_problem()
    shell = Shell(fx["nodes"] * 4.0, fx["normals"],
                  fx["stresslet_plus_complementary"], fx["M_inv"])
    sys_ = SystemFD(fibers, eta=1.0, dt=0.05, shell=shell, bodies=[body],
                    backend=OracleBackend())
    assert sys_.step(tol=1e-11, maxiter=300, restart=150)["converged"]

    # reassemble [r0_fib | r1_fib | r0_shell | r1_shell | body(from r0)]
    fib_parts, shell_parts = [], []
    for r in range(WORLD):
        sol, fib_sz, sh_sz = results[r]
        fib_parts.append(sol[:fib_sz])
        shell_parts.append(sol[fib_sz: fib_sz + sh_sz])
    body_part = results[0][0][results[0][1] + results[0][2]:]
    dist_sol = np.concatenate(fib_parts + shell_parts + [body_part])
    rel = np.linalg.norm(dist_sol - sys_.solution) / np.linalg.norm(sys_.solution)
    assert rel < 1e-7, rel


def _dist_multistep_worker(rank, world, init_file, q):
    import torch.distributed as dist
    from skellysim_amd.system_dist import DistributedSystemFD, distribute_fibers
    from skellysim_amd.system_fd import Shell
    from skellysim_amd.sharded import shard_range
    from oracle_backend import OracleBackend

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        fx, fibers, bg = _make_problem()
        N = len(fx["nodes"])
        a, b = shard_range(N, world, rank)
        shell = Shell(fx["nodes"], fx["normals"],
                      fx["stresslet_plus_complementary"][3 * a: 3 * b],
                      fx["M_inv"][3 * a: 3 * b])
        my_fibers = distribute_fibers(fibers, world, rank)
        sys_ = DistributedSystemFD(my_fibers, eta=1.0, dt=0.05, shell=shell,
                                   shell_rows=(a, b), backend=OracleBackend(),
                                   background_flow=bg)
        for _ in range(3):
            info = sys_.step(tol=1e-11, maxiter=300, restart=100)
            assert info["converged"]
        q.put((rank, [f.x for f in my_fibers]))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_distributed_multistep_matches_single_process():
    """Three full timesteps (solve -> adopt -> re-prep) in world-2 land on
    the same fiber positions as the single-process run — the distributed
    step loop, not just one solve."""
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "pg")
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_dist_multistep_worker,
                             args=(r, WORLD, init_file, q))
                 for r in range(WORLD)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(WORLD):
            rank, xs = q.get(timeout=250)
            results[rank] = xs
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0

    from skellysim_amd.system_fd import SystemFD, Shell
    from skellysim_amd.sharded import shard_range
    from oracle_backend import OracleBackend
    fx, fibers, bg = _make_problem()
    shell = Shell(fx["nodes"], fx["normals"],
                  fx["stresslet_plus_complementary"], fx["M_inv"])
    sys_ = SystemFD(fibers, eta=1.0, dt=0.05, shell=shell,
                    backend=OracleBackend(), background_flow=bg)
    for _ in range(3):
        info = sys_.step(tol=1e-11, maxiter=300, restart=100)
        assert info["converged"]

    dist_xs = []
    for r in range(WORLD):
        dist_xs.extend(results[r])
    assert len(dist_xs) == len(fibers)
    for xd, f in zip(dist_xs, sys_.fibers):
        rel = np.linalg.norm(xd - f.x) / max(np.linalg.norm(f.x), 1e-30)
        assert rel < 1e-6, rel


def _make_mixed_problem():
    """Mixed-discretization fibers (24/32 nodes) — exercises the
    per-fiber LU fallback on the distributed host preconditioner path
    (the uniform reshape used to crash on these)."""
    from skellysim_amd.fiber_fd import FiberFD
    here = os.path.dirname(os.path.abspath(__file__))
    fx = np.load(os.path.join(here, "golden", "periphery_sphere_192.npz"))
    rng = np.random.default_rng(78)
    fibers = []
    for k, n in enumerate((24, 32, 24, 32)):
        d = rng.uniform(-1, 1, 3)
        x0 = rng.uniform(-0.25, 0.25, 3)
        d /= np.linalg.norm(d)
        s = np.linspace(0, 0.5, n)
        x = x0[None, :] + s[:, None] * d[None, :]
        fibers.append(FiberFD(x, length=0.5, bending_rigidity=2.5e-3, eta=1.0,
                              minus_clamped=(k % 2 == 0), force_scale=-0.05))
    U = np.array([0.05, 0.02, -0.04])
    bg = lambda r: np.tile(U, (len(r), 1))
    return fx, fibers, bg


def _dist_mixed_worker(rank, world, init_file, q):
    import torch.distributed as dist
    from skellysim_amd.system_dist import DistributedSystemFD, distribute_fibers
    from skellysim_amd.system_fd import Shell
    from oracle_backend import OracleBackend
    from skellysim_amd.sharded import shard_range

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        fx, fibers, bg = _make_mixed_problem()
        N = len(fx["nodes"])
        a, b = shard_range(N, world, rank)
        shell = Shell(fx["nodes"], fx["normals"],
                      fx["stresslet_plus_complementary"][3 * a: 3 * b],
                      fx["M_inv"][3 * a: 3 * b])
        my_fibers = distribute_fibers(fibers, world, rank)
        sys_ = DistributedSystemFD(my_fibers, eta=1.0, dt=0.05, shell=shell,
                                   shell_rows=(a, b),
                                   backend=OracleBackend(),
                                   background_flow=bg)
        info = sys_.solve(tol=1e-11, maxiter=300, restart=100)
        q.put((rank, sys_.solution, sys_.fiber_sol_size, info["converged"]))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_distributed_mixed_discretization_matches_single_process():
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "pg")
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_dist_mixed_worker,
                             args=(r, WORLD, init_file, q))
                 for r in range(WORLD)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(WORLD):
            rank, sol, fib_size, conv = q.get(timeout=250)
            assert conv
            results[rank] = (sol, fib_size)
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0

    from skellysim_amd.system_fd import SystemFD, Shell
    from oracle_backend import OracleBackend
    fx, fibers, bg = _make_mixed_problem()
    shell = Shell(fx["nodes"], fx["normals"],
                  fx["stresslet_plus_complementary"], fx["M_inv"])
    sys_ = SystemFD(fibers, eta=1.0, dt=0.05, shell=shell,
                    backend=OracleBackend(), background_flow=bg)
    assert sys_.solve(tol=1e-11, maxiter=300, restart=100)["converged"]

    fib_parts, shell_parts = [], []
    for r in range(WORLD):
        sol, fib_size = results[r]
        fib_parts.append(sol[:fib_size])
        shell_parts.append(sol[fib_size:])
    assembled = np.concatenate(fib_parts + shell_parts)
    rel = np.linalg.norm(assembled - sys_.solution) / \
        np.linalg.norm(sys_.solution)
    assert rel < 1e-9, rel
