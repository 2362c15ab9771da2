"""GPU tests of the full solve pipeline with the HIP backend (the product
path: pair kernels, batched self-stokeslet build, batched LU, shell GEMVs on
device)."""

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def hip_backend(hip_lib_path):
    from skellysim_amd.system_fd import HipBackend
    return HipBackend()


def straight_fiber(n=32, length=1.0, direction=(0, 0, 1.0), x0=(0, 0, 0), **kw):
    from skellysim_amd.fiber_fd import FiberFD
    d = np.asarray(direction, float)
    d /= np.linalg.norm(d)
    s = np.linspace(0, length, n)
    x = np.asarray(x0, float)[None, :] + s[:, None] * d[None, :]
    return FiberFD(x, length=length, bending_rigidity=2.5e-3, eta=1.0, **kw)


def test_free_fiber_advection_hip(hip_backend):
    """The reference's tier-3 physics anchor on the product path."""
    from skellysim_amd.system_fd import SystemFD
    U = np.array([0.1, -0.05, 0.02])
    fib = straight_fiber(n=32)
    x0 = fib.x.copy()
    dt = 0.1
    sys_ = SystemFD([fib], eta=1.0, dt=dt, backend=hip_backend,
                    background_flow=lambda r: np.tile(U, (len(r), 1)))
    K = 3
    for _ in range(K):
        info = sys_.step(tol=1e-12, maxiter=300)
        assert info["converged"], info
    err = np.abs(fib.x - (x0 + U[:, None] * (K * dt))).max()
    assert err < 1e-10, err


def test_device_matvec_matches_host_matvec(hip_backend):
    """The device-resident batched matvec/preconditioner must equal the
    per-fiber host path on the same state."""
    import os
    from skellysim_amd.system_fd import SystemFD, Shell

    here = os.path.dirname(os.path.abspath(__file__))
    fx = np.load(os.path.join(here, "golden", "periphery_sphere_192.npz"))
    shell = Shell(fx["nodes"], fx["normals"], fx["stresslet_plus_complementary"],
                  fx["M_inv"])
    rng = np.random.default_rng(3)
    fibs = [straight_fiber(n=32, length=0.5, direction=rng.uniform(-1, 1, 3),
                           x0=rng.uniform(-0.25, 0.25, 3),
                           minus_clamped=(k % 2 == 0))
            for k in range(6)]
    sys_ = SystemFD(fibs, eta=1.3, dt=0.05, shell=shell, backend=hip_backend,
                    background_flow=lambda r: np.tile([0.05, 0, 0], (len(r), 1)))
    sys_.prep_state_for_solver()
    sys_._build_device_operators()
    x = rng.uniform(-1, 1, sys_.fiber_sol_size + sys_.shell_sol_size)
    x_t = hip_backend._t(x)
    mv_dev = sys_._apply_matvec_device(x_t).cpu().numpy()
    mv_host = sys_.apply_matvec(x)
    rel = np.linalg.norm(mv_dev - mv_host) / np.linalg.norm(mv_host)
    assert rel < 1e-12, rel
    pc_dev = sys_._apply_precond_device(x_t).cpu().numpy()
    pc_host = sys_.apply_preconditioner(x)
    rel = np.linalg.norm(pc_dev - pc_host) / np.linalg.norm(pc_host)
    assert rel < 1e-10, rel


def test_distributed_device_solve_world1(hip_backend):
    """DistributedSystemFD's device-resident iteration on the real HIP path
    (world 1: collectives are identities, everything else — gathered-source
    matvec, row-block shell GEMVs, device GMRES — is the multi-GPU code)."""
    import os
    from skellysim_amd.system_fd import SystemFD, Shell
    from skellysim_amd.system_dist import DistributedSystemFD

    here = os.path.dirname(os.path.abspath(__file__))
    fx = np.load(os.path.join(here, "golden", "periphery_sphere_192.npz"))
    shell = Shell(fx["nodes"], fx["normals"], fx["stresslet_plus_complementary"],
                  fx["M_inv"])
    U = np.array([0.05, 0.02, -0.04])
    bg = lambda r: np.tile(U, (len(r), 1))

    def make_fibers():
        rng = np.random.default_rng(21)
        return [straight_fiber(n=32, length=0.5,
                               direction=rng.uniform(-1, 1, 3),
                               x0=rng.uniform(-0.25, 0.25, 3),
                               minus_clamped=(k % 2 == 0))
                for k in range(5)]

    N = len(fx["nodes"])
    sys_d = DistributedSystemFD(make_fibers(), eta=1.0, dt=0.05, shell=shell,
                                shell_rows=(0, N), backend=hip_backend,
                                background_flow=bg)
    info = sys_d.solve(tol=1e-11, maxiter=300, device_mode=True)
    assert info["converged"], info

    sys_s = SystemFD(make_fibers(), eta=1.0, dt=0.05, shell=shell,
                     backend=hip_backend, background_flow=bg)
    info2 = sys_s.solve(tol=1e-11, maxiter=300)
    assert info2["converged"], info2

    rel = np.linalg.norm(sys_d.solution - sys_s.solution) / \
        np.linalg.norm(sys_s.solution)
    assert rel < 1e-8, rel


def test_hip_matches_oracle_backend_one_solve(hip_backend):
    """One multi-fiber + small-shell solve: HIP backend equals the oracle
    backend to the GMRES tolerance."""
    import os
    from skellysim_amd.system_fd import SystemFD, Shell
    from oracle_backend import OracleBackend
    from skellysim_amd.fiber_fd import FiberFD

    here = os.path.dirname(os.path.abspath(__file__))
    fx = np.load(os.path.join(here, "golden", "periphery_sphere_192.npz"))
    shell_np = Shell(fx["nodes"], fx["normals"], fx["stresslet_plus_complementary"],
                     fx["M_inv"])

    rng = np.random.default_rng(7)

    def make_fibers():
        fibs = []
        for k in range(4):
            d = rng.uniform(-1, 1, 3)
            x0 = rng.uniform(-0.25, 0.25, 3)
            fibs.append(straight_fiber(n=32, length=0.5, direction=d, x0=x0))
        return fibs

    U = np.array([0.05, 0.02, -0.04])
    bg = lambda r: np.tile(U, (len(r), 1))

    rng = np.random.default_rng(7)
    sys_hip = SystemFD(make_fibers(), eta=1.0, dt=0.05, shell=shell_np,
                       backend=hip_backend, background_flow=bg)
    info = sys_hip.solve(tol=1e-10, maxiter=300)
    assert info["converged"], info

    rng = np.random.default_rng(7)
    sys_cpu = SystemFD(make_fibers(), eta=1.0, dt=0.05, shell=shell_np,
                       backend=OracleBackend(), background_flow=bg)
    info2 = sys_cpu.solve(tol=1e-10, maxiter=300)
    assert info2["converged"], info2

    rel = np.linalg.norm(sys_hip.solution - sys_cpu.solution) / \
        np.linalg.norm(sys_cpu.solution)
    assert rel < 1e-8, rel
