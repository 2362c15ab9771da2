"""GPU test of the rigid-body solve on the product (HIP) path: the same
mobility anchors as tests/test_body.py, with every kernel leg (dense
stresslet_times_normal, stresslet double layer, center stokeslet/rotlet)
served by the HIP extension."""

import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

HERE = os.path.dirname(os.path.abspath(__file__))


def test_isolated_body_mobility_hip(hip_lib_path):
    from skellysim_amd.body import SphericalBody
    from skellysim_amd.system_fd import SystemFD, HipBackend

    fx = np.load(os.path.join(HERE, "golden", "periphery_sphere_192.npz"))
    eta, F = 1.3, np.array([0.4, -0.2, 0.7])
    T = np.array([-0.3, 0.5, 0.2])
    b = SphericalBody(fx["nodes"], -fx["normals"],
                      fx["quadrature_weights"].reshape(-1), float(fx["radius"]),
                      external_force=F, external_torque=T)
    s = SystemFD([], eta=eta, dt=0.1, bodies=[b], backend=HipBackend())
    info = s.solve(tol=1e-12, maxiter=100)
    assert info["converged"] and info["iters"] <= 3, info
    n3 = 3 * b.n_nodes
    U = s.solution[n3: n3 + 3]
    w = s.solution[n3 + 3: n3 + 6]
    U_ref = F / (6 * np.pi * eta * b.radius)
    w_ref = T / (8 * np.pi * eta * b.radius ** 3)
    assert np.linalg.norm(U - U_ref) / np.linalg.norm(U_ref) < 2e-4
    # combined F+T load: the force's quadrature error couples ~1.3e-7 abs
    # into omega (same value on the CPU path); pure-torque rotation is
    # roundoff-exact (tests/test_body.py)
    assert np.linalg.norm(w - w_ref) / np.linalg.norm(w_ref) < 1e-4


def test_coupled_fiber_body_hip_matches_oracle(hip_lib_path):
    """One coupled fiber+body step: HIP backend equals the oracle backend to
    the GMRES tolerance."""
    from skellysim_amd.body import SphericalBody
    from skellysim_amd.fiber_fd import FiberFD
    from skellysim_amd.system_fd import SystemFD, HipBackend
    from oracle_backend import OracleBackend

    fx = np.load(os.path.join(HERE, "golden", "periphery_sphere_192.npz"))
    R = float(fx["radius"])

    def build(backend):
        b = SphericalBody(fx["nodes"], -fx["normals"],
                          fx["quadrature_weights"].reshape(-1), R,
                          nucleation_sites_ref=np.array([[1.1 * R, 0.0, 0.0]]))
        s0 = np.linspace(0, 1.0, 16)
        x = b.nucleation_sites[0][None, :] + s0[:, None] * np.array([1.0, 0, 0])
        fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-3, eta=1.0,
                      minus_clamped=True, force_scale=-0.05)
        fib.binding_site = (0, 0)
        return SystemFD([fib], eta=1.0, dt=0.05, bodies=[b], backend=backend)

    s_hip = build(HipBackend())
    info = s_hip.solve(tol=1e-11, maxiter=300, restart=150)
    assert info["converged"], info
    s_cpu = build(OracleBackend())
    assert s_cpu.solve(tol=1e-11, maxiter=300, restart=150)["converged"]
    rel = np.linalg.norm(s_hip.solution - s_cpu.solution) / \
        np.linalg.norm(s_cpu.solution)
    assert rel < 1e-8, rel


def test_full_composition_hip_matches_oracle(hip_lib_path):
    """Shell + body + attached fiber on the HIP backend equals the oracle
    backend to the GMRES tolerance (the full system.cpp matvec composition
    with every kernel leg on device)."""
    from skellysim_amd.body import SphericalBody
    from skellysim_amd.fiber_fd import FiberFD
    from skellysim_amd.system_fd import SystemFD, Shell, HipBackend
    from oracle_backend import OracleBackend

    fx = np.load(os.path.join(HERE, "golden", "periphery_sphere_192.npz"))
    R = float(fx["radius"])

    def build(backend):
        shell = Shell(fx["nodes"] * 4.0, fx["normals"],
                      fx["stresslet_plus_complementary"], fx["M_inv"])
        b = SphericalBody(fx["nodes"], -fx["normals"],
                          fx["quadrature_weights"].reshape(-1), R,
                          nucleation_sites_ref=np.array([[1.1 * R, 0.0, 0.0]]))
        s0 = np.linspace(0, 1.0, 16)
        x = b.nucleation_sites[0][None, :] + s0[:, None] * np.array([1.0, 0, 0])
        fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-3, eta=1.0,
                      minus_clamped=True, force_scale=-0.05)
        fib.binding_site = (0, 0)
        return SystemFD([fib], eta=1.0, dt=0.05, shell=shell, bodies=[b],
                        backend=backend)

    s_hip = build(HipBackend())
    info = s_hip.solve(tol=1e-11, maxiter=300, restart=150)
    assert info["converged"], info
    s_cpu = build(OracleBackend())
    assert s_cpu.solve(tol=1e-11, maxiter=300, restart=150)["converged"]
    rel = np.linalg.norm(s_hip.solution - s_cpu.solution) / \
        np.linalg.norm(s_cpu.solution)
    assert rel < 1e-8, rel


def test_device_resident_bodies_match_host_path(hip_lib_path):
    """the fully device-resident GMRES iteration
    WITH BODY BLOCKS (torch link conditions, body flow and body rows inside
    _apply_matvec_device) must match the host matvec loop on real device
    tensors, for the full shell+body+fiber composition AND for a 3-step
    solve-adopt-reprep run."""
    from skellysim_amd.body import SphericalBody
    from skellysim_amd.fiber_fd import FiberFD
    from skellysim_amd.system_fd import SystemFD, Shell, HipBackend
    import torch

    fx = np.load(os.path.join(HERE, "golden", "periphery_sphere_192.npz"))
    R = float(fx["radius"])

    def build():
        shell = Shell(fx["nodes"] * 4.0, fx["normals"],
                      fx["stresslet_plus_complementary"], fx["M_inv"])
        b = SphericalBody(fx["nodes"], -fx["normals"],
                          fx["quadrature_weights"].reshape(-1), R,
                          nucleation_sites_ref=np.array([[1.1 * R, 0.0, 0.0]]),
                          external_force=(0.0, 0.0, 0.2))
        s0 = np.linspace(0, 1.0, 16)
        x = b.nucleation_sites[0][None, :] + s0[:, None] * np.array([1.0, 0, 0])
        fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-3, eta=1.0,
                      minus_clamped=True, force_scale=-0.05)
        fib.binding_site = (0, 0)
        return SystemFD([fib], eta=1.0, dt=0.05, shell=shell, bodies=[b],
                        backend=HipBackend())

    # single solve, device-resident vs host
    s_dev = build()
    i_dev = s_dev.solve(tol=1e-11, maxiter=300, restart=150, device_mode=True)
    assert i_dev["converged"], i_dev
    s_host = build()
    i_host = s_host.solve(tol=1e-11, maxiter=300, restart=150,
                          device_mode=False)
    assert i_host["converged"], i_host
    rel = np.linalg.norm(s_dev.solution - s_host.solution) / \
        np.linalg.norm(s_host.solution)
    assert rel < 1e-8, rel

    # 3-step run: positions must track to solver tolerance
    def adopt(s):
        for f, a, bnd in s._fiber_slices():
            f.step(s.solution[a:bnd])
        for b_, a, bb in s._body_sol_slices():
            b_.step(s.dt, s.solution[a:bb])
        s.repin_to_bodies()

    s_dev, s_host = build(), build()
    for _ in range(3):
        assert s_dev.solve(tol=1e-11, maxiter=300, restart=150,
                           device_mode=True)["converged"]
        adopt(s_dev)
        assert s_host.solve(tol=1e-11, maxiter=300, restart=150,
                            device_mode=False)["converged"]
        adopt(s_host)
    dx = np.linalg.norm(s_dev.fibers[0].x - s_host.fibers[0].x)
    db = np.linalg.norm(s_dev.bodies[0].position - s_host.bodies[0].position)
    assert dx < 1e-8 and db < 1e-8, (dx, db)
