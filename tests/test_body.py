"""CPU tests of rigid spherical bodies (skellysim_amd/body.py) against the
reference model (src/core/body_spherical.cpp, body_container.cpp) and
classical mobility physics, through the oracle backend."""

import os

import numpy as np
import pytest

from skellysim_amd.body import (SphericalBody, quat_mult, quat_to_rot,
                                calculate_link_conditions)
from skellysim_amd.system_fd import SystemFD
from skellysim_amd.fiber_fd import FiberFD
from oracle_backend import OracleBackend

HERE = os.path.dirname(os.path.abspath(__file__))


def sphere_fixture():
    """Body geometry from the reference-precompute sphere fixture; the
    periphery convention stores INWARD normals (precompute.py:80-81), a
    body's are outward."""
    fx = np.load(os.path.join(HERE, "golden", "periphery_sphere_192.npz"))
    return (fx["nodes"], -fx["normals"], fx["quadrature_weights"].reshape(-1),
            float(fx["radius"]))


def make_body(**kw):
    nodes, normals, w, R = sphere_fixture()
    return SphericalBody(nodes, normals, w, R, **kw)


def test_quaternion_helpers():
    from scipy.spatial.transform import Rotation
    rng = np.random.default_rng(0)
    for _ in range(5):
        q = rng.standard_normal(4)
        q /= np.linalg.norm(q)
        # scipy uses (x, y, z, w)
        ref = Rotation.from_quat([q[1], q[2], q[3], q[0]]).as_matrix()
        assert np.allclose(quat_to_rot(q), ref, atol=1e-14)
        p = rng.standard_normal(4)
        p /= np.linalg.norm(p)
        refm = (Rotation.from_quat([q[1], q[2], q[3], q[0]])
                * Rotation.from_quat([p[1], p[2], p[3], p[0]])).as_matrix()
        assert np.allclose(quat_to_rot(quat_mult(q, p)), refm, atol=1e-13)


def test_body_matvec_matches_dense_operator():
    """The dense preconditioner block IS the exact per-body operator
    (body_spherical.cpp:104-127): matvec with the self double-layer flow in
    v must reproduce A @ x."""
    eta = 1.3
    b = make_body(position=(0.3, -0.2, 0.5))
    be = OracleBackend()
    b.update_cache(eta, be)
    rng = np.random.default_rng(1)
    x = rng.uniform(-1, 1, b.solution_size)
    dens = x[: 3 * b.n_nodes].reshape(-1, 3)
    v_self = be.stresslet_normal_density(b.nodes, b.normals, dens, b.nodes, eta)
    mv = b.matvec(v_self, x)
    ref = b._A_dense @ x
    rel = np.linalg.norm(mv - ref) / np.linalg.norm(ref)
    assert rel < 1e-12, rel


def test_body_step_kinematics():
    b = make_body()
    n3 = 3 * b.n_nodes
    sol = np.zeros(b.solution_size)
    sol[n3: n3 + 3] = [0.1, -0.2, 0.3]        # U
    b.step(0.5, sol)
    assert np.allclose(b.position, [0.05, -0.1, 0.15])   # translates even
    assert np.allclose(b.orientation, [1, 0, 0, 0])      # with zero omega
    sol[n3 + 3: n3 + 6] = [0.0, 0.0, np.pi]   # rotate about z
    p0 = b.nodes[0] - b.position
    b.step(1.0, sol)
    # half-turn about z: node positions flip in x, y
    p1 = b.nodes[0] - b.position
    assert np.allclose(p1, [-p0[0], -p0[1], p0[2]], atol=1e-12)


@pytest.mark.timeout(300)
def test_isolated_body_mobility():
    """Classical sphere mobility: external force F -> U = F/(6 pi eta R),
    torque T -> omega = T/(8 pi eta R^3). The center Stokeslet/rotlet
    completion makes both exact in the continuous limit; tolerance reflects
    the 192-node quadrature."""
    eta, F = 1.3, np.array([0.4, -0.2, 0.7])
    b = make_body(external_force=F)
    s = SystemFD([], eta=eta, dt=0.1, bodies=[b], backend=OracleBackend())
    info = s.solve(tol=1e-12, maxiter=100)
    assert info["converged"], info
    n3 = 3 * b.n_nodes
    U = s.solution[n3: n3 + 3]
    w = s.solution[n3 + 3: n3 + 6]
    U_ref = F / (6 * np.pi * eta * b.radius)
    # translation: quadrature-limited (~3e-5 at 192 nodes); converges in
    # ONE GMRES iteration (the dense block preconditioner is exact)
    assert info["iters"] <= 3
    assert np.linalg.norm(U - U_ref) / np.linalg.norm(U_ref) < 2e-4
    assert np.linalg.norm(w) < 1e-4 * np.linalg.norm(U_ref) / b.radius

    T = np.array([-0.3, 0.5, 0.2])
    b2 = make_body(external_torque=T)
    s2 = SystemFD([], eta=eta, dt=0.1, bodies=[b2], backend=OracleBackend())
    assert s2.solve(tol=1e-12, maxiter=100)["converged"]
    w2 = s2.solution[n3 + 3: n3 + 6]
    w_ref = T / (8 * np.pi * eta * b2.radius ** 3)
    # rotation: the center rotlet's surface field IS the rigid rotation, so
    # this is exact to roundoff (measured 3e-16)
    assert np.linalg.norm(w2 - w_ref) / np.linalg.norm(w_ref) < 1e-12


@pytest.mark.timeout(600)
def test_two_body_interaction():
    """Two equal spheres driven along their line of centers move faster
    than isolated ones; the speed-up matches the method-of-reflections
    series U/U0 = 1 + (3/2)(R/d) - (R/d)^3 + O((R/d)^4). This exercises the
    body-body hydrodynamic coupling through the full solve (measured:
    1.3472 vs 1.3594 predicted at d=4R; 1.2426 vs 1.2454 at d=6R)."""
    nodes, normals, w, R = sphere_fixture()
    eta = 1.0
    F = np.array([0.0, 0.0, 1.0])
    U0 = 1.0 / (6 * np.pi * eta * R)
    for dfac, tol in ((4.0, 0.015), (6.0, 0.005)):
        d = dfac * R
        bs = [SphericalBody(nodes, normals, w, R, position=(0, 0, z),
                            external_force=F) for z in (0.0, d)]
        s = SystemFD([], eta=eta, dt=0.1, bodies=bs, backend=OracleBackend())
        info = s.solve(tol=1e-12, maxiter=200)
        assert info["converged"] and info["iters"] <= 10, info
        lam = R / d
        pred = 1 + 1.5 * lam - lam ** 3
        Uz = []
        off = 0
        for b in bs:
            n3 = 3 * b.n_nodes
            Uz.append(s.solution[off + n3 + 2])
            off += b.solution_size
        assert abs(Uz[0] - Uz[1]) < 1e-4 * U0          # symmetric pair
        assert abs(Uz[0] / U0 - pred) < tol * pred, (Uz[0] / U0, pred)


def test_link_conditions_formulas():
    """body_container.cpp:171-268 restated: check F/L on the body and the
    velocity rows on the fiber against directly-written expressions."""
    rng = np.random.default_rng(3)
    b = make_body(position=(0.0, 0.0, 0.0),
                  nucleation_sites_ref=np.array([[0.5, 0.0, 0.0]]))
    n = 16
    s = np.linspace(0, 1.0, n)
    x = b.nucleation_sites[0][None, :] + s[:, None] * np.array([1.0, 0, 0])
    fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-3, eta=1.0,
                  minus_clamped=True)
    fib.binding_site = (0, 0)
    fib.update_constants(1.0)
    fib.update_derivatives()

    x_fib = rng.uniform(-1, 1, 4 * n)
    body_vels = rng.uniform(-1, 1, (1, 6))
    vof, ft = calculate_link_conditions([fib], x_fib, body_vels, [b])

    m = fib.mats
    L, E = fib.length, fib.bending_rigidity
    x_new = x_fib[: 3 * n].reshape(3, n)
    T0 = x_fib[3 * n]
    xs0 = fib.xs[:, 0]
    xss0 = (2 / L) ** 2 * (x_new @ m["D_2_0"])[:, 0]
    xsss0 = (2 / L) ** 3 * (x_new @ m["D_3_0"])[:, 0]
    site = b.nucleation_sites[0] - b.position
    F_ref = -E * xsss0 + xs0 * T0
    L_ref = (-E * np.cross(site, xsss0) + np.cross(site, xs0) * T0
             + E * np.cross(xs0, xss0))
    assert np.allclose(ft[0, 0:3], F_ref, atol=1e-13)
    assert np.allclose(ft[0, 3:6], L_ref, atol=1e-13)

    U, w = body_vels[0, 0:3], body_vels[0, 3:6]
    assert np.allclose(vof[0, 0:3], -U - np.cross(w, site), atol=1e-14)
    assert np.isclose(vof[0, 3], -xs0 @ U + np.cross(xs0, site) @ w)
    assert np.allclose(vof[0, 4:7],
                       np.cross(site / np.linalg.norm(site), w), atol=1e-14)

    # unattached fibers get zero rows and do not shift the solution offset
    fib2 = FiberFD(x + 2.0, length=1.0, bending_rigidity=2.5e-3, eta=1.0)
    fib2.update_constants(1.0)
    fib2.update_derivatives()
    x2 = rng.uniform(-1, 1, 8 * n)
    x2[4 * n:] = x_fib
    vof2, ft2 = calculate_link_conditions([fib2, fib], x2, body_vels, [b])
    assert np.allclose(vof2[0], 0.0)
    assert np.allclose(vof2[1], vof[0])
    assert np.allclose(ft2, ft)


@pytest.mark.timeout(600)
def test_coupled_fiber_body_solve():
    """A motor-forced fiber attached to a body: the coupled solve converges
    and pushes the body; with the motor off and no external force the system
    stays (numerically) quiescent."""
    nodes, normals, w, R = sphere_fixture()
    site_ref = np.array([[1.1 * R, 0.0, 0.0]])  # off-surface attachment
    n = 16
    for motor, expect_motion in ((-0.05, True), (0.0, False)):
        b = SphericalBody(nodes, normals, w, R,
                          nucleation_sites_ref=site_ref)
        s0 = np.linspace(0, 1.0, n)
        x = b.nucleation_sites[0][None, :] + s0[:, None] * np.array([1.0, 0, 0])
        fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-3, eta=1.0,
                      minus_clamped=True, force_scale=motor)
        fib.binding_site = (0, 0)
        sys_ = SystemFD([fib], eta=1.0, dt=0.05, bodies=[b],
                        backend=OracleBackend())
        info = sys_.step(tol=1e-11, maxiter=300, restart=150)
        assert info["converged"], (motor, info)
        U = np.linalg.norm(b.velocity)
        if expect_motion:
            assert U > 1e-5, U
        else:
            assert U < 1e-10, U
        assert np.isfinite(fib.x).all() and np.isfinite(b.position).all()
