"""CPU tests of the finite-difference fiber model and the SystemFD
orchestration (oracle backend). The end-to-end physics anchor mirrors the
reference's own tier-3 test: a free fiber advected by a uniform background
flow moves rigidly with the flow
(tests/combined/test_fiber_uniform_background.py, rel err < 1e-13 there)."""

import numpy as np
import pytest

from skellysim_amd.fiber_fd import FiberFD, finite_diff, barycentric_matrix, fib_mats
from skellysim_amd.system_fd import SystemFD
from oracle_backend import OracleBackend


def test_finite_diff_polynomial_exactness():
    """Fornberg weights (utils.cpp:48-105) are exact on polynomials up to the
    stencil order."""
    s = np.linspace(-1, 1, 32)
    for M, n_s in [(1, 5), (2, 6), (3, 7), (4, 8)]:
        D = finite_diff(s, M, n_s)
        for deg in range(M, n_s - 1):
            p = s ** deg
            dp = D @ p
            c = 1.0
            for k in range(M):
                c *= (deg - k)
            ref = c * s ** (deg - M)
            assert np.allclose(dp, ref, atol=1e-7 * max(1, abs(ref).max())), (M, deg)


def test_barycentric_resample():
    """The reference's barycentric_matrix (utils.cpp:12-36) uses Berrut
    rational weights (+-1, halved endpoints) on the uniform alpha grid:
    exact for constants and linears, approximate for smooth functions."""
    x = np.linspace(-1, 1, 24)
    y = 2 * (0.5 + np.arange(20)) / 20 - 1
    P = barycentric_matrix(x, y)
    assert np.allclose(P @ np.ones_like(x), 1.0, atol=1e-13)
    assert np.allclose(P @ x, y, atol=1e-13)
    assert np.abs(P @ np.sin(x) - np.sin(y)).max() < 1e-3


def test_fib_mats_shapes():
    m = fib_mats(32)
    assert m["D_1_0"].shape == (32, 32)
    assert m["P_downsample_bc"].shape == (4 * 32 - 14, 4 * 32)
    assert np.isclose(m["weights_0"].sum(), 2.0)  # integrates alpha in [-1,1]


def straight_fiber(n=32, length=1.0, direction=(0, 0, 1.0), x0=(0, 0, 0), **kw):
    d = np.asarray(direction, float)
    d /= np.linalg.norm(d)
    s = np.linspace(0, length, n)
    x = np.asarray(x0, float)[None, :] + s[:, None] * d[None, :]
    return FiberFD(x, length=length, bending_rigidity=2.5e-3, eta=1.0, **kw)


def test_derivatives_of_straight_fiber():
    f = straight_fiber(n=32, direction=(0, 1, 0))
    f.update_derivatives()
    # unit tangent along y, higher derivatives zero
    assert np.allclose(f.xs[1], 1.0, atol=1e-10)
    assert np.allclose(f.xs[[0, 2]], 0.0, atol=1e-10)
    assert np.allclose(f.xss, 0.0, atol=1e-7)
    assert np.allclose(f.xssss, 0.0, atol=1e-3)  # 4th derivative noise floor


def test_batched_assembly_matches_per_fiber_loop():
    """fiber_batch.assemble_uniform must reproduce the per-fiber
    update_linear_operator/update_RHS/apply_bc_rectangular/force_operator
    results exactly, across mixed BCs."""
    from skellysim_amd.fiber_batch import assemble_uniform
    rng = np.random.default_rng(12)
    dt, eta, n = 0.05, 1.3, 24
    fibers, flows, motors = [], [], []
    for k in range(5):
        f = straight_fiber(n=n, length=0.8 + 0.1 * k,
                           direction=rng.uniform(-1, 1, 3),
                           x0=rng.uniform(-1, 1, 3),
                           minus_clamped=(k % 2 == 0))
        f.force_scale = -0.05
        f.update_constants(eta)
        f.update_derivatives()
        fibers.append(f)
        flows.append(rng.uniform(-1, 1, (3, n)))
        motors.append(rng.uniform(-1, 1, (3, n)))

    refs = []
    for f, fl, mo in zip(fibers, flows, motors):
        f.update_linear_operator(dt, eta)
        f.update_force_operator()
        f.update_RHS(dt, fl, mo)
        f.apply_bc_rectangular(dt, fl, None)
        refs.append((f.A.copy(), f.RHS.copy(), f.force_operator.copy()))

    A, RHS, F = assemble_uniform(fibers, dt, eta,
                                 flow=np.stack(flows), f_external=np.stack(motors))
    for k, (Ar, Rr, Fr) in enumerate(refs):
        assert np.allclose(A[k], Ar, atol=1e-13, rtol=1e-13), k
        assert np.allclose(RHS[k], Rr, atol=1e-13, rtol=1e-13), k
        assert np.allclose(F[k], Fr, atol=1e-13, rtol=1e-13), k

    # with external bc forces (the f_on_fiber leg of apply_bc_rectangular)
    bc_f = [rng.uniform(-1, 1, (3, n)) for _ in fibers]
    refs2 = []
    for f, fl, mo, bf in zip(fibers, flows, motors, bc_f):
        f.update_linear_operator(dt, eta)
        f.update_RHS(dt, fl, mo)
        f.apply_bc_rectangular(dt, fl, bf)
        refs2.append(f.RHS.copy())
    _, RHS2, _ = assemble_uniform(fibers, dt, eta, flow=np.stack(flows),
                                  f_external=np.stack(motors),
                                  bc_force=np.stack(bc_f))
    for k, Rr in enumerate(refs2):
        assert np.allclose(RHS2[k], Rr, atol=1e-13, rtol=1e-13), k


def test_torch_assembly_matches_numpy():
    """fiber_batch.assemble_uniform_t (the on-device per-timestep assembly)
    must equal the numpy assembly to fp64 roundoff, across mixed BCs and all
    optional force inputs."""
    from skellysim_amd.fiber_batch import assemble_uniform, assemble_uniform_t
    rng = np.random.default_rng(12)
    dt, eta, n = 0.05, 1.3, 24
    fibers = []
    for k in range(5):
        f = straight_fiber(n=n, length=0.8 + 0.1 * k,
                           direction=rng.uniform(-1, 1, 3),
                           x0=rng.uniform(-1, 1, 3),
                           minus_clamped=(k % 2 == 0))
        f.force_scale = -0.05
        f.update_constants(eta)
        f.update_derivatives()
        fibers.append(f)
    for with_forces in (False, True):
        flow = np.stack([rng.uniform(-1, 1, (3, n)) for _ in fibers]) \
            if with_forces else None
        fe = np.stack([rng.uniform(-1, 1, (3, n)) for _ in fibers]) \
            if with_forces else None
        bf = np.stack([rng.uniform(-1, 1, (3, n)) for _ in fibers]) \
            if with_forces else None
        A, R, F = assemble_uniform(fibers, dt, eta, flow=flow,
                                   f_external=fe, bc_force=bf)
        At, Rt, Ft = assemble_uniform_t(fibers, dt, eta, flow=flow,
                                        f_external=fe, bc_force=bf)
        assert np.allclose(A, At.numpy(), atol=1e-13, rtol=1e-13)
        assert np.allclose(R, Rt.numpy(), atol=1e-12, rtol=1e-13)
        assert np.allclose(F, Ft.numpy(), atol=1e-13, rtol=1e-13)


def test_periphery_repulsion_force():
    """Steric wall force (periphery.cpp:140-162): inward-pointing, decays
    with the gap, zero outside/collided, clamped node skipped."""
    n = 8
    s = np.linspace(0.0, 0.9, n)
    fib = straight_fiber(n=n, length=0.9, direction=(1.0, 0, 0))
    f = fib.periphery_repulsion("sphere", radius=1.0)
    # forces point OUTWARD toward the wall... dr = p - u_hat*radius points
    # inward (p inside): f = f_0 * dr/|dr| * exp(-gap/l_0) points toward
    # the center for interior points near the wall
    mags = np.linalg.norm(f, axis=0)
    assert mags[0] == 0.0  # node at origin: r_mag=0 -> u_hat undefined? node 0 at x=0
    assert np.all(np.diff(mags[1:]) > 0)  # monotone growth toward the wall
    # direction: -x (toward center) for points on the +x axis
    assert np.all(f[0, 1:] < 0)
    # clamped fiber skips node 0
    fib.minus_clamped = True
    f2 = fib.periphery_repulsion("sphere", radius=1.0)
    assert np.all(f2[:, 0] == 0) and np.allclose(f2[:, 1:], f[:, 1:])
    # ellipsoid variant: finite and inward on the x axis
    f3 = fib.periphery_repulsion("ellipsoid", abc=(1.0, 0.8, 0.8))
    assert np.isfinite(f3).all() and np.all(f3[0, 1:] < 0)


@pytest.mark.timeout(300)
def test_free_fiber_advects_with_uniform_flow():
    """One free fiber, uniform background U, no shell: after K backward-Euler
    steps the fiber has translated by U*K*dt and kept its shape (the
    reference pins this at rel < 1e-13 over its run;
    test_fiber_uniform_background.py:40-66)."""
    U = np.array([0.1, -0.05, 0.02])
    fib = straight_fiber(n=32, direction=(0, 0, 1.0))
    x0 = fib.x.copy()
    dt = 0.1
    sys_ = SystemFD([fib], eta=1.0, dt=dt, backend=OracleBackend(),
                    background_flow=lambda r: np.tile(U, (len(r), 1)))
    K = 5
    for _ in range(K):
        info = sys_.step(tol=1e-12, maxiter=300)
        assert info["converged"], info
    expected = x0 + U[:, None] * (K * dt)
    err = np.abs(fib.x - expected).max()
    assert err < 1e-10, err
    # tension stays ~0 for rigid advection
    assert np.abs(fib.tension).max() < 1e-8


@pytest.mark.timeout(300)
def test_adaptive_run_loop():
    """System::run mirror: adaptive dt grows to dt_max on an easy problem
    and the accepted-time bookkeeping reaches t_final exactly."""
    U = np.array([0.1, 0.0, 0.0])
    fib = straight_fiber(n=16)
    x0 = fib.x.copy()
    sys_ = SystemFD([fib], eta=1.0, dt=0.05, backend=OracleBackend(),
                    background_flow=lambda r: np.tile(U, (len(r), 1)))
    writes = []
    hist = sys_.run(t_final=0.4, dt_max=0.1, tol=1e-12, maxiter=200,
                    on_accept=lambda s, t: writes.append(t))
    assert len(hist) >= 4
    t_end = hist[-1]["time"]
    assert t_end >= 0.4
    assert hist[-1]["dt"] == pytest.approx(0.1)  # grew to dt_max
    assert writes == [h["time"] for h in hist]
    # the reference's time-accounting quirk (system.cpp:554-558, replicated
    # verbatim): the clock advances by the POST-growth dt while the state
    # moved by the dt actually taken — displacement tracks the taken dts
    taken = [0.05] + [h["dt"] for h in hist[:-1]]
    err = np.abs(fib.x - (x0 + U[:, None] * np.sum(taken))).max()
    assert err < 1e-9, err
    assert t_end >= np.sum(taken)  # clock runs ahead on growth steps


@pytest.mark.timeout(300)
def test_adaptive_run_rejects_below_dt_min():
    """Unattainable fiber-error tolerance: every step rejected, dt shrinks,
    run aborts below dt_min (system.cpp:548-551)."""
    fib = straight_fiber(n=16)
    sys_ = SystemFD([fib], eta=1.0, dt=0.05, backend=OracleBackend(),
                    background_flow=lambda r: np.tile([0.1, 0, 0], (len(r), 1)))
    with pytest.raises(RuntimeError, match="dt_min"):
        sys_.run(t_final=1.0, fiber_error_tol=0.0, dt_min=1e-3, tol=1e-12)


@pytest.mark.timeout(300)
def test_two_fiber_system_converges_and_is_finite():
    """Two interacting fibers: GMRES converges and positions remain sane."""
    f1 = straight_fiber(n=32, direction=(0, 0, 1.0), x0=(0, 0, 0))
    f2 = straight_fiber(n=32, direction=(0, 1, 0.2), x0=(0.5, 0, 0))
    sys_ = SystemFD([f1, f2], eta=1.0, dt=0.05, backend=OracleBackend(),
                    background_flow=lambda r: np.tile([0.05, 0, 0], (len(r), 1)))
    info = sys_.step(tol=1e-10, maxiter=300)
    assert info["converged"], info
    assert np.isfinite(f1.x).all() and np.isfinite(f2.x).all()
    # inextensibility: fiber length preserved to the penalty tolerance
    for f in (f1, f2):
        seg = np.diff(f.x.T, axis=0)
        length = np.linalg.norm(seg, axis=1).sum()
        assert abs(length - f.length) / f.length < 5e-3


def test_motor_activation_delay():
    """Motor forces stay off until the clock passes
    implicit_motor_activation_delay (system.cpp:417-419): a straight
    motor-forced fiber (no bending forces) is quiescent before the delay
    and translates after."""
    def make():
        fib = straight_fiber(n=16, force_scale=-0.05)
        sys_ = SystemFD([fib], eta=1.0, dt=0.05, backend=OracleBackend())
        sys_.motor_activation_delay = 0.2
        return sys_, fib

    sys_, fib = make()
    x0 = fib.x.copy()
    sys_.time = 0.0
    assert sys_.step(tol=1e-12, maxiter=200)["converged"]
    moved_before = np.abs(fib.x - x0).max()
    assert moved_before < 1e-10  # motor off, straight fiber -> no motion

    sys2, fib2 = make()
    x0b = fib2.x.copy()
    sys2.time = 0.3          # past the delay
    assert sys2.step(tol=1e-12, maxiter=200)["converged"]
    moved_after = np.abs(fib2.x - x0b).max()
    assert moved_after > 1e-4
