"""CPU tests of point/background sources (skellysim_amd/sources.py vs
src/core/point_source.cpp and background_source.cpp)."""

import numpy as np
import pytest

from skellysim_amd.sources import (PointSource, PointSourceContainer,
                                   BackgroundSource)
from skellysim_amd.fiber_fd import FiberFD
from skellysim_amd.system_fd import SystemFD
from oracle_backend import OracleBackend


def test_background_source_formula():
    """v_j = uniform_j + r[components_j] * scale_j
    (background_source.cpp:14-22), including permuted components (e.g. a
    planar extensional flow)."""
    bs = BackgroundSource(components=(1, 0, 2), scale_factor=(0.5, -0.5, 0.0),
                          uniform=(0.1, 0.0, -0.2))
    r = np.array([[1.0, 2.0, 3.0], [-1.0, 0.5, 0.0]])
    v = bs.flow(r)
    ref = np.array([[0.1 + 2.0 * 0.5, 0.0 - 1.0 * 0.5, -0.2],
                    [0.1 + 0.5 * 0.5, 0.0 + 1.0 * 0.5, -0.2]])
    assert np.allclose(v, ref, atol=1e-15)
    assert bs.is_active()
    assert not BackgroundSource().is_active()


def test_point_source_flow_and_ttl():
    """Forces flow through the regularized Stokeslet, torques through the
    rotlet; a source with time_to_live stops contributing once
    time >= ttl (point_source.cpp:16-55)."""
    import oracle
    be = OracleBackend()
    p1 = PointSource(position=(0.5, 0, 0), force=(1.0, 0, 0))
    p2 = PointSource(position=(0, 0.5, 0), torque=(0, 0, 2.0),
                     time_to_live=1.0)
    psc = PointSourceContainer([p1, p2])
    trg = np.array([[2.0, 1.0, 0.5], [-1.0, 0.3, 0.2]])
    v = psc.flow(trg, 1.3, time=0.5, backend=be)
    ref = oracle.oseen_contract(np.array([[0.5, 0, 0.]]), trg,
                                np.array([[1.0, 0, 0.]]), 1.3)
    ref += oracle.rotlet(np.array([[0, 0.5, 0.]]), trg,
                         np.array([[0, 0, 2.0]]), 1.3)
    assert np.allclose(v, ref, atol=1e-15)
    # past the torquer's ttl only the (immortal, ttl=0) forcer remains
    v2 = psc.flow(trg, 1.3, time=1.0, backend=be)
    ref2 = oracle.oseen_contract(np.array([[0.5, 0, 0.]]), trg,
                                 np.array([[1.0, 0, 0.]]), 1.3)
    assert np.allclose(v2, ref2, atol=1e-15)


@pytest.mark.timeout(300)
def test_background_source_advects_fiber():
    """The config-driven uniform background reproduces the tier-3 advection
    anchor (free fiber moves rigidly with a uniform flow)."""
    U = np.array([0.1, -0.05, 0.02])
    s = np.linspace(0, 1.0, 16)
    fib = FiberFD(s[:, None] * np.array([0.0, 0, 1.0])[None, :],
                  length=1.0, bending_rigidity=2.5e-3, eta=1.0)
    x0 = fib.x.copy()
    sys_ = SystemFD([fib], eta=1.0, dt=0.1, backend=OracleBackend())
    sys_.background_source = BackgroundSource(uniform=U)
    for _ in range(3):
        assert sys_.step(tol=1e-12, maxiter=200)["converged"]
    err = np.abs(fib.x - (x0 + U[:, None] * 0.3)).max()
    assert err < 1e-10, err


@pytest.mark.timeout(300)
def test_point_source_ttl_in_run_loop():
    """run() advances the clock (properties.time) that gates source
    lifetimes: a dead point source leaves the fiber static."""
    s = np.linspace(0, 1.0, 16)

    def build():
        fib = FiberFD(np.array([1.0, 0, 0])[None, :] +
                      s[:, None] * np.array([0.0, 0, 1.0])[None, :],
                      length=1.0, bending_rigidity=2.5e-3, eta=1.0)
        sys_ = SystemFD([fib], eta=1.0, dt=0.1, backend=OracleBackend())
        sys_.point_sources = PointSourceContainer(
            [PointSource(position=(-1.0, 0, 0), force=(0.5, 0, 0),
                         time_to_live=0.15)])
        return sys_, fib

    sys_, fib = build()
    x0 = fib.x.copy()
    sys_.run(t_final=0.1, adaptive=False, tol=1e-11)
    moved_live = np.abs(fib.x - x0).max()
    assert moved_live > 1e-5            # source alive during first step
    sys_.run(t_final=0.3, adaptive=False, tol=1e-11)  # source now dead
    x_dead_start = fib.x.copy()
    sys_.run(t_final=0.4, adaptive=False, tol=1e-11)
    moved_dead = np.abs(fib.x - x_dead_start).max()
    assert moved_dead < moved_live * 0.2
