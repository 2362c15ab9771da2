"""Fiber-fiber steric repulsion (SystemFD._fiber_fiber_repulsion — engine
extension, default off; the reference has only fiber-periphery steric,
periphery.cpp:140-162)."""

import numpy as np
import pytest

from skellysim_amd.fiber_fd import FiberFD
from skellysim_amd.system_fd import SystemFD
from oracle_backend import OracleBackend


def two_fibers(sep, n=16):
    x1 = np.linspace([0, 0, 0], [0, 0, 1.0], n)
    x2 = np.linspace([sep, 0, 0], [sep, 0, 1.0], n)
    fibs = [FiberFD(x, length=1.0, bending_rigidity=2.5e-3, eta=1.0,
                    minus_clamped=True) for x in (x1, x2)]
    return SystemFD(fibs, eta=1.0, dt=0.01, backend=OracleBackend())


def test_pairwise_forces_push_apart_and_balance():
    s = two_fibers(0.05)
    f = s._fiber_fiber_repulsion(f_0=20.0, l_0=0.05)
    n = s.fibers[0].n_nodes
    # fiber 1 pushed toward -x, fiber 2 toward +x; y components zero
    # (z picks up end effects from diagonal node pairs, cancelling in sum)
    assert np.all(f[:n, 0] < 0) and np.all(f[n:, 0] > 0)
    assert np.allclose(f[:, 1], 0.0, atol=1e-14)
    # Newton's third law: zero net force
    assert np.allclose(f.sum(axis=0), 0.0, atol=1e-12)
    # magnitude follows the exponential law for the directly-opposite pair
    gap = 0.05 - 2 * 0.0125
    # each node interacts with several nodes of the other fiber; the
    # dominant (aligned) contribution has magnitude f_0*exp(-gap/l_0)
    assert f[n:, 0].max() > 20.0 * np.exp(-gap / 0.05)


def test_no_self_interaction_and_cutoff():
    s = two_fibers(5.0)  # far beyond any cutoff
    f = s._fiber_fiber_repulsion(f_0=20.0, l_0=0.05)
    assert np.allclose(f, 0.0)
    # a single bent fiber with nearby OWN nodes gets no forces
    n = 16
    th = np.linspace(0, np.pi, n)
    x = np.stack([0.2 * np.cos(th), 0.2 * np.sin(th), np.zeros(n)], axis=1)
    fib = FiberFD(x, length=float(0.2 * np.pi), bending_rigidity=2.5e-3,
                  eta=1.0)
    s1 = SystemFD([fib], eta=1.0, dt=0.01, backend=OracleBackend())
    assert np.allclose(s1._fiber_fiber_repulsion(f_0=20.0, l_0=0.05), 0.0)


@pytest.mark.timeout(300)
def test_steric_solve_separates_close_fibers():
    """Two close parallel fibers under repulsion: the coupled solve
    converges and the first step moves them apart."""
    s = two_fibers(0.06)
    s.steric_interaction = dict(f_0=5.0, l_0=0.05)
    info = s.step(tol=1e-10, maxiter=200, restart=100)
    assert info["converged"], info
    sep_after = s.fibers[1].x[0, 1:] - s.fibers[0].x[0, 1:]  # free nodes
    assert np.all(sep_after > 0.06 - 1e-12)
    assert sep_after.mean() > 0.06  # pushed apart on average
