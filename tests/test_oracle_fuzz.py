"""Property-based fuzzing of the CPU oracle (oracle/): the C restatement
and the independent numpy restatement of the same reference formulas must
agree bit-for-bit-close across random shapes, scales, viscosities and
coincident/near-coincident points — strengthening the §8c pinning beyond
the fixed golden clouds."""

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

import oracle


def clouds(draw, max_n=40):
    n_src = draw(st.integers(1, max_n))
    n_trg = draw(st.integers(1, max_n))
    seed = draw(st.integers(0, 2 ** 31 - 1))
    scale = draw(st.sampled_from([1e-3, 1.0, 1e3]))
    rng = np.random.default_rng(seed)
    r_src = scale * rng.uniform(-1, 1, (n_src, 3))
    r_trg = scale * rng.uniform(-1, 1, (n_trg, 3))
    if draw(st.booleans()) and n_trg <= n_src:
        r_trg = r_src[:n_trg].copy()      # exact coincidences (r == 0 paths)
    eta = draw(st.sampled_from([0.3, 1.0, 7.5]))
    return rng, r_src, r_trg, eta


@st.composite
def _case(draw):
    return clouds(draw)


@settings(max_examples=60, deadline=None)
@given(_case())
def test_stokeslet_c_vs_numpy(case):
    rng, r_src, r_trg, eta = case
    f = rng.uniform(-1, 1, (len(r_src), 3))
    c = oracle.stokeslet(r_src, f, r_trg, eta)
    n = oracle.np_stokeslet(r_src, f, r_trg, eta)
    assert np.all(np.isfinite(c))
    assert np.allclose(c, n, rtol=1e-13, atol=1e-13 * np.abs(n).max() + 1e-300)


@settings(max_examples=60, deadline=None)
@given(_case())
def test_stresslet_c_vs_numpy(case):
    rng, r_src, r_trg, eta = case
    f9 = rng.uniform(-1, 1, (len(r_src), 9))
    c = oracle.stresslet(r_src, f9, r_trg, eta)
    n = oracle.np_stresslet(r_src, f9, r_trg, eta)
    assert np.all(np.isfinite(c))
    assert np.allclose(c, n, rtol=1e-13, atol=1e-13 * np.abs(n).max() + 1e-300)


@settings(max_examples=60, deadline=None)
@given(_case())
def test_oseen_contract_c_vs_numpy(case):
    rng, r_src, r_trg, eta = case
    rho = rng.uniform(-1, 1, (len(r_src), 3))
    c = oracle.oseen_contract(r_src, r_trg, rho, eta)
    n = oracle.np_oseen_contract(r_src, r_trg, rho, eta)
    assert np.all(np.isfinite(c))
    assert np.allclose(c, n, rtol=1e-13, atol=1e-13 * np.abs(n).max() + 1e-300)


@settings(max_examples=60, deadline=None)
@given(_case())
def test_rotlet_c_vs_numpy(case):
    rng, r_src, r_trg, eta = case
    rho = rng.uniform(-1, 1, (len(r_src), 3))
    c = oracle.rotlet(r_src, r_trg, rho, eta)
    n = oracle.np_rotlet(r_src, r_trg, rho, eta)
    assert np.all(np.isfinite(c))
    assert np.allclose(c, n, rtol=1e-13, atol=1e-13 * np.abs(n).max() + 1e-300)


@settings(max_examples=30, deadline=None)
@given(st.integers(0, 2 ** 31 - 1), st.integers(2, 24),
       st.sampled_from([0.3, 1.0, 7.5]))
def test_linearity_and_eta_scaling(seed, n, eta):
    """Exact-by-construction properties of every oracle kernel: linearity
    in the source strengths and 1/eta scaling (kernels.cpp:66,82,358,365)."""
    rng = np.random.default_rng(seed)
    r = rng.uniform(-1, 1, (n, 3))
    for fn, dim in ((lambda d, e: oracle.stokeslet(r, d, r, e), 3),
                    (lambda d, e: oracle.stresslet(r, d, r, e), 9),
                    (lambda d, e: oracle.oseen_contract(r, r, d, e), 3),
                    (lambda d, e: oracle.rotlet(r, r, d, e), 3)):
        a = rng.uniform(-1, 1, (n, dim))
        b = rng.uniform(-1, 1, (n, dim))
        s = fn(a + b, eta)
        ref = fn(a, eta) + fn(b, eta)
        assert np.allclose(s, ref, rtol=1e-12,
                           atol=1e-12 * np.abs(ref).max() + 1e-300)
        ref_eta = fn(a, 1.0) / eta
        assert np.allclose(fn(a, eta), ref_eta, rtol=1e-13,
                           atol=1e-13 * np.abs(ref_eta).max() + 1e-300)
