"""CPU tests of the parity oracle itself: C restatement vs independent numpy
restatement, and both against the committed golden fixtures (which were pinned
against the reference's own Python kernel statements by oracle/make_golden.py
in the build container)."""

import os

import numpy as np
import pytest


def rel(a, b):
    return np.linalg.norm(a - b) / max(np.linalg.norm(b), 1e-300)


@pytest.fixture(scope="module")
def clouds():
    rng = np.random.default_rng(42)
    S, T = 211, 97
    return dict(
        r_src=rng.uniform(-1, 1, (S, 3)),
        r_trg=rng.uniform(-1, 1, (T, 3)),
        f3=rng.uniform(-1, 1, (S, 3)),
        f9=rng.uniform(-1, 1, (S, 9)),
    )


def test_c_vs_numpy_stokeslet(oracle_mod, clouds):
    c = oracle_mod.stokeslet(clouds["r_src"], clouds["f3"], clouds["r_trg"], 1.7)
    n = oracle_mod.np_stokeslet(clouds["r_src"], clouds["f3"], clouds["r_trg"], 1.7)
    assert rel(c, n) < 1e-13


def test_c_vs_numpy_stresslet(oracle_mod, clouds):
    c = oracle_mod.stresslet(clouds["r_src"], clouds["f9"], clouds["r_trg"], 1.7)
    n = oracle_mod.np_stresslet(clouds["r_src"], clouds["f9"], clouds["r_trg"], 1.7)
    assert rel(c, n) < 1e-13


def test_c_vs_numpy_oseen(oracle_mod, clouds):
    c = oracle_mod.oseen_contract(clouds["r_src"], clouds["r_trg"], clouds["f3"], 0.9)
    n = oracle_mod.np_oseen_contract(clouds["r_src"], clouds["r_trg"], clouds["f3"], 0.9)
    assert rel(c, n) < 1e-13


def test_c_vs_numpy_rotlet(oracle_mod, clouds):
    c = oracle_mod.rotlet(clouds["r_src"], clouds["r_trg"], clouds["f3"], 0.9)
    n = oracle_mod.np_rotlet(clouds["r_src"], clouds["r_trg"], clouds["f3"], 0.9)
    assert rel(c, n) < 1e-13


def test_goldens_kernel_test_recipe(oracle_mod, golden_dir):
    """The reference kernel-parity recipe (kernel_test.cpp:25-27): regenerating
    from committed inputs must reproduce the committed outputs bit-for-bit
    modulo thread-count (chunking is thread-count independent)."""
    g = np.load(os.path.join(golden_dir, "kernel_test_1229x743.npz"))
    eta = float(g["eta"])
    u = oracle_mod.stokeslet(g["r_src"], g["f3"], g["r_trg"], eta)
    assert np.array_equal(u, g["u_stokeslet"])
    u = oracle_mod.stresslet(g["r_src"], g["f9"], g["r_trg"], eta)
    assert np.array_equal(u, g["u_stresslet"])
    u = oracle_mod.oseen_contract(g["r_src"], g["r_trg"], g["f3"], eta)
    assert np.array_equal(u, g["u_oseen"])
    u = oracle_mod.rotlet(g["r_src"][:37], g["r_trg"], g["f3"][:37], eta)
    assert np.array_equal(u, g["u_rotlet"])


def test_goldens_refpy_small(oracle_mod, golden_dir):
    g = np.load(os.path.join(golden_dir, "refpy_small.npz"))
    eta = float(g["eta"])
    assert np.array_equal(oracle_mod.stokeslet(g["r_src"], g["f3"], g["r_trg"], eta),
                          g["u_stokeslet"])
    assert np.array_equal(oracle_mod.stresslet(g["r_src"], g["f9"], g["r_trg"], eta),
                          g["u_stresslet"])
    assert np.array_equal(
        oracle_mod.oseen_contract(g["r_src"][:40], g["near_trg"], g["f3"][:40], eta),
        g["u_oseen_near"])
    assert np.array_equal(oracle_mod.rotlet(g["r_src"], g["r_trg"], g["rho"], eta),
                          g["u_rotlet"])


def test_goldens_edge_selfdup(oracle_mod, golden_dir):
    """Coincident points (src==trg and duplicated sources) stay finite and
    reproduce the committed outputs."""
    g = np.load(os.path.join(golden_dir, "edge_selfdup.npz"))
    eta = float(g["eta"])
    for name, fn, f in [("u_stokeslet", oracle_mod.stokeslet, g["f3"]),
                        ("u_stresslet", oracle_mod.stresslet, g["f9"])]:
        u = fn(g["r"], f, g["r"], eta)
        assert np.all(np.isfinite(u))
        assert np.array_equal(u, g[name])
    u = oracle_mod.oseen_contract(g["r"], g["r"], g["f3"], eta)
    assert np.all(np.isfinite(u)) and np.array_equal(u, g["u_oseen"])
    u = oracle_mod.rotlet(g["r"], g["r"], g["f3"], eta)
    assert np.all(np.isfinite(u)) and np.array_equal(u, g["u_rotlet"])


def test_c_vs_numpy_stresslet_normal_density(oracle_mod, clouds):
    r = clouds["r_src"]
    n, rho = clouds["f3"], clouds["f3"][::-1]
    c = oracle_mod.stresslet_times_normal_times_density(r, n, rho)
    np_ = oracle_mod.np_stresslet_times_normal_times_density(r, n, rho)
    assert rel(c, np_) < 1e-13


def test_c_vs_numpy_oseen_tensor(oracle_mod, clouds):
    """Dense self-Oseen builder (kernels.cpp:146-195): C vs numpy, symmetry,
    zero diagonal blocks, and consistency with the contraction kernel
    (G @ rho == oseen_contract for the same cloud)."""
    r = clouds["r_src"][:64]
    G = oracle_mod.oseen_tensor(r, 1.3)
    Gn = oracle_mod.np_oseen_tensor(r, 1.3)
    assert rel(G, Gn) < 1e-13
    assert np.linalg.norm(G - G.T) < 1e-12
    for i in range(len(r)):
        assert np.all(G[3 * i: 3 * i + 3, 3 * i: 3 * i + 3] == 0.0)
    rho = clouds["f3"][:64]
    u_mat = (G @ rho.reshape(-1)).reshape(-1, 3)
    u_contract = oracle_mod.oseen_contract(r, r, rho, 1.3)
    assert rel(u_mat, u_contract) < 1e-13


def test_empty_sources(oracle_mod):
    r_trg = np.random.default_rng(0).uniform(-1, 1, (10, 3))
    u = oracle_mod.stokeslet(np.empty((0, 3)), np.empty((0, 3)), r_trg, 1.0)
    assert u.shape == (10, 3) and np.all(u == 0)


def test_empty_targets(oracle_mod):
    r_src = np.random.default_rng(0).uniform(-1, 1, (10, 3))
    u = oracle_mod.stokeslet(r_src, r_src, np.empty((0, 3)), 1.0)
    assert u.shape == (0, 3)


def test_eta_scaling(oracle_mod, clouds):
    """u scales as 1/eta (the reference divides the evaluator result by eta)."""
    u1 = oracle_mod.stokeslet(clouds["r_src"], clouds["f3"], clouds["r_trg"], 1.0)
    u2 = oracle_mod.stokeslet(clouds["r_src"], clouds["f3"], clouds["r_trg"], 2.0)
    assert rel(u1, 2.0 * u2) < 1e-15


def test_linearity_in_strengths(oracle_mod, clouds):
    """Size-independent property: the evaluation is linear in f."""
    r_src, r_trg = clouds["r_src"], clouds["r_trg"]
    f, g = clouds["f3"], clouds["f3"][::-1]
    u_sum = oracle_mod.stokeslet(r_src, f + g, r_trg, 1.0)
    u_parts = (oracle_mod.stokeslet(r_src, f, r_trg, 1.0)
               + oracle_mod.stokeslet(r_src, g, r_trg, 1.0))
    assert rel(u_sum, u_parts) < 1e-13
