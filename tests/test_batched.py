"""Tests of the batched per-fiber algebra (CPU here; same torch code path is
rocSOLVER/rocBLAS batched on GPU — exercised on device in test_gpu_flows)."""

import numpy as np
import pytest
import torch

from skellysim_amd.batched import BatchedLU, batched_matvec


def test_batched_lu_solve_matches_direct():
    rng = np.random.default_rng(0)
    nf, m = 64, 128  # 64 fibers, 4n with n=32
    A = rng.uniform(-1, 1, (nf, m, m)) + 4 * np.eye(m)
    b = rng.uniform(-1, 1, (nf, m))
    lu = BatchedLU(torch.from_numpy(A))
    x = lu.solve(torch.from_numpy(b))
    ref = np.stack([np.linalg.solve(A[i], b[i]) for i in range(nf)])
    assert np.linalg.norm(x.numpy() - ref) / np.linalg.norm(ref) < 1e-10


def test_batched_lu_reuse_across_solves():
    rng = np.random.default_rng(1)
    nf, m = 8, 64
    A = rng.uniform(-1, 1, (nf, m, m)) + 3 * np.eye(m)
    lu = BatchedLU(torch.from_numpy(A))
    for seed in range(3):  # one factorization, many solves (per-iteration use)
        b = np.random.default_rng(seed).uniform(-1, 1, (nf, m))
        x = lu.solve(torch.from_numpy(b))
        ref = np.stack([np.linalg.solve(A[i], b[i]) for i in range(nf)])
        assert np.allclose(x.numpy(), ref, atol=1e-10)


def test_batched_matvec():
    rng = np.random.default_rng(2)
    nf, m = 16, 96
    A = rng.uniform(-1, 1, (nf, m, m))
    v = rng.uniform(-1, 1, (nf, m))
    out = batched_matvec(torch.from_numpy(A), torch.from_numpy(v))
    ref = np.einsum("bij,bj->bi", A, v)
    assert np.allclose(out.numpy(), ref, atol=1e-12)


def test_shape_and_dtype_validation():
    with pytest.raises(ValueError):
        BatchedLU(torch.zeros(3, 4, 5, dtype=torch.float64))
    with pytest.raises(TypeError):
        BatchedLU(torch.zeros(3, 4, 4, dtype=torch.float32))


def test_lu_trsm_path_matches_lu_solve(monkeypatch):
    """SKELLY_LU_TRSM=1 (round-2 experiment: batched solve_triangular
    instead of magma lu_solve) must produce the same solutions."""
    import torch
    from skellysim_amd.batched import BatchedLU
    rng = np.random.default_rng(3)
    mats = torch.from_numpy(rng.uniform(-1, 1, (5, 24, 24))
                            + 24 * np.eye(24)[None])
    rhs = torch.from_numpy(rng.uniform(-1, 1, (5, 24)))
    lu = BatchedLU(mats.clone())
    x_ref = lu.solve(rhs)
    monkeypatch.setenv("SKELLY_LU_TRSM", "1")
    x_trsm = lu.solve(rhs)
    assert torch.allclose(x_trsm, x_ref, atol=1e-12)
    direct = torch.linalg.solve(mats, rhs.unsqueeze(-1)).squeeze(-1)
    assert torch.allclose(x_trsm, direct, atol=1e-10)
