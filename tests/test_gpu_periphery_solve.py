"""End-to-end shell solve on GPU (config-4 machinery): reference-generated
periphery precompute (tests/golden/periphery_sphere_192.npz, built by
oracle/make_periphery_fixture.py with the reference's own Python tooling)
-> resident ShellOperator + our GMRES (right-preconditioned, ICGS) -> HIP
stresslet evaluation of the resulting density at interior points.

Physics: rigid fixed spherical shell in uniform background flow U; the
solved density must cancel U throughout the interior (Stokes uniqueness).
The bound is the quadrature-limited residual recorded by the fixture
generator on CPU (x1.5 slack)."""

import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def fix(golden_dir, hip_lib_path):
    return np.load(os.path.join(golden_dir, "periphery_sphere_192.npz"))


def test_shell_solve_and_interior_cancellation(fix):
    from skellysim_amd.flows import ShellOperator, periphery_flow
    from skellysim_amd.gmres import gmres

    dev = torch.device("cuda:0")
    A = torch.from_numpy(fix["stresslet_plus_complementary"]).to(dev)
    M_inv = torch.from_numpy(fix["M_inv"]).to(dev)
    N = fix["nodes"].shape[0]
    eta = float(fix["eta"])
    U = fix["U"]

    op = ShellOperator(M_inv, A)
    rhs = torch.from_numpy(-np.tile(U, N)).to(dev)  # RHS_ = -v_on_shell
    v0 = torch.zeros_like(rhs)

    q, info = gmres(lambda x: op.matvec(x, v0), rhs,
                    precond=op.apply_preconditioner, tol=1e-10, maxiter=50,
                    restart=30)
    assert info["converged"], info
    # with the exact inverse as right preconditioner: a couple of iterations
    assert info["iters"] <= 5, info

    # solution matches the fixture's direct solve
    q_ref = fix["density"]
    rel = np.linalg.norm(q.cpu().numpy() - q_ref) / np.linalg.norm(q_ref)
    assert rel < 1e-8

    # interior physics: U + D[q] ~ 0 at quadrature accuracy
    pts = torch.from_numpy(fix["interior_pts"]).to(dev)
    dens = q.reshape(N, 3)
    u = periphery_flow(torch.from_numpy(fix["nodes"]).to(dev),
                       torch.from_numpy(fix["normals"]).to(dev),
                       dens, pts, eta)
    torch.cuda.synchronize()
    resid = np.abs(u.cpu().numpy() + U[None, :]).max()
    bound = 1.5 * float(fix["interior_resid_max"])
    assert resid < bound, (resid, bound)


def test_on_gpu_assembly_matches_reference_precompute(fix):
    """The on-GPU shell-operator assembly (periphery_precompute.py, built on
    the engine's own dense/contraction kernels) must reproduce the
    reference-Python-generated operator (precompute.py:104-135) on the same
    nodes/normals/weights."""
    from skellysim_amd.periphery_precompute import assemble_shell_operator

    dev = torch.device("cuda:0")
    nodes = torch.from_numpy(fix["nodes"]).to(dev)
    normals = torch.from_numpy(fix["normals"]).to(dev)
    w = torch.from_numpy(fix["quadrature_weights"]).to(dev)
    A, M_inv = assemble_shell_operator(nodes, normals, w)
    torch.cuda.synchronize()
    A_ref = fix["stresslet_plus_complementary"]
    rel = np.linalg.norm(A.cpu().numpy() - A_ref) / np.linalg.norm(A_ref)
    assert rel < 1e-12, rel
    # our rocSOLVER inverse is an inverse of the same operator
    I = A @ M_inv
    err = float(torch.norm(I - torch.eye(len(I), dtype=torch.float64, device=dev)))
    assert err < 1e-9, err
    rel_inv = np.linalg.norm(M_inv.cpu().numpy() - fix["M_inv"]) / np.linalg.norm(fix["M_inv"])
    assert rel_inv < 1e-8, rel_inv


def test_config4_ellipsoid_8k_shell_end_to_end(golden_dir, hip_lib_path):
    """BASELINE config-4 periphery at full scale: 8192-node ellipsoid
    (reference EllipsoidalPeriphery defaults x1.04), operator assembled AND
    inverted on device, GMRES solve, interior cancellation to ~1e-6
    (measured 9.6e-7 on MI355X; x5 slack)."""
    from skellysim_amd.periphery_precompute import assemble_shell_operator
    from skellysim_amd.flows import ShellOperator, periphery_flow
    from skellysim_amd.gmres import gmres

    fix = np.load(os.path.join(golden_dir, "ellipsoid_8192_nodes.npz"))
    dev = torch.device("cuda:0")
    nodes = torch.from_numpy(fix["nodes"]).to(dev)
    normals = torch.from_numpy(fix["normals"]).to(dev)
    w = torch.from_numpy(fix["quadrature_weights"]).to(dev)
    N = len(nodes)
    A, M_inv = assemble_shell_operator(nodes, normals, w)
    op = ShellOperator(M_inv, A)
    U = np.array([0.3, -0.2, 0.7])
    rhs = torch.from_numpy(-np.tile(U, N)).to(dev)
    v0 = torch.zeros_like(rhs)
    q, info = gmres(lambda x: op.matvec(x, v0), rhs, precond=op.apply_preconditioner,
                    tol=1e-10, maxiter=30, restart=30)
    assert info["converged"] and info["iters"] <= 3

    rng = np.random.default_rng(5)
    a, b, c = float(fix["a"]), float(fix["b"]), float(fix["c"])
    pts = rng.uniform(-1, 1, (2000, 3)) * np.array([a, b, c])
    lvl = (pts[:, 0] / a) ** 2 + (pts[:, 1] / b) ** 2 + (pts[:, 2] / c) ** 2
    pts = pts[lvl < 0.5][:200]
    u = periphery_flow(nodes, normals, q.reshape(N, 3),
                       torch.from_numpy(pts).to(dev), 1.0)
    torch.cuda.synchronize()
    resid = np.abs(u.cpu().numpy() + U[None, :]).max()
    assert resid < 5e-6, resid


def test_shell_operator_consistency_with_fixture(fix):
    """A @ q == rhs and M_inv is A's inverse (reference precompute contract)."""
    from skellysim_amd.flows import ShellOperator

    dev = torch.device("cuda:0")
    A = torch.from_numpy(fix["stresslet_plus_complementary"]).to(dev)
    M_inv = torch.from_numpy(fix["M_inv"]).to(dev)
    N = fix["nodes"].shape[0]
    op = ShellOperator(M_inv, A)
    q = torch.from_numpy(fix["density"]).to(dev)
    rhs = -np.tile(fix["U"], N)
    r = op.matvec(q, torch.zeros_like(q)).cpu().numpy() - rhs
    assert np.linalg.norm(r) / np.linalg.norm(rhs) < 1e-12
    x = torch.from_numpy(np.random.default_rng(0).uniform(-1, 1, 3 * N)).to(dev)
    y = op.apply_preconditioner(op.matvec(x, torch.zeros_like(x)))
    assert float(torch.norm(y - x) / torch.norm(x)) < 1e-9
