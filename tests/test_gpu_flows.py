"""GPU tests of the container-level flow operators (SURVEY §8a a5/a6 and
§8f next-row 1) against numpy restatements of the reference host prep +
the CPU oracle pair kernels."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def rel(a, b):
    return np.linalg.norm(a - b) / max(np.linalg.norm(b), 1e-300)


@pytest.fixture(scope="module")
def flows(hip_lib_path):
    from skellysim_amd import flows
    return flows


def test_periphery_flow_vs_oracle(flows, oracle_mod):
    """Periphery::flow (periphery.cpp:55-79): f_dl = 2 eta n_i d_j then
    stresslet."""
    rng = np.random.default_rng(21)
    n, t, eta = 3000, 1200, 1.7
    pos = rng.uniform(-1, 1, (n, 3))
    normal = rng.uniform(-1, 1, (n, 3))
    dens = rng.uniform(-1, 1, (n, 3))
    trg = rng.uniform(-1, 1, (t, 3))
    dev = torch.device("cuda:0")
    u = flows.periphery_flow(torch.from_numpy(pos).to(dev), torch.from_numpy(normal).to(dev),
                             torch.from_numpy(dens).to(dev), torch.from_numpy(trg).to(dev), eta)
    torch.cuda.synchronize()
    f_dl = 2.0 * eta * np.einsum("ni,nj->nij", normal, dens).reshape(n, 9)
    ref = oracle_mod.stresslet(pos, f_dl, trg, eta)
    assert rel(u.cpu().numpy(), ref) < 1e-10


def test_periphery_flow_empty(flows):
    dev = torch.device("cuda:0")
    trg = torch.rand(5, 3, dtype=torch.float64, device=dev)
    z = torch.zeros(0, 3, dtype=torch.float64, device=dev)
    u = flows.periphery_flow(z, z, z, trg, 1.0)
    assert u.shape == (5, 3) and torch.all(u == 0)


def test_fiber_flow_with_self_subtraction(flows, oracle_mod):
    """FiberContainer flow (f_c_fd.cpp:172-214): weighted forces, stokeslet,
    per-fiber self-term subtraction (uniform batched path + ragged path)."""
    rng = np.random.default_rng(8)
    n_fib, n_nodes, eta = 24, 32, 1.1
    N = n_fib * n_nodes
    r_fib = rng.uniform(-1, 1, (N, 3))
    forces = rng.uniform(-1, 1, (N, 3))
    weights = rng.uniform(0.1, 1.0, N)
    extra_trg = rng.uniform(-1, 1, (500, 3))
    r_trg = np.vstack([r_fib, extra_trg])  # targets start with fiber nodes
    stokeslets = rng.uniform(-1, 1, (n_fib, 3 * n_nodes, 3 * n_nodes))

    dev = torch.device("cuda:0")
    u = flows.fiber_flow(torch.from_numpy(r_fib).to(dev), torch.from_numpy(forces).to(dev),
                         torch.from_numpy(weights).to(dev), torch.from_numpy(r_trg).to(dev),
                         eta, fiber_sizes=[n_nodes] * n_fib,
                         self_stokeslets=torch.from_numpy(stokeslets).to(dev))
    torch.cuda.synchronize()

    wf = forces * weights[:, None]
    ref = oracle_mod.stokeslet(r_fib, wf, r_trg, eta)
    for i in range(n_fib):
        sl = slice(i * n_nodes, (i + 1) * n_nodes)
        ref[sl] -= (stokeslets[i] @ wf[sl].reshape(-1)).reshape(n_nodes, 3)
    assert rel(u.cpu().numpy(), ref) < 1e-10

    # ragged path (list of per-fiber matrices, unequal sizes)
    sizes = [32, 16, 48]
    N2 = sum(sizes)
    r2 = torch.from_numpy(r_fib[:N2]).to(dev)
    f2 = torch.from_numpy(forces[:N2]).to(dev)
    w2 = torch.from_numpy(weights[:N2]).to(dev)
    mats = [torch.from_numpy(rng.uniform(-1, 1, (3 * s, 3 * s))).to(dev) for s in sizes]
    u2 = flows.fiber_flow(r2, f2, w2, r2, eta, fiber_sizes=sizes, self_stokeslets=mats)
    torch.cuda.synchronize()
    wf2 = (forces[:N2] * weights[:N2, None])
    ref2 = oracle_mod.stokeslet(r_fib[:N2], wf2, r_fib[:N2], eta)
    off = 0
    for M, s in zip(mats, sizes):
        ref2[off:off + s] -= (M.cpu().numpy() @ wf2[off:off + s].reshape(-1)).reshape(s, 3)
        off += s
    assert rel(u2.cpu().numpy(), ref2) < 1e-10


def test_body_flow_and_velocity_at_targets(flows, oracle_mod):
    """BodyContainer::flow composite (stresslet nodes + stokeslet centers +
    rotlet centers) and the velocity_at_targets sum, vs oracle composition."""
    rng = np.random.default_rng(33)
    eta = 1.25
    nb_nodes, nb, t = 600, 4, 800
    body_nodes = rng.uniform(-1, 1, (nb_nodes, 3))
    body_norms = rng.uniform(-1, 1, (nb_nodes, 3))
    dens = rng.uniform(-1, 1, (nb_nodes, 3))
    centers = rng.uniform(-1, 1, (nb, 3))
    forces = rng.uniform(-1, 1, (nb, 3))
    torques = rng.uniform(-1, 1, (nb, 3))
    trg = rng.uniform(-1, 1, (t, 3))
    dev = torch.device("cuda:0")
    T = lambda a: torch.from_numpy(a).to(dev)

    u = flows.body_flow(T(body_nodes), T(body_norms), T(dens), T(centers), T(forces),
                        T(torques), T(trg), eta)
    torch.cuda.synchronize()
    f_dl = 2.0 * eta * np.einsum("ni,nj->nij", body_norms, dens).reshape(-1, 9)
    ref = (oracle_mod.stresslet(body_nodes, f_dl, trg, eta)
           + oracle_mod.stokeslet(centers, forces, trg, eta)
           + oracle_mod.rotlet(centers, trg, torques, eta))
    assert rel(u.cpu().numpy(), ref) < 1e-10

    # composite velocity_at_targets: fibers + shell + bodies
    nf_nodes = 512
    fib_pos = rng.uniform(-1, 1, (nf_nodes, 3))
    fib_f = rng.uniform(-1, 1, (nf_nodes, 3))
    fib_w = rng.uniform(0.1, 1, nf_nodes)
    sh_pos = rng.uniform(-1, 1, (700, 3))
    sh_n = rng.uniform(-1, 1, (700, 3))
    sh_d = rng.uniform(-1, 1, (700, 3))
    u2 = flows.velocity_at_targets(
        T(trg), eta,
        fiber=dict(r_src=T(fib_pos), forces=T(fib_f), weights=T(fib_w)),
        shell=dict(node_pos=T(sh_pos), node_normal=T(sh_n), density=T(sh_d)),
        bodies=dict(node_pos=T(body_nodes), node_normals=T(body_norms),
                    densities=T(dens), centers=T(centers), forces=T(forces),
                    torques=T(torques)))
    torch.cuda.synchronize()
    f_dl_sh = 2.0 * eta * np.einsum("ni,nj->nij", sh_n, sh_d).reshape(-1, 9)
    ref2 = (ref
            + oracle_mod.stokeslet(fib_pos, fib_f * fib_w[:, None], trg, eta)
            + oracle_mod.stresslet(sh_pos, f_dl_sh, trg, eta))
    assert rel(u2.cpu().numpy(), ref2) < 1e-10


def test_batched_lu_on_device(flows):
    """next-row 2: batched per-fiber LU (rocSOLVER) resident on device."""
    from skellysim_amd.batched import BatchedLU
    rng = np.random.default_rng(17)
    nf, m = 512, 128  # 512 fibers x (4*32)
    A = rng.uniform(-1, 1, (nf, m, m)) + 4 * np.eye(m)
    b = rng.uniform(-1, 1, (nf, m))
    dev = torch.device("cuda:0")
    lu = BatchedLU(torch.from_numpy(A).to(dev))
    x = lu.solve(torch.from_numpy(b).to(dev))
    torch.cuda.synchronize()
    resid = np.einsum("bij,bj->bi", A, x.cpu().numpy()) - b
    assert np.abs(resid).max() < 1e-9


def test_gmres_on_device_with_shell_operator(flows):
    """GMRES (next-row 3) running entirely on cuda:0 with a resident dense
    operator and its inverse as right preconditioner."""
    from skellysim_amd.gmres import gmres
    rng = np.random.default_rng(13)
    n = 600
    A = rng.uniform(-1, 1, (n, n)) + 20 * np.eye(n)
    b = rng.uniform(-1, 1, n)
    dev = torch.device("cuda:0")
    At = torch.from_numpy(A).to(dev)
    bt = torch.from_numpy(b).to(dev)
    x, info = gmres(lambda v: At @ v, bt, tol=1e-11, maxiter=300, restart=60)
    assert info["converged"]
    assert x.is_cuda
    r = np.linalg.norm(A @ x.cpu().numpy() - b) / np.linalg.norm(b)
    assert r < 1e-10


def test_shell_operator_gemvs(flows):
    """Periphery dense GEMVs (periphery.cpp:21-47) with resident matrices."""
    rng = np.random.default_rng(4)
    N = 1024  # 3N x 3N operators
    M_inv = rng.uniform(-1, 1, (3 * N, 3 * N))
    SPC = rng.uniform(-1, 1, (3 * N, 3 * N))
    x = rng.uniform(-1, 1, 3 * N)
    v = rng.uniform(-1, 1, 3 * N)
    dev = torch.device("cuda:0")
    op = flows.ShellOperator(torch.from_numpy(M_inv).to(dev), torch.from_numpy(SPC).to(dev))
    p = op.apply_preconditioner(torch.from_numpy(x).to(dev))
    m = op.matvec(torch.from_numpy(x).to(dev), torch.from_numpy(v).to(dev))
    torch.cuda.synchronize()
    assert rel(p.cpu().numpy(), M_inv @ x) < 1e-13
    assert rel(m.cpu().numpy(), SPC @ x + v) < 1e-13
