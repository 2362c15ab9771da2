"""hipGraph-captured GMRES iteration (gmres.py use_graph /
SKELLY_HIPGRAPH=1): the graph-replayed solve must converge to the same
solution as the eager device-resident solve on a real coupled system
(fibers + shell, split-path pair kernels under capture via the persistent
split-K workspace)."""

import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

HERE = os.path.dirname(os.path.abspath(__file__))


def _build():
    import torch
    from skellysim_amd.fiber_fd import FiberFD
    from skellysim_amd.system_fd import SystemFD, Shell, HipBackend

    fx = np.load(os.path.join(HERE, "golden", "periphery_sphere_192.npz"))
    rng = np.random.default_rng(5)
    fibers = []
    for k in range(24):
        d = rng.uniform(-1, 1, 3)
        x0 = rng.uniform(-0.3, 0.3, 3)
        d /= np.linalg.norm(d)
        s = np.linspace(0, 0.5, 32)
        x = x0[None, :] + s[:, None] * d[None, :]
        fibers.append(FiberFD(x, length=0.5, bending_rigidity=2.5e-3,
                              eta=1.0, minus_clamped=(k % 2 == 0),
                              force_scale=-0.05))
    shell = Shell(fx["nodes"] * 4.0, fx["normals"],
                  fx["stresslet_plus_complementary"], fx["M_inv"])
    U = np.array([0.05, 0.02, -0.04])
    bg = lambda r: np.tile(U, (len(r), 1))
    return SystemFD(fibers, eta=1.0, dt=0.05, shell=shell,
                    backend=HipBackend(), background_flow=bg)


@pytest.mark.timeout(600)
def test_graph_solve_matches_eager(hip_lib_path):
    s_e = _build()
    i_e = s_e.solve(tol=1e-11, maxiter=300, restart=120)
    assert i_e["converged"], i_e

    os.environ["SKELLY_HIPGRAPH"] = "1"
    try:
        s_g = _build()
        i_g = s_g.solve(tol=1e-11, maxiter=300, restart=120)
    finally:
        os.environ.pop("SKELLY_HIPGRAPH", None)
    assert i_g["converged"], i_g
    assert i_g["true_residual"] is not None and \
        i_g["true_residual"] < 1e-10, i_g["true_residual"]
    rel = np.linalg.norm(s_g.solution - s_e.solution) / \
        np.linalg.norm(s_e.solution)
    assert rel < 1e-9, rel
    # iteration counts agree (ULP-level ICGS differences only)
    assert abs(i_g["iters"] - i_e["iters"]) <= 2, (i_g["iters"], i_e["iters"])
