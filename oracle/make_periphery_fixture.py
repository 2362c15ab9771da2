"""Generate the periphery (shell) precompute fixture with the reference's OWN
Python tooling, plus a physics-calibration record. Build-container only
(reads /root/reference); the GPU box consumes the committed fixture.

Replicates src/skelly_sim/precompute.py::precompute_periphery line-for-line
for a 192-node sphere periphery of attachment radius 1.0 (node radius x1.04,
precompute.py:34): ShapeGallery sphere nodes (shape_gallery.py:69-104),
inward normals (precompute.py "Normals are in the opposite direction"),
ConvexHull triangulation, RBF quadrature weights
(Smooth_Closed_Surface_Quadrature_RBF), singularity-subtracted stresslet
matrix + complementary kernel + inverse preconditioner
(precompute.py:113-135). Imports run under a no-op numba shim and a stub
function_generator (the sphere path never calls it).

Physics calibration stored in the fixture: for a rigid fixed shell in a
uniform background flow U, the shell density q = A^-1 (-U) must cancel the
background in the ENTIRE interior (Stokes uniqueness: u = U + D[q] = 0 on
the boundary => u = 0 inside). The residual interior velocity magnitude is
quadrature-limited; the observed max over 100 interior points (r < 0.6 R)
with the CPU oracle stresslet is stored as `interior_resid_max` and the GPU
test asserts the same bound (x1.5 slack).
"""

import sys
import time
import types
import warnings
import os

import numpy as np


def install_shims():
    numba = types.ModuleType("numba")

    def njit(*a, **k):
        if a and callable(a[0]):
            return a[0]
        return lambda f: f

    numba.njit = njit
    numba.prange = range
    sys.modules.setdefault("numba", numba)

    fg = types.ModuleType("function_generator")

    class FunctionGenerator:
        def __init__(self, *a, **k):
            pass

    fg.FunctionGenerator = FunctionGenerator
    sys.modules.setdefault("function_generator", fg)


def main():
    install_shims()
    sys.path.insert(0, "/root/reference/src")
    from skelly_sim.shape_gallery import ShapeGallery
    import skelly_sim.Smooth_Closed_Surface_Quadrature_RBF as quadlib
    import skelly_sim.periphery as periphery
    import skelly_sim.quaternion as quaternion
    import skelly_sim.kernels as kernels
    from scipy.spatial import ConvexHull
    import scipy.linalg as scla

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, repo)
    import oracle

    t0 = time.time()
    n_nodes = 192
    radius = 1.0 * 1.04  # attachment radius 1.0 x periphery_node_scale_factor
    b = ShapeGallery("sphere", n_nodes, radius=radius)
    nodes = b.nodes
    normals = -b.node_normals  # precompute.py:80-81
    hull = ConvexHull(nodes)
    with warnings.catch_warnings():
        warnings.filterwarnings("ignore")
        qw = quadlib.Smooth_Closed_Surface_Quadrature_RBF(nodes, hull.simplices, b.h, b.gradh)
    print(f"quadrature: {time.time()-t0:.1f}s, sum={qw.sum():.6f} "
          f"(4 pi R^2 = {4*np.pi*radius**2:.6f})")

    shell = periphery.Periphery(np.array([0.0, 0.0, 0.0]),
                                quaternion.Quaternion([1.0, 0.0, 0.0, 0.0]),
                                nodes, normals, qw)
    shell.get_singularity_subtraction_vectors(eta=1.0)
    trg = shell.get_r_vectors()
    nrm = shell.get_normals()
    N = shell.Nblobs
    w = shell.quadrature_weights

    S = kernels.stresslet_kernel_times_normal_numba(trg, nrm, eta=1.0)
    I = np.zeros((3 * N, 3 * N))
    for i in range(N):  # precompute.py:117-121
        I[3 * i: 3 * (i + 1), 3 * i + 0] = shell.ex[3 * i: 3 * (i + 1)] / w[i]
        I[3 * i: 3 * (i + 1), 3 * i + 1] = shell.ey[3 * i: 3 * (i + 1)] / w[i]
        I[3 * i: 3 * (i + 1), 3 * i + 2] = shell.ez[3 * i: 3 * (i + 1)] / w[i]
    Iv = np.ones(3 * N)
    Iv[0::3] /= w
    Iv[1::3] /= w
    Iv[2::3] /= w
    S += -I - np.diag(Iv)  # precompute.py:126
    C = kernels.complementary_kernel(trg, nrm)
    A = S + C  # stresslet_plus_complementary
    M_inv = scla.inv(A)  # precompute.py:135
    print(f"assembled 3N={3*N} operator: {time.time()-t0:.1f}s")

    # ---- physics calibration (CPU oracle stresslet) ----
    eta = 1.0
    U = np.array([0.3, -0.2, 0.7])
    rhs = -np.tile(U, N)  # update_RHS: RHS_ = -v_on_shell (periphery.cpp:86)
    q = M_inv @ rhs
    rng = np.random.default_rng(5)
    pts = rng.uniform(-0.6, 0.6, (400, 3))
    pts = pts[np.linalg.norm(pts, axis=1) < 0.6 * radius][:100]
    nrm_rows = nrm.reshape(N, 3)
    q_rows = q.reshape(N, 3)
    f_dl = 2.0 * eta * np.einsum("ni,nj->nij", nrm_rows, q_rows).reshape(N, 9)
    u_shell = oracle.stresslet(nodes, f_dl, pts, eta)
    resid = np.abs(u_shell + U[None, :]).max()
    gmres_resid = np.linalg.norm(A @ q - rhs) / np.linalg.norm(rhs)
    print(f"interior |U + D[q]| max = {resid:.3e}; direct-solve resid {gmres_resid:.1e}")

    out = os.path.join(repo, "tests", "golden", "periphery_sphere_192.npz")
    np.savez_compressed(out,
                        nodes=nodes, normals=normals, quadrature_weights=qw,
                        stresslet_plus_complementary=A, M_inv=M_inv,
                        radius=radius, eta=eta, U=U, interior_pts=pts,
                        interior_resid_max=resid, density=q)
    print("wrote", out, os.path.getsize(out) / 1e6, "MB")


if __name__ == "__main__":
    main()
