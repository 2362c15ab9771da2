"""Generate the config-4 ellipsoidal-periphery GEOMETRY fixture (nodes,
inward normals, RBF quadrature weights) with the reference's own tooling.
Build-container only (reads /root/reference).

BASELINE config 4: 8192-node ellipsoidal periphery with the reference's
EllipsoidalPeriphery defaults a=7.8, b=4.16, c=4.16 (skelly_config.py:548-550)
scaled x1.04 (precompute.py:34,57-59). Only the small geometry arrays are
committed (tests/golden/ellipsoid_8k_nodes.npz, ~0.5 MB); the 24576^2 dense
operators are assembled ON the GPU by
skellysim_amd/periphery_precompute.py (validated against the reference
assembly at 192 nodes in tests/test_gpu_periphery_solve.py).
"""

import os
import sys
import time
import warnings

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from make_periphery_fixture import install_shims  # noqa: E402


def main(n_nodes=8192):
    install_shims()
    sys.path.insert(0, "/root/reference/src")
    from skelly_sim.shape_gallery import ShapeGallery
    import skelly_sim.Smooth_Closed_Surface_Quadrature_RBF as quadlib
    from scipy.spatial import ConvexHull

    scale = 1.04
    a, b, c = 7.8 * scale, 4.16 * scale, 4.16 * scale
    t0 = time.time()
    boundary = ShapeGallery("ellipsoid", n_nodes, a=a, b=b, c=c)
    nodes = boundary.nodes
    normals = -boundary.node_normals  # precompute.py:80-81
    hull = ConvexHull(nodes)
    print(f"shape+hull: {time.time()-t0:.1f}s")
    with warnings.catch_warnings():
        warnings.filterwarnings("ignore")
        qw = quadlib.Smooth_Closed_Surface_Quadrature_RBF(
            nodes, hull.simplices, boundary.h, boundary.gradh)
    print(f"quadrature: {time.time()-t0:.1f}s; area sum={qw.sum():.4f}")

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = os.path.join(repo, "tests", "golden", f"ellipsoid_{n_nodes}_nodes.npz")
    np.savez_compressed(out, nodes=nodes, normals=normals, quadrature_weights=qw,
                        a=a, b=b, c=c)
    print("wrote", out, os.path.getsize(out) / 1e6, "MB")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 8192)
