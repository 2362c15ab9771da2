"""Generate a small ellipsoidal BODY geometry fixture (nodes, OUTWARD
normals, RBF quadrature weights) with the reference's own tooling — the
body analog of make_ellipsoid_nodes.py (bodies keep the outward normals;
the periphery fixtures negate them, precompute.py:80-81). Build-container
only (reads /root/reference)."""

import os
import sys
import time
import warnings

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from make_periphery_fixture import install_shims  # noqa: E402


def main(n_nodes=200, a=1.5, b=1.0, c=1.0):
    install_shims()
    sys.path.insert(0, "/root/reference/src")
    from skelly_sim.shape_gallery import ShapeGallery
    import skelly_sim.Smooth_Closed_Surface_Quadrature_RBF as quadlib
    from scipy.spatial import ConvexHull

    t0 = time.time()
    boundary = ShapeGallery("ellipsoid", n_nodes, a=a, b=b, c=c)
    nodes = boundary.nodes
    normals = boundary.node_normals  # OUTWARD for a body
    hull = ConvexHull(nodes)
    with warnings.catch_warnings():
        warnings.filterwarnings("ignore")
        qw = quadlib.Smooth_Closed_Surface_Quadrature_RBF(
            nodes, hull.simplices, boundary.h, boundary.gradh)
    print(f"{len(nodes)} nodes in {time.time()-t0:.1f}s; "
          f"area sum={qw.sum():.4f}")

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = os.path.join(repo, "tests", "golden", "ellipsoid_body_nodes.npz")
    np.savez_compressed(out, nodes=nodes, normals=normals,
                        quadrature_weights=qw, a=a, b=b, c=c)
    print("wrote", out, os.path.getsize(out) / 1e6, "MB")


if __name__ == "__main__":
    main()
