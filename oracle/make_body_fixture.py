"""Generate the reference-default spherical BODY precompute fixture
(600 nodes, skelly_config.py:737-738) with the reference's own tooling:
sphere surface nodes + OUTWARD normals + RBF quadrature weights, plus a
Fibonacci-lattice set of nucleation sites at the attachment radius
(above the hydrodynamic surface — skelly_config.py:733-734).
Build-container only (reads /root/reference)."""

import os
import sys
import time
import warnings

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from make_periphery_fixture import install_shims  # noqa: E402


def fibonacci_sphere(n, radius):
    """shape_gallery.py:69-84-style lattice for the nucleation sites."""
    i = np.arange(n) + 0.5
    phi = np.arccos(1 - 2 * i / n)
    golden = np.pi * (1 + 5 ** 0.5)
    theta = golden * i
    return radius * np.stack([np.cos(theta) * np.sin(phi),
                              np.sin(theta) * np.sin(phi),
                              np.cos(phi)], axis=1)


def main(n_nodes=600, radius=0.5, attachment_scale=1.1, n_sites=50):
    install_shims()
    sys.path.insert(0, "/root/reference/src")
    from skelly_sim.shape_gallery import ShapeGallery
    import skelly_sim.Smooth_Closed_Surface_Quadrature_RBF as quadlib
    from scipy.spatial import ConvexHull

    t0 = time.time()
    boundary = ShapeGallery("sphere", n_nodes, radius=radius)
    nodes = boundary.nodes
    normals = boundary.node_normals  # OUTWARD for a body
    hull = ConvexHull(nodes)
    with warnings.catch_warnings():
        warnings.filterwarnings("ignore")
        qw = quadlib.Smooth_Closed_Surface_Quadrature_RBF(
            nodes, hull.simplices, boundary.h, boundary.gradh)
    print(f"{len(nodes)} nodes in {time.time()-t0:.1f}s; "
          f"area={qw.sum():.4f} (4 pi r^2 = {4*np.pi*radius**2:.4f})")
    sites = fibonacci_sphere(n_sites, attachment_scale * radius)

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = os.path.join(repo, "tests", "golden", "body_sphere_600.npz")
    np.savez_compressed(out, nodes=nodes, normals=normals,
                        quadrature_weights=qw, radius=radius,
                        nucleation_sites=sites,
                        attachment_radius=attachment_scale * radius)
    print("wrote", out, os.path.getsize(out) / 1e6, "MB")


if __name__ == "__main__":
    main()
