"""Fixture for the reference's body-fiber-periphery COMPRESSION regression
(tests/combined/regression_tests/test_body_fdfiber_compression.py):

- the exact geometry the reference's own precompute pipeline generates for
  that test's config (gen_config): a 2000-node spherical periphery of
  config radius 4.0/1.04 (node radius x1.04 = 4.0, precompute.py:34) and a
  400-node spherical body of config radius 0.5 (quadrature surface at
  0.5 - body_quadrature_radius_offset_low = 0.4, precompute.py:27,156-157),
  both via ShapeGallery + RBF quadrature run from /root/reference;
- the reference test's OWN pinned answers, copied verbatim as golden
  vectors: final fiber/body positions (fdfiber_compression_finalpositions
  .npz) and the frame-98 velocity field (fdfiber_compression_data.npy).

Build-container only (reads /root/reference); tests/test_gpu_compression.py
consumes the committed fixture on the GPU box.
"""

import os
import sys
import time
import warnings

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from make_periphery_fixture import install_shims  # noqa: E402


def sphere_geometry(n_nodes, radius, inward):
    from skelly_sim.shape_gallery import ShapeGallery
    import skelly_sim.Smooth_Closed_Surface_Quadrature_RBF as quadlib
    from scipy.spatial import ConvexHull

    t0 = time.time()
    b = ShapeGallery("sphere", n_nodes, radius=radius)
    nodes = b.nodes
    normals = -b.node_normals if inward else b.node_normals
    hull = ConvexHull(nodes)
    with warnings.catch_warnings():
        warnings.filterwarnings("ignore")
        qw = quadlib.Smooth_Closed_Surface_Quadrature_RBF(
            nodes, hull.simplices, b.h, b.gradh)
    print(f"{n_nodes} nodes r={radius}: {time.time()-t0:.1f}s, "
          f"area={qw.sum():.4f} (4 pi r^2 = {4*np.pi*radius**2:.4f})")
    return nodes, normals, qw


def main():
    install_shims()
    sys.path.insert(0, "/root/reference/src")

    shell_cfg_radius = 4.0 / 1.04          # gen_config periphery.radius
    shell_nodes, shell_normals, shell_w = sphere_geometry(
        2000, shell_cfg_radius * 1.04, inward=True)
    body_nodes, body_normals, body_w = sphere_geometry(
        400, 0.5 - 0.1, inward=False)      # quadrature offset, precompute.py:156

    ref = "/root/reference/tests/combined/regression_tests"
    pins = np.load(os.path.join(ref, "fdfiber_compression_finalpositions.npz"))
    vfield = np.load(os.path.join(ref, "fdfiber_compression_data.npy"))

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = os.path.join(repo, "tests", "golden", "compression_regression.npz")
    np.savez_compressed(
        out,
        shell_nodes=shell_nodes, shell_normals=shell_normals,
        shell_weights=shell_w, shell_radius=shell_cfg_radius,
        body_nodes=body_nodes, body_normals=body_normals,
        body_weights=body_w, body_radius=0.5,
        xlast=pins["xlast"], ylast=pins["ylast"], zlast=pins["zlast"],
        bodylast=pins["bodylast"],
        velocity_field_pinned=vfield)
    print("wrote", out, os.path.getsize(out) / 1e6, "MB")


if __name__ == "__main__":
    main()
