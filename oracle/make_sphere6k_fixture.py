import sys, os
sys.path.insert(0, '/root/repo/oracle')
from make_periphery_fixture import install_shims
import numpy as np, time, warnings

install_shims()
sys.path.insert(0, "/root/reference/src")
from skelly_sim.shape_gallery import ShapeGallery
import skelly_sim.Smooth_Closed_Surface_Quadrature_RBF as quadlib
from scipy.spatial import ConvexHull

n_nodes = 6000
radius = 15.0 * 1.04  # oocyte-scale spherical periphery (config 5: ~4k fibers inside)
t0 = time.time()
b = ShapeGallery("sphere", n_nodes, radius=radius)
nodes = b.nodes; normals = -b.node_normals
hull = ConvexHull(nodes)
with warnings.catch_warnings():
    warnings.filterwarnings("ignore")
    qw = quadlib.Smooth_Closed_Surface_Quadrature_RBF(nodes, hull.simplices, b.h, b.gradh)
print(f"quad {time.time()-t0:.0f}s sum={qw.sum():.2f} (4piR^2={4*np.pi*radius**2:.2f})")
out = "/root/repo/tests/golden/sphere_6000_nodes.npz"
np.savez_compressed(out, nodes=nodes, normals=normals, quadrature_weights=qw, radius=radius)
print("wrote", out, os.path.getsize(out)/1e6, "MB")
