"""Generate the config-5 OOCYTE periphery geometry (surface of revolution)
with the reference's own tooling. Build-container only.

The reference's shape_gallery surface_of_revolution path needs the
`function_generator` package (absent here): its FunctionGenerator fits a
function on [a, b] for fast evaluation/differentiation. The shim below is a
numpy Chebyshev interpolant with an accuracy check that RAISES on a poor fit
— exactly what the reference's Envelope retry loop expects (it shrinks the
bounds away from the endpoint singularities until the fit succeeds,
shape_gallery.py:25-42).

Envelope parameters: the oocyte example defaults
(examples/oocyte/gen_config.py:48-59: T=0.72, p1=0.4, p2=0.2, length=7.5,
bounds +-3.75, ~6000 node target), node scale x1.04 (precompute.py:34,70-72).
"""

import os
import sys
import time
import types
import warnings

import numpy as np


class ChebFunctionGenerator:
    """Drop-in for function_generator.FunctionGenerator: Chebyshev
    interpolation on [a, b] with __call__ / differentiate and a fit-quality
    check that raises (triggering the Envelope's bound-shrinking retry)."""

    DEG = 512
    TOL = 1e-6

    def __init__(self, f, a, b, *args, **kwargs):
        from numpy.polynomial import chebyshev as C

        a, b = float(a), float(b)
        with warnings.catch_warnings():
            warnings.filterwarnings("error")  # overflow in f -> retry
            cheb = C.Chebyshev.interpolate(f, self.DEG, domain=[a, b])
            xs = np.linspace(a, b, 4096)
            err = np.max(np.abs(cheb(xs) - f(xs)))
            scale = max(1.0, np.max(np.abs(f(xs))))
        if not np.isfinite(err) or err > self.TOL * scale:
            raise RuntimeError(f"Chebyshev fit error {err}")
        # only install attributes on success: the Envelope retry loop detects
        # failure by the ABSENCE of self.a (shape_gallery.py:38-41)
        self.a, self.b = a, b
        self._cheb = cheb
        self._dcheb = cheb.deriv()

    def __call__(self, x):
        return self._cheb(x)

    def differentiate(self, x):
        return self._dcheb(x)


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
    fg.FunctionGenerator = ChebFunctionGenerator
    sys.modules["function_generator"] = fg  # override any stub


def main(n_nodes_target=6000):
    install_shims()
    sys.path.insert(0, "/root/reference/src")
    from skelly_sim.shape_gallery import ShapeGallery
    import skelly_sim.Smooth_Closed_Surface_Quadrature_RBF as quadlib
    from scipy.spatial import ConvexHull

    envelope_config = {
        "n_nodes_target": n_nodes_target,
        "lower_bound": -3.75,
        "upper_bound": 3.75,
        "height": "0.5 * T * ((1 + 2*x/length)**p1) * ((1 - 2*x/length)**p2) * length",
        "T": 0.72,
        "p1": 0.4,
        "p2": 0.2,
        "length": 7.5,
    }
    t0 = time.time()
    b = ShapeGallery("surface_of_revolution", 0, envelope_config=envelope_config,
                     scale_factor=1.04)
    nodes = b.nodes
    normals = -b.node_normals  # precompute.py:80-81
    print(f"shape: {len(nodes)} nodes in {time.time()-t0:.1f}s")
    hull = ConvexHull(nodes)
    with warnings.catch_warnings():
        warnings.filterwarnings("ignore")
        qw = quadlib.Smooth_Closed_Surface_Quadrature_RBF(nodes, hull.simplices,
                                                          b.h, b.gradh)
    print(f"quadrature: {time.time()-t0:.1f}s; area sum = {qw.sum():.3f}")

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = os.path.join(repo, "tests", "golden", "oocyte_nodes.npz")
    np.savez_compressed(out, nodes=nodes, normals=normals, quadrature_weights=qw,
                        envelope_T=0.72, envelope_p1=0.4, envelope_p2=0.2,
                        envelope_length=7.5, scale_factor=1.04)
    print("wrote", out, os.path.getsize(out) / 1e6, "MB")


if __name__ == "__main__":
    main()
