#!/usr/bin/env python3
"""Generate surface-geometry npz files (nodes/normals/weights) with the
engine's OWN precompute (skellysim_amd/precompute.py) — no reference
tooling needed. Output is consumable by run_sim --shell-geometry /
--body-geometry and skellysim_amd.config.build_system.

    python tools/make_geometry.py sphere --n-nodes 6000 --radius 4.0 \
        --inward --out shell.npz
    python tools/make_geometry.py ellipsoid --n-nodes 8192 \
        --abc 8.112 4.3264 4.3264 --inward --out ellipsoid_shell.npz
    python tools/make_geometry.py sphere --n-nodes 600 --radius 0.5 \
        --nucleation-sites 50 --attachment-scale 1.1 --out body.npz

(Peripheries use inward normals, bodies outward — the reference's
precompute.py:80-81 convention.)"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from skellysim_amd.precompute import (sphere_geometry, ellipsoid_geometry,
                                      fibonacci_sphere)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("shape", choices=["sphere", "ellipsoid",
                                      "surface_of_revolution"])
    ap.add_argument("--n-nodes", type=int, required=True)
    ap.add_argument("--radius", type=float, default=1.0)
    ap.add_argument("--abc", type=float, nargs=3, default=None)
    ap.add_argument("--inward", action="store_true",
                    help="inward normals (periphery convention)")
    ap.add_argument("--nucleation-sites", type=int, default=0)
    ap.add_argument("--attachment-scale", type=float, default=1.1,
                    help="site radius / surface radius (sites must sit OFF "
                         "the quadrature surface)")
    ap.add_argument("--envelope", type=float, nargs=4, default=None,
                    metavar=("T", "P1", "P2", "LENGTH"),
                    help="surface_of_revolution envelope parameters "
                         "(oocyte defaults: 0.72 0.4 0.2 7.5)")
    ap.add_argument("--scale-factor", type=float, default=1.04)
    ap.add_argument("--out", required=True)
    args = ap.parse_args()

    if args.shape == "sphere":
        g = sphere_geometry(args.n_nodes, args.radius,
                            inward_normals=args.inward)
    elif args.shape == "ellipsoid":
        if args.abc is None:
            ap.error("ellipsoid requires --abc a b c")
        g = ellipsoid_geometry(args.n_nodes, *args.abc,
                               inward_normals=args.inward)
    else:
        from skellysim_amd.precompute import surface_of_revolution_geometry
        if args.envelope is None:
            ap.error("surface_of_revolution requires --envelope T p1 p2 L")
        g = surface_of_revolution_geometry(args.n_nodes, *args.envelope,
                                           scale_factor=args.scale_factor,
                                           inward_normals=args.inward)
    out = dict(nodes=g["nodes"], normals=g["normals"], weights=g["weights"],
               quadrature_weights=g["weights"])
    for k in ("radius", "a", "b", "c", "envelope_T", "envelope_p1",
              "envelope_p2", "envelope_length", "scale_factor"):
        if k in g:
            out[k] = g[k]
    if args.nucleation_sites:
        out["nucleation_sites"] = fibonacci_sphere(
            args.nucleation_sites, args.attachment_scale * args.radius)
    np.savez(args.out, **out)
    print(f"wrote {args.out}: {len(g['nodes'])} nodes, "
          f"area {g['weights'].sum():.6f}")


if __name__ == "__main__":
    main()
