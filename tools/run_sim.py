#!/usr/bin/env python3
"""Simulation driver CLI — the engine's analog of the reference binary
(src/skelly_sim.cpp): reference TOML config in, adaptive timestep loop,
reference-format msgpack trajectory out (readable by the reference's own
TrajectoryReader / analysis tools).

    python tools/run_sim.py --config-file skelly_config.toml \
        --shell-geometry tests/golden/ellipsoid_8192_nodes.npz \
        --t-final 0.1 --out skelly_sim.out

The shell geometry npz (nodes/normals/quadrature_weights) replaces the
reference's precompute .npz; the dense operators are assembled on device."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from skellysim_amd.config import load_config, build_system, periphery_interaction_from
from skellysim_amd.system_fd import HipBackend
from skellysim_amd.trajectory import TrajectoryWriter


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config-file", required=True)
    ap.add_argument("--shell-geometry", default=None)
    ap.add_argument("--body-geometry", default=None,
                    help="npz with nodes/normals/weights (the body "
                         "precompute surface, shared by all bodies)")
    ap.add_argument("--t-final", type=float, default=None,
                    help="override params.t_final")
    ap.add_argument("--out", default="skelly_sim.out")
    ap.add_argument("--gmres-tol", type=float, default=None)
    ap.add_argument("--resume", action="store_true",
                    help="restore the last frame of --out and append "
                         "(the reference's skelly_sim --resume)")
    ap.add_argument("--overwrite", action="store_true",
                    help="allow clobbering an existing trajectory "
                         "(skelly_sim.cpp:48-49)")
    args = ap.parse_args()
    if args.resume and args.overwrite:
        ap.error("--resume and --overwrite are mutually exclusive")
    if args.resume and not os.path.exists(args.out):
        ap.error("--resume supplied without existing trajectory")
    if not args.resume and not args.overwrite and os.path.exists(args.out):
        ap.error(f"existing trajectory {args.out!r}; supply --overwrite")

    cfg = load_config(args.config_file)
    p = cfg.get("params", {})
    t_final = args.t_final if args.t_final is not None else p.get("t_final", 1.0)
    tol = args.gmres_tol if args.gmres_tol is not None else p.get("gmres_tol", 1e-10)

    t0 = time.perf_counter()
    sys_ = build_system(cfg, backend=HipBackend(),
                        shell_geometry=args.shell_geometry,
                        body_geometry=args.body_geometry,
                        config_dir=os.path.dirname(
                            os.path.abspath(args.config_file)))
    print(f"system: {len(sys_.fibers)} fibers"
          + (f" + {sys_.shell.n_nodes}-node shell" if sys_.shell else "")
          + (f" + {len(sys_.bodies)} bodies" if sys_.bodies else "")
          + f", built in {time.perf_counter()-t0:.1f}s", flush=True)

    if args.resume:
        from skellysim_amd.trajectory import resume_from_trajectory
        n_frames = resume_from_trajectory(sys_, args.out)
        print(f"resumed from frame {n_frames - 1} at t={sys_.time:.4f}",
              flush=True)

    dt_write = p.get("dt_write", 0.1)

    with TrajectoryWriter(args.out, append=args.resume) as tw:
        def on_accept(s, t):
            # the reference's write-cadence test (system.cpp:560-561):
            # write when the clock crosses a dt_write boundary
            if int(t / dt_write) > int((t - s.dt) / dt_write):
                tw.write_frame(s, t, s.dt)

        t0 = time.perf_counter()
        hist = sys_.run(t_final=t_final, adaptive=p.get("adaptive_timestep_flag", True),
                        dt_min=p.get("dt_min", 1e-4), dt_max=p.get("dt_max", sys_.dt),
                        beta_up=p.get("beta_up", 1.2), beta_down=p.get("beta_down", 0.5),
                        fiber_error_tol=p.get("fiber_error_tol", 0.1),
                        periphery_shape=periphery_interaction_from(
                            {**cfg, "params": {**p, "periphery_interaction_flag": True}}),
                        tol=tol, on_accept=on_accept)
        wall = time.perf_counter() - t0
    if hist:
        print(f"{len(hist)} accepted steps to t={hist[-1]['time']:.4f} in {wall:.1f}s "
              f"({len(hist)/wall:.3f} steps/s); iters/step: "
              f"{[h['iters'] for h in hist[:12]]}", flush=True)
    else:
        print(f"0 accepted steps (time already >= t_final) in {wall:.1f}s",
              flush=True)


if __name__ == "__main__":
    main()
