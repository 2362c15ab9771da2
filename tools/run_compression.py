#!/usr/bin/env python3
"""CPU (oracle-backend) rehearsal of the compression-regression mirror
(tests/compression_common.py): runs the reference protocol end to end and
prints the final-position total error against the reference's pins.
Usage: python tools/run_compression.py [--t-final 5.0] [--hip]"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

import numpy as np

from compression_common import (build_system, run_protocol,
                                final_position_error)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--t-final", type=float, default=5.0)
    ap.add_argument("--hip", action="store_true")
    ap.add_argument("--device-mode", action="store_true",
                    help="opt into the device-resident iteration (with "
                         "bodies present the default is the host matvec)")
    args = ap.parse_args()

    t0 = time.perf_counter()
    if args.hip:
        from skellysim_amd.system_fd import HipBackend
        backend = HipBackend()
    else:
        from oracle_backend import OracleBackend
        backend = OracleBackend()
    s = build_system(backend, device=args.hip)
    print(f"build {time.perf_counter()-t0:.1f}s", flush=True)

    t0 = time.perf_counter()
    frames, hist = run_protocol(s, t_final=args.t_final)
    wall = time.perf_counter() - t0
    print(f"{len(hist)} accepted steps, {len(frames)} frames, {wall:.1f}s "
          f"({len(hist)/wall:.3f} steps/s); "
          f"iters: {[h['iters'] for h in hist[:8]]}...", flush=True)
    print(f"final body pos {frames[-1]['body_pos']}")
    print(f"fiber plus end {frames[-1]['fiber_x'][:, -1]}")
    if args.t_final == 5.0:
        err = final_position_error(frames)
        print(f"final-position total error vs reference pins: {err:.3e} "
              f"(reference gate 1e-5)")
        from compression_common import velocity_field_error
        for cand in (96, 97, 98):
            if cand < len(frames):
                ve, npts = velocity_field_error(frames, backend, frame_no=cand)
                print(f"velocity-field error vs pins at frame {cand} "
                      f"(t={frames[cand]['time']:.3f}, {npts} pts): {ve:.3e}")
    else:
        print("(partial run; pin comparison skipped)")


if __name__ == "__main__":
    main()
