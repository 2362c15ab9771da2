"""Listener service — the reference's post-processing protocol
(src/core/listener.cpp) served by this engine: size-prefixed msgpack
requests on stdin, responses on stdout, against a trajectory file.

Wire contract (listener.cpp:86-137 / reader.py Listener.request):
  request  — uint64-LE size + msgpack map {frame_no, evaluator,
             streamlines{...}, vortexlines{...}, velocity_field{x}}
             (arrays __eigen__-encoded by the reference client)
  response — uint64-LE size + msgpack map {time, i_frame, n_frames,
             streamlines, vortexlines, velocity_field}
  size 0 in → terminate; invalid frame -> size-0 response (listener.cpp:110-115)

Velocity-field semantics mirror System::velocity_at_targets
(system.cpp:330-384) for fibers + shell + bodies + sources: fiber forces
are force_operator @ fiber_solution (apply_fiber_force on the frame's
positions+tension), fiber flow WITHOUT self-subtraction (flow(..., false),
system.cpp:355), plus the shell double layer from the frame's
solution_vec_, the bodies' double layer + link-force center
stokeslet/rotlet with the interior rigid-velocity override
(system.cpp:349-371), and the configured point/background sources.
Streamlines are integrated per the reference's adaptive
5(4) RK scheme (see integrate_streamline); vortex lines are streamlines of
the central-difference curl of the velocity field (streamline.cpp:16-35,
115-165), with the singularity stop still on the VELOCITY norm
(streamline.cpp:51, same observer for both line kinds)."""

import argparse
import os
import struct
import sys

import msgpack
import numpy as np

from .fiber_fd import FiberFD


def eigen_decode(d):
    """reader.py::_eigen_to_numpy (same rules)."""
    if isinstance(d, list):
        if d and d[0] == "__eigen__":
            if d[1] == 1 or d[2] == 1:
                return np.array(d[3:])
            if d[1] == 3:
                return np.array(d[3:]).reshape((d[2], d[1]))
            return np.array(d[3:]).reshape((d[2], d[1])).transpose()
        return [eigen_decode(x) for x in d]
    if isinstance(d, dict):
        return {k: eigen_decode(v) for k, v in d.items()}
    return d


def eigen_encode_3xn(a):
    """(n, 3) -> ['__eigen__', 3, n, colmajor] (eigen_matrix_plugin.h)."""
    a = np.asarray(a, dtype=np.float64)
    return ["__eigen__", 3, len(a), *a.reshape(-1).tolist()]


class Trajectory:
    """Minimal reader of the reference-format trajectory stream."""

    def __init__(self, path):
        self.frames = []
        with open(path, "rb") as fh:
            unp = msgpack.Unpacker(fh, raw=False)
            first = next(unp)
            if isinstance(first, dict) and "trajversion" in first:
                self.header = first
            else:
                self.header = {}
                self.frames.append(eigen_decode(first))
            for obj in unp:
                self.frames.append(eigen_decode(obj))

    def __len__(self):
        return len(self.frames)


def fibers_from_frame(frame, eta):
    ftype, fmaps = frame["fibers"]
    fibers = []
    for fm in fmaps:
        f = FiberFD(fm["x_"], length=fm["length_"],
                    bending_rigidity=fm["bending_rigidity_"], eta=eta,
                    radius=fm["radius_"], force_scale=fm["force_scale_"],
                    minus_clamped=bool(fm["minus_clamped_"]),
                    penalty_param=fm["penalty_param_"],
                    beta_tstep=fm["beta_tstep_"])
        f.length_prev = fm["length_prev_"]
        f.binding_site = tuple(fm.get("binding_site_", (-1, -1)))
        t = np.asarray(fm["tension_"], float).reshape(-1)
        f.tension = t if t.size == f.n_nodes else np.zeros(f.n_nodes)
        f.update_derivatives()
        f.update_force_operator()
        fibers.append(f)
    return fibers


def bodies_from_frame(frame, body_geometry):
    """Reconstruct spherical bodies from a frame's body maps
    (body_spherical.hpp:77: radius_, position_, orientation_,
    solution_vec_) and the reference-frame geometry npz/dict
    (nodes/normals/weights, as the precompute emits)."""
    from .body import SphericalBody
    maps = frame.get("bodies", [[], [], []])[0]
    bodies = []
    for m in maps:
        q = m["orientation_"]
        assert q[0] == "__quat__"
        sites = body_geometry.get("nucleation_sites")
        b = SphericalBody(np.asarray(body_geometry["nodes"], float),
                          np.asarray(body_geometry["normals"], float),
                          np.asarray(body_geometry["weights"], float),
                          float(m["radius_"]),
                          position=np.asarray(m["position_"], float).reshape(-1),
                          orientation=np.asarray(q[1:5], float),
                          nucleation_sites_ref=sites)
        sol = np.asarray(m["solution_vec_"], float).reshape(-1)
        if sol.size == b.solution_size:
            b.solution_vec = sol
            n3 = 3 * b.n_nodes
            b.velocity = sol[n3: n3 + 3]
            b.angular_velocity = sol[n3 + 3: n3 + 6]
        bodies.append(b)
    return bodies


def velocity_field(frame, targets, eta, compute, shell_geometry=None,
                   body_geometry=None, sources=None):
    """System::velocity_at_targets (system.cpp:330-384) for fibers
    (+shell/+bodies when geometry is given, +point/background sources when
    `sources` = (PointSourceContainer|None, BackgroundSource|None) is
    given; their clock is the frame time). compute: object with
    stokeslet/stresslet_normal_density/rotlet (a system_fd backend)."""
    targets = np.asarray(targets, float).reshape(-1, 3)
    u = np.zeros((len(targets), 3))
    if sources is not None:
        psc, bs = sources
        if psc is not None:
            u += psc.flow(targets, eta, float(frame.get("time", 0.0)), compute)
        if bs is not None and bs.is_active():
            u += bs.flow(targets, eta)
    fibers = fibers_from_frame(frame, eta)
    if fibers:
        r_src, wf = [], []
        for f in fibers:
            sol = np.concatenate([f.x.reshape(-1), f.tension])
            ff = f.force_operator @ sol  # apply_fiber_force, system.cpp:339
            fn = np.stack([ff[i * f.n_nodes:(i + 1) * f.n_nodes]
                           for i in range(3)], axis=1)
            r_src.append(f.x.T)
            wf.append(fn * f.quadrature_weights()[:, None])
        # fc_->flow(..., subtract_self=false), system.cpp:355
        u += compute.stokeslet(np.concatenate(r_src), np.concatenate(wf),
                               targets, eta)
    if shell_geometry is not None and "shell" in frame:
        dens = np.asarray(frame["shell"].get("solution_vec_", np.zeros(0)),
                          float).reshape(-1)
        if dens.size:
            u += compute.stresslet_normal_density(
                shell_geometry["nodes"], shell_geometry["normals"],
                dens.reshape(-1, 3), targets, eta)
    bodies = bodies_from_frame(frame, body_geometry) \
        if body_geometry is not None else []
    if bodies:
        from .body import calculate_link_conditions
        # bc_.flow with the frame solution: double layer + center
        # stokeslet/rotlet of the LINK forces (system.cpp:349-357; external
        # forces are zeroed there, per the reference's own comment)
        nodes = np.concatenate([b.nodes for b in bodies])
        normals = np.concatenate([b.normals for b in bodies])
        dens = np.concatenate([b.solution_vec[: 3 * b.n_nodes].reshape(-1, 3)
                               for b in bodies])
        u += compute.stresslet_normal_density(nodes, normals, dens, targets, eta)
        body_vels = np.stack([np.concatenate([b.velocity, b.angular_velocity])
                              for b in bodies])
        x_fib = np.concatenate([np.concatenate([f.x.reshape(-1), f.tension])
                                for f in fibers]) if fibers else np.zeros(0)
        _, ft = calculate_link_conditions(fibers, x_fib, body_vels, bodies)
        centers = np.stack([b.position for b in bodies])
        u += compute.stokeslet(centers, ft[:, 0:3], targets, eta)
        u += compute.rotlet(centers, ft[:, 3:6], targets, eta)
        # points inside a body move rigidly (system.cpp:363-371)
        for b in bodies:
            dx = targets - b.position[None, :]
            inside = np.linalg.norm(dx, axis=1) < b.radius
            if inside.any():
                u[inside] = b.velocity[None, :] + \
                    np.cross(np.broadcast_to(b.angular_velocity, (int(inside.sum()), 3)),
                             dx[inside])
    return u


def vorticity(field_fn, pts, eps=1e-7):
    """Central-difference curl of the velocity field at each row of pts
    (get_vorticity_at_point, streamline.cpp:16-35: eps=1e-7, 6 evals/point;
    0.5/eps times the +eps/-eps differences = the standard central-difference
    curl)."""
    pts = np.asarray(pts, float).reshape(-1, 3)
    n = len(pts)
    probes = np.repeat(pts, 6, axis=0)
    for k in range(3):
        probes[2 * k::6, k] += eps
        probes[2 * k + 1::6, k] -= eps
    v = field_fn(probes).reshape(n, 6, 3)
    return 0.5 / eps * np.stack([
        (v[:, 2, 2] - v[:, 3, 2]) - (v[:, 4, 1] - v[:, 5, 1]),
        (v[:, 4, 0] - v[:, 5, 0]) - (v[:, 0, 2] - v[:, 1, 2]),
        (v[:, 0, 1] - v[:, 1, 1]) - (v[:, 2, 0] - v[:, 3, 0]),
    ], axis=1)


def integrate_streamline(field_fn, x0, dt_init=0.1, t_final=1.0, abs_err=1e-10,
                         rel_err=1e-6, back_integrate=True, rhs_fn=None):
    """One streamline (StreamLine::compute, src/core/streamline.cpp:66-118):
    adaptive 5(4) Runge-Kutta integration of dx/dt = u(x) forward to t_final
    (and backward when requested), with the reference's |u| > 1e3 singularity
    stop. The reference uses Boost odeint's Cash-Karp 5(4); scipy's RK45
    (Dormand-Prince 5(4)) integrates the same ODE to the same tolerances —
    step placement differs, the curve does not.

    rhs_fn overrides the integrated field (vortex lines integrate the curl,
    VortexLine::compute streamline.cpp:115-165) while the singularity stop
    stays on the VELOCITY field_fn (streamline.cpp:51).

    Returns dict {x: (npts, 3), val: (npts, 3), time: [..]} (streamline.hpp:29),
    val being the integrated field at the path points (velocity for
    streamlines, vorticity for vortex lines)."""
    from scipy.integrate import solve_ivp

    rfn = rhs_fn if rhs_fn is not None else field_fn

    def rhs(t, x):
        return rfn(x.reshape(1, 3)).reshape(3)

    def singularity(t, x):
        return 1e3 - np.linalg.norm(field_fn(x.reshape(1, 3)))

    singularity.terminal = True

    def run(t_end, first):
        sol = solve_ivp(rhs, (0.0, t_end), np.asarray(x0, float).reshape(3),
                        method="RK45", first_step=abs(first), atol=abs_err,
                        rtol=rel_err, events=singularity, dense_output=False)
        return sol.t, sol.y.T

    t_fwd, x_fwd = run(t_final, dt_init)
    if back_integrate:
        t_back, x_back = run(-t_final, dt_init)
        # join: reversed backward path (minus the duplicate seed) + forward
        # (streamline.cpp:55-64)
        t = np.concatenate([t_back[:0:-1], t_fwd])
        x = np.concatenate([x_back[:0:-1], x_fwd])
    else:
        t, x = t_fwd, x_fwd
    val = np.stack([rfn(p.reshape(1, 3)).reshape(3) for p in x])
    return {"x": x, "val": val, "time": t.tolist()}


def process_streamlines(frame, req, eta, compute, shell_geometry,
                        vortex=False, body_geometry=None, sources=None):
    """process_streamlines / process_vortexlines (listener.cpp:51-74): one
    line per seed column; vortex=True integrates the curl field instead."""
    req = req or {}
    x0 = eigen_decode(req.get("x0", []))
    x0 = np.asarray(x0, float).reshape(-1, 3) if np.size(x0) else np.zeros((0, 3))
    if not len(x0):
        return []
    field = lambda pts: velocity_field(frame, pts, eta, compute,
                                       shell_geometry, body_geometry,
                                       sources)
    rhs_fn = (lambda pts: vorticity(field, pts)) if vortex else None
    out = []
    for seed in x0:
        s = integrate_streamline(field, seed,
                                 dt_init=float(req.get("dt_init", 0.1)),
                                 t_final=float(req.get("t_final", 1.0)),
                                 abs_err=float(req.get("abs_err", 1e-10)),
                                 rel_err=float(req.get("rel_err", 1e-6)),
                                 back_integrate=bool(req.get("back_integrate", True)),
                                 rhs_fn=rhs_fn)
        out.append({"x": eigen_encode_3xn(s["x"]),
                    "val": eigen_encode_3xn(s["val"]),
                    "time": s["time"]})
    return out


def serve(stdin, stdout, traj, compute, eta=1.0, shell_geometry=None,
          body_geometry=None, sources=None):
    """The stdin/stdout request loop (listener.cpp:86-137)."""
    while True:
        raw = stdin.read(8)
        if len(raw) < 8:
            return
        (msgsize,) = struct.unpack("<Q", raw)
        if msgsize == 0:
            return
        payload = b""
        while len(payload) < msgsize:
            chunk = stdin.read(msgsize - len(payload))
            if not chunk:
                return
            payload += chunk
        cmd = msgpack.unpackb(payload, raw=False)
        frame_no = int(cmd.get("frame_no", 0))
        if frame_no < 0 or frame_no >= len(traj):
            stdout.write(struct.pack("<Q", 0))
            stdout.flush()
            continue
        frame = traj.frames[frame_no]
        vf = cmd.get("velocity_field", {}) or {}
        x = eigen_decode(vf.get("x", []))
        x = np.asarray(x, float).reshape(-1, 3) if np.size(x) else np.zeros((0, 3))
        u = velocity_field(frame, x, eta, compute, shell_geometry,
                           body_geometry, sources) \
            if len(x) else np.zeros((0, 3))
        response = {
            "time": float(frame["time"]),
            "i_frame": frame_no,
            "n_frames": len(traj),
            "streamlines": process_streamlines(frame, cmd.get("streamlines"),
                                               eta, compute, shell_geometry,
                                               body_geometry=body_geometry,
                                               sources=sources),
            "vortexlines": process_streamlines(frame, cmd.get("vortexlines"),
                                               eta, compute, shell_geometry,
                                               vortex=True,
                                               body_geometry=body_geometry,
                                               sources=sources),
            "velocity_field": eigen_encode_3xn(u),
        }
        out = msgpack.packb(response)
        stdout.write(struct.pack("<Q", len(out)))
        stdout.write(out)
        stdout.flush()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trajectory", default="skelly_sim.out")
    ap.add_argument("--eta", type=float, default=1.0)
    ap.add_argument("--shell-geometry", default=None,
                    help="npz with nodes/normals (the precompute geometry)")
    ap.add_argument("--body-geometry", default=None,
                    help="npz with nodes/normals/weights (body reference "
                         "frame geometry from the precompute)")
    ap.add_argument("--config-file", default=None,
                    help="reference TOML config; supplies point/background "
                         "sources and eta for the velocity field")
    ap.add_argument("--listen", action="store_true",
                    help="compatibility no-op: the reference client spawns "
                         "`<binary> --listen` (reader.py:155); this CLI "
                         "always listens")
    args = ap.parse_args()
    # the reference convention: skelly_config.toml in the working directory
    # (reader.py:138); pick it up when present and not overridden
    if args.config_file is None and os.path.exists("skelly_config.toml"):
        args.config_file = "skelly_config.toml"

    from .system_fd import HipBackend  # product path: the MI355X engine
    compute = HipBackend()
    shell_geometry = None
    if args.shell_geometry:
        from .config import _geometry_fields
        fx = np.load(args.shell_geometry)
        nodes, normals, _ = _geometry_fields(fx)
        shell_geometry = {"nodes": nodes, "normals": normals}
    body_geometry = None
    if args.body_geometry:
        from .config import _geometry_fields
        bx = np.load(args.body_geometry)
        nodes, normals, weights = _geometry_fields(bx)
        body_geometry = {"nodes": nodes, "normals": normals,
                         "weights": weights}
        if "nucleation_sites" in bx:
            body_geometry["nucleation_sites"] = bx["nucleation_sites"]
    sources = None
    eta = args.eta
    if args.config_file:
        from .config import load_config
        from .sources import PointSourceContainer, BackgroundSource
        cfg = load_config(args.config_file)
        eta = cfg.get("params", {}).get("eta", eta)
        psc = PointSourceContainer.from_config(cfg["point_sources"]) \
            if cfg.get("point_sources") else None
        bs = BackgroundSource.from_config(cfg["background"]) \
            if "background" in cfg else None
        if psc is not None or bs is not None:
            sources = (psc, bs)
    traj = Trajectory(args.trajectory)
    serve(sys.stdin.buffer, sys.stdout.buffer, traj, compute, eta=eta,
          shell_geometry=shell_geometry, body_geometry=body_geometry,
          sources=sources)


if __name__ == "__main__":
    main()
