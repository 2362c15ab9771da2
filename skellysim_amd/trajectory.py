"""Reference-format trajectory output: writes the
msgpack stream SkellySim's own Python tooling (src/skelly_sim/reader.py)
reads, so the reference's analysis/paraview ecosystem consumes this engine's
output unchanged.

Format contract (verified against the reference sources):
  header  — header_map_t (include/io_maps.hpp:44-57): map {trajversion=1
            (CMakeLists.txt:12), number_mpi_ranks, fiber_type,
            skellysim_version, skellysim_commit, simdate, hostname}
  frame   — output_map_t (io_maps.hpp:31-41): map {time, dt, rng_state,
            fibers, bodies, shell}
    fibers — FiberContainerFiniteDifference, packed as the ARRAY
             [fiber_type_, [fiber maps]] (MSGPACK_DEFINE,
             fiber_container_finite_difference.hpp:126; FIBERTYPE
             FiniteDifference == 1, fiber_container_base.hpp:19)
    fiber  — map with the keys of fiber_finite_difference.hpp:160-161
    bodies — [spherical, deformable, ellipsoidal] lists (body_container.hpp:158)
    shell  — map {solution_vec_} (periphery.hpp:120)
    Eigen matrices — ['__eigen__', rows, cols, column-major values]
             (include/eigen_matrix_plugin.h:30-41)
"""

import datetime
import socket

import msgpack
import numpy as np

TRAJECTORY_VERSION = 1  # reference CMakeLists.txt:12
FIBERTYPE_FINITEDIFFERENCE = 1  # fiber_container_base.hpp:19


def eigen(a):
    """['__eigen__', rows, cols, col-major...] (eigen_matrix_plugin.h).
    (n, 3) numpy arrays become 3 x n Eigen matrices (point-major values are
    identical); 1-D arrays become n x 1 vectors."""
    a = np.asarray(a, dtype=np.float64)
    if a.ndim == 1:
        return ["__eigen__", a.size, 1] + a.tolist()
    if a.ndim == 2 and a.shape[1] == 3:
        return ["__eigen__", 3, a.shape[0]] + a.reshape(-1).tolist()
    if a.ndim == 2:  # general: emit as (rows, cols) col-major
        return ["__eigen__", a.shape[0], a.shape[1]] + a.T.reshape(-1).tolist()
    raise ValueError(f"unsupported shape {a.shape}")


def body_map(b):
    """SphericalBody msgpack map (body_spherical.hpp:77:
    radius_, position_, orientation_, solution_vec_); quaternions are
    ['__quat__', w, x, y, z] (eigen_quaternion_plugin.h)."""
    w, x, y, z = (float(v) for v in b.orientation)
    return {
        "radius_": b.radius,
        "position_": eigen(b.position),
        "orientation_": ["__quat__", w, x, y, z],
        "solution_vec_": eigen(np.asarray(b.solution_vec)),
    }


def fiber_map(f):
    """fiber_finite_difference.hpp:160-161 (non-resume field set)."""
    return {
        "n_nodes_": f.n_nodes,
        "radius_": f.radius,
        "length_": f.length,
        "length_prev_": f.length_prev,
        "bending_rigidity_": f.bending_rigidity,
        "penalty_param_": f.penalty_param,
        "force_scale_": f.force_scale,
        "beta_tstep_": f.beta_tstep,
        "binding_site_": list(getattr(f, "binding_site", (-1, -1))),
        "tension_": eigen(f.tension),
        "x_": eigen(f.x.T),  # (n, 3) -> 3 x n Eigen
        "minus_clamped_": f.minus_clamped,
    }


class TrajectoryWriter:
    """Streams reference-format frames for a SystemFD run. append=True
    continues an existing trajectory without re-writing the header (the
    reference's resume path, system.cpp:708-712)."""

    def __init__(self, path, version="skelly-hip-0.1.0", commit="unknown",
                 append=False):
        if append:
            self._fh = open(path, "ab")
            return
        self._fh = open(path, "wb")
        header = {
            "trajversion": TRAJECTORY_VERSION,
            "number_mpi_ranks": 1,
            "fiber_type": FIBERTYPE_FINITEDIFFERENCE,
            "skellysim_version": version,
            "skellysim_commit": commit,
            "simdate": datetime.datetime.now().isoformat(),
            "hostname": socket.gethostname(),
        }
        self._fh.write(msgpack.packb(header))

    def write_frame(self, system, time, dt):
        # shell block only: [fibers | shell | bodies] layout (system.cpp:78)
        a = system.fiber_sol_size
        b = a + system.shell_sol_size
        shell_sol = (system.solution[a:b]
                     if getattr(system, "solution", None) is not None
                     and system.shell else np.zeros(0))
        frame = {
            "time": float(time),
            "dt": float(dt),
            # the reference stores its Philox counters here
            # (io_maps.hpp:31-41); this engine stores its numpy PCG64 state
            # (128-bit ints as decimal strings, msgpack has no int128) so a
            # resumed dynamic-instability run continues the same stream
            "rng_state": _pack_rng(getattr(system, "rng", None)),
            "fibers": [FIBERTYPE_FINITEDIFFERENCE,
                       [fiber_map(f) for f in system.fibers]],
            # [spherical, deformable, ellipsoidal] (body_container.hpp:158)
            "bodies": [[body_map(b) for b in getattr(system, "bodies", [])],
                       [], []],
            "shell": {"solution_vec_": eigen(np.asarray(shell_sol))},
        }
        self._fh.write(msgpack.packb(frame))
        self._fh.flush()

    def close(self):
        self._fh.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


def _pack_rng(rng):
    """Serialize as a list of [string, string] pairs — the same msgpack
    shape as the reference's vector<pair<string,string>> rng_state
    (io_maps.hpp:31-41), so the reference binary can at least parse (and
    skip) the field; the VALUES are this engine's numpy PCG64 state
    (128-bit ints as decimal strings), not Philox counters."""
    if rng is None:
        return []
    st = rng.bit_generator.state
    pairs = [["bit_generator", str(st["bit_generator"])]]
    pairs += [["state." + k, str(v)] for k, v in st["state"].items()]
    pairs += [["has_uint32", str(int(st.get("has_uint32", 0)))],
              ["uinteger", str(int(st.get("uinteger", 0)))]]
    return pairs


def _restore_rng(rng, packed):
    if rng is None:
        return
    if isinstance(packed, (list, tuple)):  # current pair-list format
        packed = {k: v for k, v in packed}
        if packed.get("bit_generator") != \
                rng.bit_generator.state["bit_generator"]:
            return
        rng.bit_generator.state = {
            "bit_generator": packed["bit_generator"],
            "state": {k[len("state."):]: int(v) for k, v in packed.items()
                      if k.startswith("state.")},
            "has_uint32": int(packed.get("has_uint32", 0)),
            "uinteger": int(packed.get("uinteger", 0))}
        return
    if not isinstance(packed, dict):  # legacy map format
        return
    if packed.get("bit_generator") != rng.bit_generator.state["bit_generator"]:
        return
    rng.bit_generator.state = {
        "bit_generator": packed["bit_generator"],
        "state": {k: int(v) for k, v in packed["state"].items()},
        "has_uint32": int(packed.get("has_uint32", 0)),
        "uinteger": int(packed.get("uinteger", 0))}


def resume_from_trajectory(system, path):
    """System::resume_from_trajectory (system.cpp:223-228 +
    TrajectoryReader::unpack_current_frame): restore the system to the LAST
    frame of an existing trajectory — fibers rebuilt from their serialized
    state, body positions/orientations/solutions restored onto the
    configured bodies, the clock and dt adopted. Returns the frame count.
    The rng_state field (this engine's PCG64 state, or the reference's
    Philox counters which are skipped) is restored so a resumed
    dynamic-instability run continues the exact random stream."""
    from .listener import Trajectory, fibers_from_frame
    traj = Trajectory(path)
    if not traj.frames:
        raise ValueError(f"no frames in {path}")
    frame = traj.frames[-1]
    system.time = float(frame["time"])
    system.dt = float(frame["dt"])
    _restore_rng(getattr(system, "rng", None), frame.get("rng_state"))
    system.fibers = fibers_from_frame(frame, system.eta)
    system._uniform = all(f.n_nodes == system.fibers[0].n_nodes
                          for f in system.fibers) if system.fibers else True
    body_maps = frame.get("bodies", [[], [], []])[0]
    if len(body_maps) != len(system.bodies):
        raise ValueError("trajectory body count does not match the system")
    for b, m in zip(system.bodies, body_maps):
        q = m["orientation_"]
        b.place(np.asarray(m["position_"], float).reshape(-1),
                np.asarray(q[1:5], float))
        sol = np.asarray(m["solution_vec_"], float).reshape(-1)
        if sol.size == b.solution_size:
            b.solution_vec = sol
            n3 = 3 * b.n_nodes
            b.velocity = sol[n3: n3 + 3].copy()
            b.angular_velocity = sol[n3 + 3: n3 + 6].copy()
    # reassemble the global solution vector (used by frame writes and
    # velocity fields before the next solve)
    parts = [np.concatenate([f.x.reshape(-1), f.tension])
             for f in system.fibers]
    if system.shell is not None:
        sh = np.asarray(frame.get("shell", {}).get("solution_vec_",
                                                   np.zeros(0)), float)
        parts.append(sh.reshape(-1))
    parts += [b.solution_vec for b in system.bodies]
    system.solution = np.concatenate(parts) if parts else np.zeros(0)
    return len(traj)
