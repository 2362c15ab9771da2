"""CPU tests of the listener protocol (skellysim_amd/listener.py) against the
reference's wire format: requests encoded exactly as reader.py's
Listener.request does (__eigen__ ndencode), responses decoded with its
_eigen_to_numpy rules."""

import io
import struct

import msgpack
import numpy as np
import pytest

from skellysim_amd.fiber_fd import FiberFD
from skellysim_amd.system_fd import SystemFD
from oracle_backend import OracleBackend
from skellysim_amd.trajectory import TrajectoryWriter
from skellysim_amd.listener import Trajectory, serve, eigen_decode


def _ndencode(obj):  # reader.py:15-19
    if isinstance(obj, np.ndarray):
        return ["__eigen__", obj.shape[1], obj.shape[0]] + obj.ravel().tolist()
    return obj


def _request_bytes(frame_no, x):
    cmd = {
        "frame_no": frame_no,
        "evaluator": "GPU",
        "streamlines": {"dt_init": 0.1, "t_final": 1.0, "abs_err": 1e-10,
                        "rel_err": 1e-6, "back_integrate": True,
                        "x0": np.zeros((0, 3))},
        "vortexlines": {"dt_init": 0.1, "t_final": 1.0, "abs_err": 1e-10,
                        "rel_err": 1e-6, "back_integrate": True,
                        "x0": np.zeros((0, 3))},
        "velocity_field": {"x": np.asarray(x, float)},
    }
    msg = msgpack.packb(cmd, default=_ndencode)
    return struct.pack("<Q", len(msg)) + msg


def _write_traj(tmp_path, steps=2):
    s = np.linspace(0, 1.0, 16)
    fibers = [FiberFD(np.array([0.2 * k, 0, 0])[None, :] +
                      s[:, None] * np.array([0.0, 0, 1.0])[None, :],
                      length=1.0, bending_rigidity=2.5e-3, eta=1.0,
                      force_scale=-0.02)
              for k in range(2)]
    sys_ = SystemFD(fibers, eta=1.0, dt=0.1, backend=OracleBackend(),
                    background_flow=lambda r: np.tile([0.1, 0, 0], (len(r), 1)))
    path = str(tmp_path / "skelly_sim.out")
    with TrajectoryWriter(path) as tw:
        t = 0.0
        for _ in range(steps):
            assert sys_.step(tol=1e-11, maxiter=200)["converged"]
            t += sys_.dt
            tw.write_frame(sys_, t, sys_.dt)
    return path, sys_


def _roundtrip(path, requests, compute):
    traj = Trajectory(path)
    stdin = io.BytesIO(b"".join(requests) + struct.pack("<Q", 0))
    stdout = io.BytesIO()
    serve(stdin, stdout, traj, compute, eta=1.0)
    stdout.seek(0)
    responses = []
    while True:
        raw = stdout.read(8)
        if len(raw) < 8:
            break
        (size,) = struct.unpack("<Q", raw)
        if size == 0:
            responses.append(None)
            continue
        responses.append(msgpack.unpackb(stdout.read(size), raw=False))
    return responses


def test_velocity_field_request(tmp_path):
    path, sys_ = _write_traj(tmp_path)
    targets = np.array([[0.5, 0.3, 0.5], [1.0, -0.2, 0.1], [0.0, 0.0, 2.0]])
    reqs = [_request_bytes(1, targets)]
    (res,) = _roundtrip(path, reqs, OracleBackend())
    assert res["i_frame"] == 1 and res["n_frames"] == 2
    assert res["time"] == pytest.approx(0.2)
    u = eigen_decode(res["velocity_field"])
    assert u.shape == (3, 3)

    # semantics: flow of force_operator@solution, quadrature-weighted, NO
    # self subtraction (system.cpp:339,355) — recompute directly
    import oracle
    r_src, wf = [], []
    for f in sys_.fibers:
        sol = np.concatenate([f.x.reshape(-1), f.tension])
        ff = f.force_operator @ sol
        fn = np.stack([ff[i * f.n_nodes:(i + 1) * f.n_nodes] for i in range(3)],
                      axis=1)
        r_src.append(f.x.T)
        wf.append(fn * f.quadrature_weights()[:, None])
    ref = oracle.stokeslet(np.concatenate(r_src), np.concatenate(wf), targets, 1.0)
    assert np.allclose(u, ref, atol=1e-12)


def test_streamline_request(tmp_path):
    """Streamlines: seeds integrate along the frame's velocity field
    (midpoint finite differences of the returned path match the returned
    velocities), back+forward paths joined with monotone time. Uses a BENT
    fiber (a straight unloaded fiber exerts zero bending force -> zero
    field)."""
    s = np.linspace(0, 1.0, 24)
    x = np.stack([0.15 * np.sin(2 * np.pi * s), np.zeros_like(s), s], axis=1)
    fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-2, eta=1.0)
    sys_ = SystemFD([fib], eta=1.0, dt=0.1, backend=OracleBackend())
    path = str(tmp_path / "skelly_sim.out")
    with TrajectoryWriter(path) as tw:
        tw.write_frame(sys_, 0.1, 0.1)
    cmd = {
        "frame_no": 0,
        "evaluator": "GPU",
        "streamlines": {"dt_init": 0.05, "t_final": 0.5, "abs_err": 1e-10,
                        "rel_err": 1e-8, "back_integrate": True,
                        "x0": np.array([[0.6, 0.2, 0.4]])},
        "vortexlines": {"dt_init": 0.1, "t_final": 1.0, "abs_err": 1e-10,
                        "rel_err": 1e-6, "back_integrate": True,
                        "x0": np.zeros((0, 3))},
        "velocity_field": {"x": np.zeros((0, 3))},
    }
    msg = msgpack.packb(cmd, default=_ndencode)
    (res,) = _roundtrip(path, [struct.pack("<Q", len(msg)) + msg], OracleBackend())
    assert len(res["streamlines"]) == 1
    sl = res["streamlines"][0]
    x = eigen_decode(sl["x"])
    val = eigen_decode(sl["val"])
    t = np.asarray(sl["time"])
    assert x.shape == val.shape and x.shape[1] == 3 and len(t) == len(x)
    assert np.all(np.diff(t) > 0) and t[0] < 0 < t[-1]
    # the seed lies on the path at t == 0
    i0 = np.argmin(np.abs(t))
    assert np.allclose(x[i0], [0.6, 0.2, 0.4], atol=1e-12)
    # midpoint finite differences track the reported velocities
    dxdt = (x[2:] - x[:-2]) / (t[2:] - t[:-2])[:, None]
    mid = val[1:-1]
    mask = np.linalg.norm(mid, axis=1) > 1e-8
    relerr = np.linalg.norm(dxdt[mask] - mid[mask], axis=1) / \
        np.linalg.norm(mid[mask], axis=1)
    assert np.median(relerr) < 0.25


def test_vorticity_helper():
    """vorticity() = central-difference curl (streamline.cpp:16-35): exact
    on rigid rotation (curl = 2*Omega), ~1e-8 on the analytic Stokeslet
    vorticity f x r / (4 pi eta r^3)."""
    from skellysim_amd.listener import vorticity
    Om = np.array([0.3, -0.7, 0.5])
    rot = lambda p: np.cross(np.tile(Om, (len(p), 1)), p)
    pts = np.array([[0.2, 0.1, -0.3], [1.0, 2.0, 0.5]])
    w = vorticity(rot, pts)
    assert np.allclose(w, 2 * Om, atol=1e-7)

    import oracle
    src = np.array([[0.1, -0.2, 0.3]])
    f = np.array([[0.7, 0.4, -0.9]])
    eta = 1.3
    trg = np.array([[1.5, 2.1, 1.7], [-1.8, 0.9, 2.2]])
    field = lambda p: oracle.stokeslet(src, f, p, eta)
    w = vorticity(field, trg)
    r = trg - src
    wa = np.cross(np.tile(f, (len(trg), 1)), r) \
        / (4 * np.pi * eta * np.linalg.norm(r, axis=1)[:, None] ** 3)
    assert np.linalg.norm(w - wa) / np.linalg.norm(wa) < 1e-6


def test_vortexline_request(tmp_path):
    """Vortex lines: seeds integrate along the CURL of the frame's velocity
    field (VortexLine::compute, streamline.cpp:115-165) — path tangents match
    the returned vorticities, and val recomputes as vorticity()."""
    from skellysim_amd.listener import vorticity, velocity_field
    s = np.linspace(0, 1.0, 24)
    x = np.stack([0.15 * np.sin(2 * np.pi * s), np.zeros_like(s), s], axis=1)
    fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-2, eta=1.0)
    sys_ = SystemFD([fib], eta=1.0, dt=0.1, backend=OracleBackend())
    path = str(tmp_path / "skelly_sim.out")
    with TrajectoryWriter(path) as tw:
        tw.write_frame(sys_, 0.1, 0.1)
    cmd = {
        "frame_no": 0,
        "evaluator": "GPU",
        "streamlines": {"dt_init": 0.1, "t_final": 1.0, "abs_err": 1e-10,
                        "rel_err": 1e-6, "back_integrate": True,
                        "x0": np.zeros((0, 3))},
        "vortexlines": {"dt_init": 0.05, "t_final": 0.3, "abs_err": 1e-10,
                        "rel_err": 1e-8, "back_integrate": True,
                        "x0": np.array([[0.6, 0.2, 0.4]])},
        "velocity_field": {"x": np.zeros((0, 3))},
    }
    msg = msgpack.packb(cmd, default=_ndencode)
    (res,) = _roundtrip(path, [struct.pack("<Q", len(msg)) + msg], OracleBackend())
    assert res["streamlines"] == []
    assert len(res["vortexlines"]) == 1
    vl = res["vortexlines"][0]
    x = eigen_decode(vl["x"])
    val = eigen_decode(vl["val"])
    t = np.asarray(vl["time"])
    assert x.shape == val.shape and x.shape[1] == 3 and len(t) == len(x)
    assert np.all(np.diff(t) > 0) and t[0] < 0 < t[-1]
    i0 = np.argmin(np.abs(t))
    assert np.allclose(x[i0], [0.6, 0.2, 0.4], atol=1e-12)
    # val really is the vorticity of the frame's field at the path points
    traj = Trajectory(path)
    field = lambda p: velocity_field(traj.frames[0], p, 1.0, OracleBackend())
    w = vorticity(field, x)
    assert np.linalg.norm(w - val) / np.linalg.norm(w) < 1e-5
    # path tangents track the vorticity (the integrated field)
    dxdt = (x[2:] - x[:-2]) / (t[2:] - t[:-2])[:, None]
    mid = val[1:-1]
    mask = np.linalg.norm(mid, axis=1) > 1e-8
    relerr = np.linalg.norm(dxdt[mask] - mid[mask], axis=1) / \
        np.linalg.norm(mid[mask], axis=1)
    assert np.median(relerr) < 0.25


def test_velocity_field_with_shell(tmp_path):
    """Shell branch of listener velocity_field: the frame's solution_vec_
    drives the double-layer term (system.cpp:330-384, shell_->flow of the
    stored density) on top of the fiber stokeslet flow."""
    import os
    from skellysim_amd.system_fd import Shell
    here = os.path.dirname(os.path.abspath(__file__))
    fx = np.load(os.path.join(here, "golden", "periphery_sphere_192.npz"))
    shell = Shell(fx["nodes"], fx["normals"], fx["stresslet_plus_complementary"],
                  fx["M_inv"])
    s = np.linspace(0, 1.0, 16)
    x = np.stack([0.1 * np.sin(np.pi * s), np.zeros_like(s), 0.6 * s - 0.8],
                 axis=1)
    fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-2, eta=1.0,
                  force_scale=-0.02)
    sys_ = SystemFD([fib], eta=1.0, dt=0.05, shell=shell,
                    backend=OracleBackend())
    path = str(tmp_path / "skelly_sim.out")
    with TrajectoryWriter(path) as tw:
        assert sys_.step(tol=1e-11, maxiter=300)["converged"]
        tw.write_frame(sys_, 0.05, 0.05)
    dens = sys_.solution[sys_.fiber_sol_size:].reshape(-1, 3)
    assert np.linalg.norm(dens) > 0  # the shell really participates

    targets = np.array([[0.3, 0.1, -0.2], [0.0, 0.4, 0.5], [-0.5, 0.0, 0.0]])
    traj = Trajectory(path)
    stdin = io.BytesIO(_request_bytes(0, targets) + struct.pack("<Q", 0))
    stdout = io.BytesIO()
    serve(stdin, stdout, traj, OracleBackend(), eta=1.0,
          shell_geometry={"nodes": fx["nodes"], "normals": fx["normals"]})
    stdout.seek(0)
    (size,) = struct.unpack("<Q", stdout.read(8))
    res = msgpack.unpackb(stdout.read(size), raw=False)
    u = eigen_decode(res["velocity_field"])

    import oracle
    f = sys_.fibers[0]
    f.update_derivatives()        # the listener rebuilds operators from the
    f.update_force_operator()     # frame's (post-step) positions
    sol = np.concatenate([f.x.reshape(-1), f.tension])
    ff = f.force_operator @ sol
    fn = np.stack([ff[i * f.n_nodes:(i + 1) * f.n_nodes] for i in range(3)],
                  axis=1)
    ref = oracle.stokeslet(f.x.T, fn * f.quadrature_weights()[:, None],
                           targets, 1.0)
    f_dl = 2.0 * np.einsum("ni,nj->nij", fx["normals"], dens).reshape(-1, 9)
    ref += oracle.stresslet(fx["nodes"], f_dl, targets, 1.0)
    assert np.allclose(u, ref, atol=1e-12)

    # without shell_geometry the shell term is (documented) absent
    stdin = io.BytesIO(_request_bytes(0, targets) + struct.pack("<Q", 0))
    stdout2 = io.BytesIO()
    serve(stdin, stdout2, traj, OracleBackend(), eta=1.0)
    stdout2.seek(0)
    (size,) = struct.unpack("<Q", stdout2.read(8))
    res2 = msgpack.unpackb(stdout2.read(size), raw=False)
    u2 = eigen_decode(res2["velocity_field"])
    assert not np.allclose(u2, ref, atol=1e-12)


def test_velocity_field_with_bodies(tmp_path):
    """Body branch of the listener velocity field: the frame's body maps
    (radius/position/orientation/solution_vec_) + the reference-frame
    geometry reproduce System::velocity_at_targets' body flow (double layer
    + link-force center stokeslet/rotlet + interior rigid override)."""
    import os
    from skellysim_amd.body import SphericalBody, calculate_link_conditions
    from skellysim_amd.listener import velocity_field
    fx = np.load(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                              "golden", "periphery_sphere_192.npz"))
    R = float(fx["radius"])
    geom = {"nodes": fx["nodes"], "normals": -fx["normals"],
            "weights": fx["quadrature_weights"].reshape(-1),
            "nucleation_sites": np.array([[1.1 * R, 0.0, 0.0]])}
    b = SphericalBody(geom["nodes"], geom["normals"], geom["weights"], R,
                      position=(0.1, -0.2, 0.3),
                      nucleation_sites_ref=np.array([[1.1 * R, 0.0, 0.0]]))
    s0 = np.linspace(0, 1.0, 16)
    x = b.nucleation_sites[0][None, :] + s0[:, None] * np.array([1.0, 0, 0])
    fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-3, eta=1.0,
                  minus_clamped=True, force_scale=-0.05)
    fib.binding_site = (0, 0)
    sys_ = SystemFD([fib], eta=1.0, dt=0.05, bodies=[b],
                    backend=OracleBackend())
    path = str(tmp_path / "skelly_sim.out")
    with TrajectoryWriter(path) as tw:
        assert sys_.step(tol=1e-11, maxiter=300, restart=150)["converged"]
        tw.write_frame(sys_, 0.05, 0.05)
    traj = Trajectory(path)
    frame = traj.frames[0]
    # body round trip through the reference wire format
    bm = frame["bodies"][0][0]
    assert bm["radius_"] == R and bm["orientation_"][0] == "__quat__"
    assert np.allclose(np.asarray(bm["position_"]).reshape(-1), b.position)
    assert np.allclose(np.asarray(bm["solution_vec_"]).reshape(-1),
                       b.solution_vec)

    targets = np.array([[2.5, 0.4, 0.0], [0.0, 2.2, 1.0],
                        b.position + [0.3, 0.1, 0.0]])  # last is INSIDE
    u = velocity_field(frame, targets, 1.0, OracleBackend(),
                       body_geometry=geom)

    import oracle
    f = sys_.fibers[0]
    f.update_derivatives()
    f.update_force_operator()
    sol = np.concatenate([f.x.reshape(-1), f.tension])
    ff = f.force_operator @ sol
    fn = np.stack([ff[i * f.n_nodes:(i + 1) * f.n_nodes] for i in range(3)],
                  axis=1)
    ref = oracle.stokeslet(f.x.T, fn * f.quadrature_weights()[:, None],
                           targets, 1.0)
    dens = b.solution_vec[: 3 * b.n_nodes].reshape(-1, 3)
    f_dl = 2.0 * np.einsum("ni,nj->nij", b.normals, dens).reshape(-1, 9)
    ref += oracle.stresslet(b.nodes, f_dl, targets, 1.0)
    bv = np.concatenate([b.velocity, b.angular_velocity])[None]
    _, ft = calculate_link_conditions([f], sol, bv, [b])
    ref += oracle.stokeslet(b.position[None], ft[:, 0:3], targets, 1.0)
    ref += oracle.rotlet(b.position[None], targets, ft[:, 3:6], 1.0)
    dx = targets[2] - b.position
    ref[2] = b.velocity + np.cross(b.angular_velocity, dx)
    assert np.allclose(u, ref, atol=1e-12), np.abs(u - ref).max()


def test_invalid_frame_gives_empty_response(tmp_path):
    path, _ = _write_traj(tmp_path)
    reqs = [_request_bytes(99, np.zeros((1, 3))), _request_bytes(0, np.zeros((1, 3)))]
    res = _roundtrip(path, reqs, OracleBackend())
    assert res[0] is None          # size-0 response (listener.cpp:110-115)
    assert res[1] is not None      # listener continues serving
    assert res[1]["i_frame"] == 0


@pytest.mark.timeout(300)
def test_listener_cli_subprocess(tmp_path):
    """The CLI process path the reference's Listener client drives
    (reader.py:155 spawns `<binary> --listen`): tools/skelly_sim_hip serves
    the wire protocol over real pipes. Metadata-only request (no compute,
    so it runs without a GPU)."""
    import os
    import subprocess
    path, sys_ = _write_traj(tmp_path)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    launcher = os.path.join(repo, "tools", "skelly_sim_hip")
    cmd = {
        "frame_no": 1, "evaluator": "GPU",
        "streamlines": {"dt_init": 0.1, "t_final": 1.0, "abs_err": 1e-10,
                        "rel_err": 1e-6, "back_integrate": True,
                        "x0": np.zeros((0, 3))},
        "vortexlines": {"dt_init": 0.1, "t_final": 1.0, "abs_err": 1e-10,
                        "rel_err": 1e-6, "back_integrate": True,
                        "x0": np.zeros((0, 3))},
        "velocity_field": {"x": np.zeros((0, 3))},
    }
    msg = msgpack.packb(cmd, default=_ndencode)
    proc = subprocess.Popen([launcher, "--listen", "--trajectory", path],
                            stdin=subprocess.PIPE, stdout=subprocess.PIPE,
                            cwd=str(tmp_path))
    try:
        proc.stdin.write(struct.pack("<Q", len(msg)) + msg)
        proc.stdin.flush()
        (size,) = struct.unpack("<Q", proc.stdout.read(8))
        res = msgpack.unpackb(proc.stdout.read(size), raw=False)
        assert res["i_frame"] == 1 and res["n_frames"] == 2
        assert res["time"] == pytest.approx(0.2)
        assert res["streamlines"] == [] and res["vortexlines"] == []
        proc.stdin.write(struct.pack("<Q", 0))  # terminate (listener.cpp:110)
        proc.stdin.flush()
        assert proc.wait(timeout=30) == 0
    finally:
        if proc.poll() is None:
            proc.kill()
