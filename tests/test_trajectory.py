"""Trajectory output in the reference's msgpack format (§8f row 4):
structural checks against the serialization contract, and — when the
reference tree is present (build container) — a round trip through the
REFERENCE'S OWN TrajectoryReader (src/skelly_sim/reader.py)."""

import os
import sys
import types

import msgpack
import numpy as np
import pytest

from skellysim_amd.fiber_fd import FiberFD
from skellysim_amd.system_fd import SystemFD
from oracle_backend import OracleBackend
from skellysim_amd.trajectory import TrajectoryWriter

REFERENCE = "/root/reference"


def _small_system():
    rng = np.random.default_rng(4)
    s = np.linspace(0, 1.0, 16)
    d = np.array([0.0, 0.0, 1.0])
    fibers = [FiberFD(np.array([0.1 * k, 0, 0])[None, :] + s[:, None] * d[None, :],
                      length=1.0, bending_rigidity=2.5e-3, eta=1.0)
              for k in range(2)]
    U = np.array([0.1, 0.0, 0.0])
    return SystemFD(fibers, eta=1.0, dt=0.1, backend=OracleBackend(),
                    background_flow=lambda r: np.tile(U, (len(r), 1)))


def _write_traj(dirpath, steps=2):
    sys_ = _small_system()
    path = os.path.join(dirpath, "skelly_sim.out")
    times = []
    with TrajectoryWriter(path) as tw:
        t = 0.0
        for _ in range(steps):
            info = sys_.step(tol=1e-11, maxiter=200)
            assert info["converged"]
            t += sys_.dt
            times.append(t)
            tw.write_frame(sys_, t, sys_.dt)
    return path, times, sys_


def test_structure_matches_serialization_contract(tmp_path):
    path, times, sys_ = _write_traj(str(tmp_path))
    with open(path, "rb") as fh:
        unp = msgpack.Unpacker(fh, raw=False)
        header = next(unp)
        assert header["trajversion"] == 1
        assert header["fiber_type"] == 1
        frames = list(unp)
    assert len(frames) == len(times)
    fr = frames[-1]
    assert set(fr.keys()) == {"time", "dt", "rng_state", "fibers", "bodies", "shell"}
    assert fr["time"] == pytest.approx(times[-1])
    ftype, fibs = fr["fibers"]
    assert ftype == 1 and len(fibs) == 2
    f0 = fibs[0]
    assert f0["n_nodes_"] == 16
    x = f0["x_"]
    assert x[0] == "__eigen__" and x[1] == 3 and x[2] == 16
    got = np.array(x[3:]).reshape(16, 3)
    assert np.allclose(got, sys_.fibers[0].x.T)
    assert fr["bodies"] == [[], [], []]


def _body_system():
    from skellysim_amd.body import SphericalBody
    fx = np.load(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                              "golden", "periphery_sphere_192.npz"))
    R = float(fx["radius"])
    b = SphericalBody(fx["nodes"], -fx["normals"],
                      fx["quadrature_weights"].reshape(-1), R,
                      position=(0.2, -0.1, 0.3),
                      external_force=(0.1, 0.0, -0.2),
                      external_torque=(0.0, 0.3, 0.0))
    return SystemFD([], eta=1.0, dt=0.1, bodies=[b],
                    backend=OracleBackend()), b


def test_body_frame_structure(tmp_path):
    """Spherical bodies serialize per body_spherical.hpp:77 into the first
    sublist of the [spherical, deformable, ellipsoidal] bodies array."""
    sys_, b = _body_system()
    path = str(tmp_path / "skelly_sim.out")
    with TrajectoryWriter(path) as tw:
        assert sys_.step(tol=1e-11, maxiter=100)["converged"]
        tw.write_frame(sys_, 0.1, 0.1)
    with open(path, "rb") as fh:
        unp = msgpack.Unpacker(fh, raw=False)
        next(unp)  # header
        fr = next(unp)
    sph, deform, ellip = fr["bodies"]
    assert deform == [] and ellip == [] and len(sph) == 1
    bm = sph[0]
    assert set(bm.keys()) == {"radius_", "position_", "orientation_",
                              "solution_vec_"}
    assert bm["radius_"] == b.radius
    q = bm["orientation_"]
    assert q[0] == "__quat__" and len(q) == 5
    assert np.allclose(q[1:], b.orientation)
    pos = bm["position_"]
    assert pos[0] == "__eigen__" and np.allclose(pos[3:], b.position)
    sol = bm["solution_vec_"]
    assert sol[1] == b.solution_size and sol[2] == 1
    assert np.allclose(sol[3:], b.solution_vec)


@pytest.mark.timeout(600)
def test_resume_matches_uninterrupted_run(tmp_path):
    """resume_from_trajectory (system.cpp:223-228): a run interrupted after
    2 steps and resumed into a FRESH system continues to the same state as
    an uninterrupted 4-step run (solves are deterministic without dynamic
    instability), and the appended trajectory has continuous times with a
    single header."""
    from skellysim_amd.trajectory import resume_from_trajectory
    from skellysim_amd.listener import Trajectory

    def fresh():
        rng = np.random.default_rng(9)
        s = np.linspace(0, 1.0, 16)
        x = np.stack([0.1 * np.sin(2 * np.pi * s), np.zeros_like(s), s],
                     axis=1)
        fib = FiberFD(x, length=1.0, bending_rigidity=2.5e-2, eta=1.0,
                      force_scale=-0.02)
        return SystemFD([fib], eta=1.0, dt=0.1, backend=OracleBackend(),
                        background_flow=lambda r: np.tile([0.05, 0, 0],
                                                          (len(r), 1)))

    # uninterrupted 4 steps
    sA = fresh()
    sA.run(t_final=0.4, adaptive=False, tol=1e-12)

    # 2 steps, write, resume into a FRESH system, 2 more
    path = str(tmp_path / "skelly_sim.out")
    sB = fresh()
    with TrajectoryWriter(path) as tw:
        sB.run(t_final=0.2, adaptive=False, tol=1e-12,
               on_accept=lambda s_, t: tw.write_frame(s_, t, s_.dt))
    sC = fresh()
    n = resume_from_trajectory(sC, path)
    assert n == 2 and sC.time == pytest.approx(0.2)
    with TrajectoryWriter(path, append=True) as tw:
        sC.run(t_final=0.4, adaptive=False, tol=1e-12,
               on_accept=lambda s_, t: tw.write_frame(s_, t, s_.dt))

    assert np.allclose(sC.fibers[0].x, sA.fibers[0].x, atol=1e-12)
    assert np.allclose(sC.fibers[0].tension, sA.fibers[0].tension, atol=1e-12)

    traj = Trajectory(path)
    assert traj.header["trajversion"] == 1
    times = [f["time"] for f in traj.frames]
    assert times == pytest.approx([0.1, 0.2, 0.3, 0.4])


@pytest.mark.skipif(not os.path.isdir(REFERENCE), reason="reference tree absent")
def test_reference_reader_round_trip(tmp_path):
    """The REFERENCE'S OWN reader.py must load our trajectory: times indexed,
    frames decoded, fiber positions identical."""
    # shims for the reference python package's optional deps
    for name, mod in {
        "toml": dict(load=lambda f: {}, dump=lambda *a, **k: None),
        "dataclass_utils": dict(check_type=lambda *a, **k: None),
        "numba": dict(njit=lambda *a, **k: (a[0] if a and callable(a[0])
                                            else (lambda f: f)), prange=range),
    }.items():
        m = types.ModuleType(name)
        for k, v in mod.items():
            setattr(m, k, v)
        sys.modules.setdefault(name, m)
    if "nptyping" not in sys.modules:
        m = types.ModuleType("nptyping")

        class _Sub:
            def __class_getitem__(cls, item):
                return np.ndarray

        m.NDArray = _Sub
        m.Shape = _Sub
        m.Float64 = float
        sys.modules["nptyping"] = m
    fg = types.ModuleType("function_generator")
    fg.FunctionGenerator = type("FunctionGenerator", (), {"__init__": lambda s, *a, **k: None})
    sys.modules.setdefault("function_generator", fg)

    sys.path.insert(0, os.path.join(REFERENCE, "src"))
    try:
        from skelly_sim.reader import TrajectoryReader
    except Exception as e:
        pytest.skip(f"reference reader not importable here: {e}")

    path, times, sys_ = _write_traj(str(tmp_path))
    toml_file = os.path.join(str(tmp_path), "skelly_config.toml")
    open(toml_file, "w").write("")

    traj = TrajectoryReader(toml_file)
    assert traj.trajectory_version == 1
    assert list(traj.times) == pytest.approx(times)
    traj.load_frame(len(times) - 1)
    fibs = traj["fibers"]
    assert len(fibs) == 2
    assert np.allclose(fibs[0]["x_"], sys_.fibers[0].x.T)
    assert traj["time"] == pytest.approx(times[-1])

    # bodies through the reference reader (flattened sublists,
    # reader.py:333-341; quaternion decoded as [w, x, y, z] array)
    os.remove(os.path.join(str(tmp_path), "skelly_sim.out"))
    sys_b, b = _body_system()
    with TrajectoryWriter(os.path.join(str(tmp_path), "skelly_sim.out")) as tw:
        assert sys_b.step(tol=1e-11, maxiter=100)["converged"]
        tw.write_frame(sys_b, 0.1, 0.1)
    traj2 = TrajectoryReader(toml_file)
    traj2.load_frame(0)
    bodies = traj2["bodies"]
    assert len(bodies) == 1
    bm = bodies[0]
    assert bm["radius_"] == b.radius
    assert np.allclose(bm["position_"], b.position)
    assert np.allclose(bm["orientation_"], b.orientation)
    assert np.allclose(bm["solution_vec_"], b.solution_vec)


def test_shell_plus_body_frame_slices(tmp_path):
    """Regression: with BOTH a shell and bodies, the frame's shell
    solution_vec_ is exactly the shell block of the [fibers|shell|bodies]
    solution layout (not the tail including bodies)."""
    from skellysim_amd.body import SphericalBody
    from skellysim_amd.system_fd import Shell
    fx = np.load(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                              "golden", "periphery_sphere_192.npz"))
    shell = Shell(fx["nodes"], fx["normals"],
                  fx["stresslet_plus_complementary"], fx["M_inv"])
    R = float(fx["radius"])
    b = SphericalBody(fx["nodes"] * 0.2, -fx["normals"],
                      fx["quadrature_weights"].reshape(-1) * 0.04, 0.2 * R,
                      position=(0.2, 0.0, 0.0),
                      external_force=(0.05, 0.0, 0.0))
    sys_ = SystemFD([], eta=1.0, dt=0.05, shell=shell, bodies=[b],
                    backend=OracleBackend())
    path = str(tmp_path / "skelly_sim.out")
    with TrajectoryWriter(path) as tw:
        assert sys_.step(tol=1e-10, maxiter=200, restart=100)["converged"]
        tw.write_frame(sys_, 0.05, 0.05)
    with open(path, "rb") as fh:
        unp = msgpack.Unpacker(fh, raw=False)
        next(unp)
        fr = next(unp)
    sh = fr["shell"]["solution_vec_"]
    assert sh[1] == sys_.shell_sol_size and sh[2] == 1
    a = sys_.fiber_sol_size
    assert np.allclose(sh[3:], sys_.solution[a: a + sys_.shell_sol_size])
    bm = fr["bodies"][0][0]
    bsol = bm["solution_vec_"]
    assert bsol[0] == "__eigen__"
    assert np.allclose(bsol[3:], sys_.solution[a + sys_.shell_sol_size:])


def test_resume_with_bodies_matches_uninterrupted(tmp_path):
    """resume_from_trajectory restores body position, orientation and
    solution (body_spherical.hpp:77 fields): a 2-step + resume + 2-step run
    of a forced, torqued body lands bit-close to the uninterrupted 4-step
    run."""
    from skellysim_amd.trajectory import resume_from_trajectory

    sA, _ = _body_system()
    sA.run(t_final=0.4, adaptive=False, tol=1e-12)

    path = str(tmp_path / "skelly_sim.out")
    sB, _ = _body_system()
    with TrajectoryWriter(path) as tw:
        sB.run(t_final=0.2, adaptive=False, tol=1e-12,
               on_accept=lambda s_, t: tw.write_frame(s_, t, s_.dt))
    sC, _ = _body_system()
    n = resume_from_trajectory(sC, path)
    assert n == 2 and sC.time == pytest.approx(0.2)
    sC.run(t_final=0.4, adaptive=False, tol=1e-12)

    bA, bC = sA.bodies[0], sC.bodies[0]
    assert np.allclose(bC.position, bA.position, atol=1e-10)
    qA = np.array(bA.orientation, float)
    qC = np.array(bC.orientation, float)
    # q and -q are the same rotation
    assert min(np.abs(qC - qA).max(), np.abs(qC + qA).max()) < 1e-10
