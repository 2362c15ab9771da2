"""CPU tests: the reference-generator-authored config fixture loads and
builds the config-4 system (fiber geometry consistent with the committed
ellipsoid periphery geometry)."""

import os

import numpy as np

from skellysim_amd.config import load_config, build_fibers, periphery_interaction_from

HERE = os.path.dirname(os.path.abspath(__file__))
CFG = os.path.join(HERE, "golden", "skelly_config_ellipsoid.toml")


def test_load_reference_config():
    cfg = load_config(CFG)
    assert cfg["params"]["eta"] == 1.0
    assert cfg["params"]["fiber_type"] == "FiniteDifference"
    assert cfg["periphery"]["shape"] == "ellipsoid"
    assert len(cfg["fibers"]) == 512


def test_build_fibers_geometry():
    cfg = load_config(CFG)
    fibers = build_fibers(cfg, eta=cfg["params"]["eta"])
    assert len(fibers) == 512
    assert all(f.n_nodes == 64 and f.minus_clamped for f in fibers)
    a, b, c = (cfg["periphery"][k] for k in "abc")
    for f in fibers[::64]:
        # minus end sits just inside the attachment ellipsoid (a, b, c);
        # the periphery node surface is a further x1.04 out (precompute.py:34)
        p = f.x[:, 0]
        lvl = (p[0] / a) ** 2 + (p[1] / b) ** 2 + (p[2] / c) ** 2
        assert 0.8 < lvl < 1.0, lvl
        # fiber points inward: second node strictly inside the first
        q = f.x[:, 1]
        lvl2 = (q[0] / a) ** 2 + (q[1] / b) ** 2 + (q[2] / c) ** 2
        assert lvl2 < lvl
        # fiber arclength == declared length
        seg = np.diff(f.x.T, axis=0)
        assert abs(np.linalg.norm(seg, axis=1).sum() - f.length) < 1e-2


def test_fibers_inside_committed_periphery_geometry():
    """The committed 8192-node geometry fixture (nodes at x1.04 the
    attachment surface) encloses every fiber node."""
    cfg = load_config(CFG)
    g = np.load(os.path.join(HERE, "golden", "ellipsoid_8192_nodes.npz"))
    a, b, c = float(g["a"]), float(g["b"]), float(g["c"])
    fibers = build_fibers(cfg, eta=1.0)
    for f in fibers:
        lvl = ((f.x[0] / a) ** 2 + (f.x[1] / b) ** 2 + (f.x[2] / c) ** 2)
        assert np.all(lvl < 1.0)


def test_run_sim_cli_end_to_end(tmp_path, monkeypatch):
    """tools/run_sim.py plumbing on CPU: tiny hand-written reference-format
    config -> run loop -> reference-format trajectory with the configured
    dt_write cadence. The CLI's HipBackend is patched to the oracle backend
    (the CLI itself stays HIP-only)."""
    import sys
    import importlib
    from oracle_backend import OracleBackend

    s = np.linspace(0, 1.0, 16)
    pts = np.stack([0.1 * np.sin(2 * np.pi * s), np.zeros_like(s), s], axis=1)
    flat = ", ".join(repr(float(v)) for v in pts.reshape(-1))
    cfg_path = tmp_path / "tiny.toml"
    cfg_path.write_text(f"""
[params]
eta = 1.0
dt_initial = 0.1
dt_write = 0.1
t_final = 0.25
gmres_tol = 1e-11
adaptive_timestep_flag = false
fiber_type = "FiniteDifference"

[[fibers]]
length = 1.0
bending_rigidity = 2.5e-2
radius = 0.0125
force_scale = -0.02
minus_clamped = false
n_nodes = 16
x = [{flat}]
""")
    out = tmp_path / "traj.out"
    sys.path.insert(0, os.path.join(os.path.dirname(HERE), "tools"))
    try:
        run_sim = importlib.import_module("run_sim")
    finally:
        sys.path.pop(0)
    monkeypatch.setattr(run_sim, "HipBackend", OracleBackend)
    monkeypatch.setattr(sys, "argv",
                        ["run_sim.py", "--config-file", str(cfg_path),
                         "--out", str(out)])
    run_sim.main()

    from skellysim_amd.listener import Trajectory
    traj = Trajectory(str(out))
    assert traj.header["trajversion"] == 1
    assert len(traj) >= 2                      # t=0.1 and t=0.2 writes
    times = [f["time"] for f in traj.frames]
    assert times == sorted(times)
    f0 = traj.frames[0]["fibers"][1][0]
    assert np.asarray(f0["x_"]).shape == (16, 3)
    assert np.isfinite(np.asarray(f0["x_"])).all()


def test_body_config_build_and_solve(tmp_path):
    """[[bodies]] tables (skelly_config.py:720-751) build spherical bodies;
    a fiber with parent_body/parent_site couples to it; the built system
    solves on the oracle backend."""
    import sys
    sys.path.insert(0, HERE)
    from oracle_backend import OracleBackend
    from skellysim_amd.config import load_config, build_bodies, build_system

    from oracle_backend import OracleBackend
    fx = np.load(os.path.join(HERE, "golden", "periphery_sphere_192.npz"))
    R = float(fx["radius"])
    geom = {"nodes": fx["nodes"], "normals": -fx["normals"],
            "weights": fx["quadrature_weights"].reshape(-1)}
    s = np.linspace(0, 1.0, 16)
    x0 = np.array([1.1 * R + 0.3, -0.2, 0.1])
    pts = (x0[None, :] + s[:, None] * np.array([1.0, 0, 0]))
    flat = ", ".join(repr(float(v)) for v in pts.reshape(-1))
    site = [1.1 * R, 0.0, 0.0]  # attachment radius above the surface
    cfg_path = tmp_path / "body.toml"
    cfg_path.write_text(f"""
[params]
eta = 1.0
dt_initial = 0.05
fiber_type = "FiniteDifference"

[[bodies]]
shape = "sphere"
radius = {R!r}
position = [0.3, -0.2, 0.1]
orientation = [0.0, 0.0, 0.0, 1.0]
nucleation_sites = {site!r}
external_force = [0.0, 0.0, 0.1]

[[fibers]]
length = 1.0
bending_rigidity = 2.5e-3
force_scale = -0.02
parent_body = 0
parent_site = 0
n_nodes = 16
x = [{flat}]
""")
    cfg = load_config(str(cfg_path))
    bodies = build_bodies(cfg, geom)
    assert len(bodies) == 1
    b = bodies[0]
    assert b.radius == R and np.allclose(b.position, [0.3, -0.2, 0.1])
    assert np.allclose(b.external_force, [0, 0, 0.1])
    assert b.nucleation_sites_ref.shape == (1, 3)

    sys_ = build_system(cfg, backend=OracleBackend(), body_geometry=geom)
    assert len(sys_.bodies) == 1 and len(sys_.fibers) == 1
    f = sys_.fibers[0]
    assert f.binding_site == (0, 0) and f.minus_clamped
    info = sys_.step(tol=1e-11, maxiter=300, restart=150)
    assert info["converged"], info
    assert np.linalg.norm(sys_.bodies[0].velocity) > 1e-5  # pushed by F_ext


def test_ellipsoid_body_config():
    """shape = "ellipsoid" builds an EllipsoidalBody with axis_length
    (skelly_config.py:735-736)."""
    from skellysim_amd.config import build_bodies
    from skellysim_amd.body import EllipsoidalBody
    g = np.load(os.path.join(HERE, "golden", "ellipsoid_body_nodes.npz"))
    geom = {"nodes": g["nodes"], "normals": g["normals"],
            "weights": g["quadrature_weights"].reshape(-1)}
    cfg = {"bodies": [dict(shape="ellipsoid",
                           axis_length=[float(g["a"]), float(g["b"]),
                                        float(g["c"])],
                           position=[0.5, 0.0, -0.2])]}
    (b,) = build_bodies(cfg, geom)
    assert isinstance(b, EllipsoidalBody)
    assert np.allclose(b.axis_length, [g["a"], g["b"], g["c"]])
    assert b.radius == float(g["a"])  # max axis as the scalar radius
    assert not b.check_collision(b)   # reference stubs ellipsoid collisions


def test_periphery_interaction_parsing():
    cfg = load_config(CFG)
    assert periphery_interaction_from(cfg) is None  # flag off in the example
    cfg["params"]["periphery_interaction_flag"] = True
    pi = periphery_interaction_from(cfg)
    assert pi["kind"] == "ellipsoid" and pi["abc"] == (7.8, 4.16, 4.16)
    assert pi["f_0"] == 20.0 and pi["l_0"] == 0.05


def test_run_sim_cli_dynamic_instability(tmp_path, monkeypatch):
    """End-to-end CLI with a [params.dynamic_instability] table and a body:
    the adaptive run nucleates/removes fibers mid-run and the trajectory
    records the changing population."""
    import sys
    import importlib
    sys.path.insert(0, HERE)
    from oracle_backend import OracleBackend

    fx = np.load(os.path.join(HERE, "golden", "body_sphere_600.npz"))
    geom_path = tmp_path / "body_geom.npz"
    np.savez(geom_path, nodes=fx["nodes"], normals=fx["normals"],
             weights=fx["quadrature_weights"].reshape(-1),
             nucleation_sites=fx["nucleation_sites"])
    sites_flat = ", ".join(repr(float(v))
                           for v in np.asarray(fx["nucleation_sites"])[:6].reshape(-1))
    cfg_path = tmp_path / "di.toml"
    cfg_path.write_text(f"""
[params]
eta = 1.0
dt_initial = 0.1
dt_write = 0.1
t_final = 0.4
gmres_tol = 1e-10
adaptive_timestep_flag = false
seed = 21
fiber_type = "FiniteDifference"

[params.dynamic_instability]
n_nodes = 8
v_growth = 0.2
f_catastrophe = 1.0
nucleation_rate = 20.0
min_length = 0.4

[[bodies]]
shape = "sphere"
radius = {float(fx["radius"])!r}
nucleation_sites = [{sites_flat}]
""")
    out = tmp_path / "traj.out"
    sys.path.insert(0, os.path.join(os.path.dirname(HERE), "tools"))
    try:
        run_sim = importlib.import_module("run_sim")
    finally:
        sys.path.pop(0)
    monkeypatch.setattr(run_sim, "HipBackend", OracleBackend)
    monkeypatch.setattr(sys, "argv",
                        ["run_sim.py", "--config-file", str(cfg_path),
                         "--body-geometry", str(geom_path),
                         "--out", str(out)])
    run_sim.main()

    from skellysim_amd.listener import Trajectory
    traj = Trajectory(str(out))
    counts = [len(f["fibers"][1]) for f in traj.frames]
    assert len(traj) >= 3
    assert max(counts) >= 1          # nucleation happened
    assert max(counts) <= 6          # bounded by the sites
    for f in traj.frames:            # bodies recorded every frame
        assert len(f["bodies"][0]) == 1


def test_reference_precompute_key_aliases():
    """Geometry loaders accept the reference precompute's own key names
    (node_positions_ref/node_normals_ref/node_weights for bodies,
    precompute.py:187; nodes/normals/quadrature_weights(+operators) for
    peripheries, precompute.py:141-148)."""
    from skellysim_amd.config import build_bodies, _geometry_fields
    from oracle_backend import OracleBackend
    fx = np.load(os.path.join(HERE, "golden", "periphery_sphere_192.npz"))
    R = float(fx["radius"])
    ref_style = {"node_positions_ref": fx["nodes"],
                 "node_normals_ref": -fx["normals"],
                 "node_weights": fx["quadrature_weights"]}
    nodes, normals, w = _geometry_fields(ref_style)
    assert nodes.shape == (192, 3) and w.shape == (192,)
    cfg = {"bodies": [dict(shape="sphere", radius=R)]}
    (b,) = build_bodies(cfg, ref_style)
    assert b.n_nodes == 192 and b.radius == R
    # engine-style keys still work
    eng_style = {"nodes": fx["nodes"], "normals": -fx["normals"],
                 "weights": fx["quadrature_weights"]}
    (b2,) = build_bodies(cfg, eng_style)
    assert np.allclose(b2.nodes, b.nodes)


def test_shell_precompute_operator_adoption():
    """A full reference precompute npz (with stresslet_plus_complementary
    and M_inv) is adopted directly as the shell operators instead of
    re-assembling (precompute.py:141-148 file layout)."""
    import torch
    from skellysim_amd.config import build_system
    from oracle_backend import OracleBackend
    fx = np.load(os.path.join(HERE, "golden", "periphery_sphere_192.npz"))
    cfg = {"params": {"eta": 1.0}, "periphery": {"shape": "sphere",
                                                 "radius": float(fx["radius"])},
           "fibers": []}
    sys_ = build_system(cfg, backend=OracleBackend(),
                        shell_geometry=dict(fx), device="cpu")
    assert sys_.shell is not None
    assert torch.is_tensor(sys_.shell.A)
    assert np.allclose(sys_.shell.A.numpy(),
                       fx["stresslet_plus_complementary"])
    assert np.allclose(sys_.shell.M_inv.numpy(), fx["M_inv"])


def test_run_sim_cli_resume(tmp_path, monkeypatch):
    """run_sim --resume (the reference's skelly_sim --resume): a run to
    t_final=0.2 followed by a --resume run with t_final=0.4 yields the same
    trajectory as one uninterrupted run to 0.4 — continuous times, one
    header, matching final fiber state."""
    import sys
    import importlib
    from oracle_backend import OracleBackend

    s = np.linspace(0, 1.0, 16)
    pts = np.stack([0.1 * np.sin(2 * np.pi * s), np.zeros_like(s), s], axis=1)
    flat = ", ".join(repr(float(v)) for v in pts.reshape(-1))

    def write_cfg(path, t_final):
        path.write_text(f"""
[params]
eta = 1.0
dt_initial = 0.1
dt_write = 0.1
t_final = {t_final}
gmres_tol = 1e-11
adaptive_timestep_flag = false
fiber_type = "FiniteDifference"

[[fibers]]
length = 1.0
bending_rigidity = 2.5e-2
radius = 0.0125
force_scale = -0.02
minus_clamped = false
n_nodes = 16
x = [{flat}]
""")

    sys.path.insert(0, os.path.join(os.path.dirname(HERE), "tools"))
    try:
        run_sim = importlib.import_module("run_sim")
    finally:
        sys.path.pop(0)
    monkeypatch.setattr(run_sim, "HipBackend", OracleBackend)

    def run(argv):
        monkeypatch.setattr(sys, "argv", ["run_sim.py"] + argv)
        run_sim.main()

    # uninterrupted
    cfgA = tmp_path / "a.toml"
    write_cfg(cfgA, 0.4)
    outA = tmp_path / "a.out"
    run(["--config-file", str(cfgA), "--out", str(outA)])

    # interrupted + resumed
    cfgB1 = tmp_path / "b1.toml"
    write_cfg(cfgB1, 0.2)
    outB = tmp_path / "b.out"
    run(["--config-file", str(cfgB1), "--out", str(outB)])
    cfgB2 = tmp_path / "b2.toml"
    write_cfg(cfgB2, 0.4)
    run(["--config-file", str(cfgB2), "--out", str(outB), "--resume"])

    from skellysim_amd.listener import Trajectory
    trA, trB = Trajectory(str(outA)), Trajectory(str(outB))
    tA = [f["time"] for f in trA.frames]
    tB = [f["time"] for f in trB.frames]
    assert np.allclose(tB, tA, atol=1e-12)
    xA = np.asarray(trA.frames[-1]["fibers"][1][0]["x_"], float)
    xB = np.asarray(trB.frames[-1]["fibers"][1][0]["x_"], float)
    assert np.allclose(xB, xA, atol=1e-12)


def test_precompute_file_auto_pickup(tmp_path):
    """build_system honors the config's own precompute_file paths
    (params.hpp:61 / body_ellipsoidal.cpp:220-221) when no explicit
    geometry is supplied: the periphery operators come from the named npz
    (adopted directly when it holds the prebuilt matrices) and each body's
    surface from its table's file."""
    import shutil
    import torch
    from oracle_backend import OracleBackend
    from skellysim_amd.config import build_system

    src = os.path.join(HERE, "golden", "periphery_sphere_192.npz")
    shutil.copy(src, tmp_path / "periphery_precompute.npz")
    shutil.copy(src, tmp_path / "body_precompute.npz")
    fx = np.load(src)
    cfg = {"params": {"eta": 1.0},
           "periphery": {"shape": "sphere", "radius": float(fx["radius"]),
                         "precompute_file": "periphery_precompute.npz"},
           "bodies": [{"radius": float(fx["radius"]),
                       "precompute_file": "body_precompute.npz"}],
           "fibers": []}
    sys_ = build_system(cfg, backend=OracleBackend(), device="cpu",
                        config_dir=str(tmp_path))
    assert sys_.shell is not None
    assert np.allclose(sys_.shell.A.numpy(),
                       fx["stresslet_plus_complementary"])
    assert len(sys_.bodies) == 1
    assert sys_.bodies[0].n_nodes == len(fx["nodes"])


def test_skelly_sim_hip_run_dispatch(tmp_path, monkeypatch):
    """tools/skelly_sim_hip without --listen runs the simulation with the
    reference CLI surface (skelly_sim.cpp:27-30): --config-file=X,
    --overwrite, --no-* forms ignored, trajectory at ./skelly_sim.out."""
    import importlib.machinery
    import importlib.util
    import sys as _sys
    from oracle_backend import OracleBackend

    s = np.linspace(0, 1.0, 16)
    pts = np.stack([0.1 * np.sin(2 * np.pi * s), np.zeros_like(s), s], axis=1)
    flat = ", ".join(repr(float(v)) for v in pts.reshape(-1))
    (tmp_path / "skelly_config.toml").write_text(f"""
[params]
eta = 1.0
dt_initial = 0.1
dt_write = 0.1
t_final = 0.2
gmres_tol = 1e-11
adaptive_timestep_flag = false
fiber_type = "FiniteDifference"

[[fibers]]
length = 1.0
bending_rigidity = 2.5e-2
radius = 0.0125
force_scale = -0.02
minus_clamped = false
n_nodes = 16
x = [{flat}]
""")
    loader = importlib.machinery.SourceFileLoader(
        "skelly_sim_hip_bin",
        os.path.join(os.path.dirname(HERE), "tools", "skelly_sim_hip"))
    spec = importlib.util.spec_from_loader(loader.name, loader)
    mod = importlib.util.module_from_spec(spec)
    loader.exec_module(mod)
    import run_sim
    monkeypatch.setattr(run_sim, "HipBackend", OracleBackend)
    monkeypatch.chdir(tmp_path)
    mod.main(["--config-file=skelly_config.toml", "--overwrite",
              "--no-resume"])

    from skellysim_amd.listener import Trajectory
    traj = Trajectory(str(tmp_path / "skelly_sim.out"))
    assert np.allclose([f["time"] for f in traj.frames], [0.1, 0.2],
                       atol=1e-12)
