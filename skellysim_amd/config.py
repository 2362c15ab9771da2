"""Reference TOML config consumption: reads the config files SkellySim's own
generator (src/skelly_sim/skelly_config.py) produces — params, fibers (with
the flat col-major x arrays), periphery — and builds a SystemFD (SURVEY.md §5
"New build keeps the same TOML keys it consumes so reference configs run
unchanged").

The reference feeds the periphery matrices from its precompute .npz
(precompute_file); this engine assembles those operators ON DEVICE
(periphery_precompute.assemble_shell_operator) from the geometry
(nodes/normals/weights), which build_system takes as an npz path or dict."""

import os

import numpy as np
import tomli

from .fiber_fd import FiberFD
from .system_fd import SystemFD, Shell


def load_config(path):
    with open(path, "rb") as f:
        return tomli.load(f)


def build_fibers(cfg, eta):
    fibers = []
    for ft in cfg.get("fibers", []):
        x = np.asarray(ft["x"], float).reshape(-1, 3)  # flat col-major 3 x n
        f = FiberFD(
            x, length=ft["length"], bending_rigidity=ft["bending_rigidity"],
            eta=eta, radius=ft.get("radius", 0.0125),
            force_scale=ft.get("force_scale", 0.0),
            minus_clamped=bool(ft.get("minus_clamped", False)))
        # body attachment (fiber_finite_difference.cpp:44-45); an attached
        # fiber's minus end is clamped to the body regardless of the flag
        # (skelly_config.py:277)
        f.binding_site = (int(ft.get("parent_body", -1)),
                          int(ft.get("parent_site", -1)))
        if f.binding_site[0] >= 0:
            from .fiber_fd import BC_VELOCITY, BC_ANGULAR_VELOCITY
            f.minus_clamped = True
            f.bc_minus = (BC_VELOCITY, BC_ANGULAR_VELOCITY)
        fibers.append(f)
    return fibers


def periphery_shape_from(cfg):
    """dict(kind, radius|abc) for collision/binding geometry."""
    per = cfg.get("periphery", {})
    if per.get("shape") == "sphere":
        return dict(kind="sphere", radius=per["radius"])
    if per.get("shape") == "ellipsoid":
        return dict(kind="ellipsoid", abc=(per["a"], per["b"], per["c"]))
    return None


def periphery_interaction_from(cfg):
    """fiber-periphery steric params (params.cpp:18,75-78; defaults
    f_0=20, l_0=0.05, params.hpp:46-47) when the flag is on."""
    p = cfg.get("params", {})
    if not p.get("periphery_interaction_flag", False):
        return None
    per = cfg.get("periphery", {})
    fp = p.get("fiber_periphery_interaction", {})
    kw = dict(f_0=fp.get("f_0", 20.0), l_0=fp.get("l_0", 0.05))
    if per.get("shape") == "sphere":
        return dict(kind="sphere", radius=per["radius"], **kw)
    if per.get("shape") == "ellipsoid":
        return dict(kind="ellipsoid", abc=(per["a"], per["b"], per["c"]), **kw)
    return None


def _geometry_fields(g):
    """Accept both this engine's key names (nodes/normals/weights|
    quadrature_weights) and the reference precompute's
    (node_positions_ref/node_normals_ref/node_weights,
    precompute.py:187,245 / body_spherical.cpp:199-203)."""
    def pick(*names):
        for k in names:
            if k in g:
                return np.asarray(g[k], float)
        raise KeyError(f"none of {names} in geometry ({list(g.keys())})")
    nodes = pick("nodes", "node_positions_ref")
    normals = pick("normals", "node_normals_ref")
    weights = pick("weights", "quadrature_weights", "node_weights")
    return nodes, normals, weights.reshape(-1)


def build_bodies(cfg, body_geometry):
    """Spherical bodies from reference [[bodies]] tables
    (skelly_config.py:720-751 / body_spherical.cpp:213-275). body_geometry:
    an npz path or dict (nodes/normals/weights — the body precompute
    surface) shared by all bodies, or a list with one entry per body. TOML
    orientation is Eigen coeffs order [x, y, z, w]
    (parse_util convert_array<Quaterniond>); nucleation_sites is the flat
    col-major 3 x m array."""
    from .body import SphericalBody
    tables = cfg.get("bodies", [])
    if not tables:
        return []
    geoms = body_geometry if isinstance(body_geometry, (list, tuple)) \
        else [body_geometry] * len(tables)
    bodies = []
    for bt, g in zip(tables, geoms):
        g = dict(np.load(g)) if isinstance(g, str) else g
        q = bt.get("orientation", [0.0, 0.0, 0.0, 1.0])
        sites = np.asarray(bt.get("nucleation_sites", []),
                           float).reshape(-1, 3)
        shape = bt.get("shape", "sphere")
        kw = dict(position=bt.get("position", [0.0, 0.0, 0.0]),
                  orientation=(q[3], q[0], q[1], q[2]),
                  nucleation_sites_ref=sites if len(sites) else None,
                  external_force=bt.get("external_force", [0.0, 0.0, 0.0]),
                  external_torque=bt.get("external_torque", [0.0, 0.0, 0.0]),
                  external_force_type=bt.get("external_force_type", "Linear"),
                  oscillation_amplitude=bt.get(
                      "external_oscillation_force_amplitude", 0.0),
                  # omega = 2 pi * frequency (body_spherical.cpp:252-253)
                  oscillation_omega=2.0 * np.pi * bt.get(
                      "external_oscillation_force_frequency", 0.0),
                  oscillation_phase=bt.get(
                      "external_oscillation_force_phase", 0.0))
        nodes, normals, w = _geometry_fields(g)
        if shape == "sphere":
            bodies.append(SphericalBody(nodes, normals, w,
                                        bt.get("radius", 1.0), **kw))
        elif shape == "ellipsoid":
            from .body import EllipsoidalBody
            bodies.append(EllipsoidalBody(
                nodes, normals, w,
                bt.get("axis_length", [1.0, 1.0, 1.0]), **kw))
        else:
            raise NotImplementedError(f"body shape {shape!r}")
    return bodies


def _resolve_precompute(table, config_dir):
    """The reference loads the periphery/body surface data from the
    precompute_file named in the TOML table (params.hpp:61 /
    body_ellipsoidal.cpp:220-221, defaults periphery_precompute.npz /
    body_precompute.npz, skelly_config.py:446,762). Returns a path if the
    table names one that exists (cwd-relative like the reference, else
    config-dir-relative), None otherwise."""
    pf = table.get("precompute_file")
    if not pf:
        return None
    if os.path.exists(pf):
        return pf
    if config_dir is not None:
        cand = os.path.join(config_dir, pf)
        if os.path.exists(cand):
            return cand
    return None


def build_system(cfg, backend=None, shell_geometry=None, dt=None,
                 background_flow=None, body_geometry=None, device=None,
                 config_dir=None):
    """SystemFD from a reference config. shell_geometry: npz path or dict
    with nodes/normals/quadrature_weights (the periphery geometry the
    reference's precompute generates; operators are assembled on device).
    When shell_geometry/body_geometry are not supplied, the precompute_file
    paths named in the config are picked up when they exist on disk (the
    reference's own loading convention)."""
    params = cfg.get("params", {})
    eta = params.get("eta", 1.0)
    dt = dt if dt is not None else params.get("dt_initial", 0.025)
    fibers = build_fibers(cfg, eta)

    if shell_geometry is None and "periphery" in cfg:
        shell_geometry = _resolve_precompute(cfg["periphery"], config_dir)
    if body_geometry is None and cfg.get("bodies"):
        paths = [_resolve_precompute(bt, config_dir) for bt in cfg["bodies"]]
        if all(p is not None for p in paths):
            body_geometry = paths

    shell = None
    if "periphery" in cfg and shell_geometry is not None:
        import torch
        from .periphery_precompute import assemble_shell_operator

        g = np.load(shell_geometry) if isinstance(shell_geometry, str) \
            else shell_geometry
        dev = torch.device(device if device is not None else "cuda:0")
        nodes, normals, weights = _geometry_fields(g)
        if "stresslet_plus_complementary" in g and "M_inv" in g:
            # a full reference precompute file (precompute.py:141-148):
            # adopt its prebuilt dense operators directly
            A = torch.from_numpy(
                np.asarray(g["stresslet_plus_complementary"])).to(dev)
            M_inv = torch.from_numpy(np.asarray(g["M_inv"])).to(dev)
        else:
            A, M_inv = assemble_shell_operator(
                torch.from_numpy(nodes).to(dev),
                torch.from_numpy(normals).to(dev),
                torch.from_numpy(weights).to(dev))
        shell = Shell(nodes, normals, A, M_inv)

    bodies = build_bodies(cfg, body_geometry) if body_geometry is not None \
        else []
    di = params.get("dynamic_instability")
    if di is not None and di.get("n_nodes", 0) == 0:
        di = None
    system = SystemFD(fibers, eta=eta, dt=dt, shell=shell,
                      background_flow=background_flow, backend=backend,
                      periphery_interaction=periphery_interaction_from(cfg),
                      bodies=bodies,
                      periphery_shape=periphery_shape_from(cfg),
                      periphery_binding=params.get("periphery_binding"),
                      dynamic_instability=di,
                      seed=params.get("seed", 130319))
    system.motor_activation_delay = params.get(
        "implicit_motor_activation_delay", 0.0)
    # point/background sources (system.cpp:691-695)
    from .sources import PointSourceContainer, BackgroundSource
    if cfg.get("point_sources"):
        system.point_sources = PointSourceContainer.from_config(
            cfg["point_sources"])
    if "background" in cfg:
        system.background_source = BackgroundSource.from_config(
            cfg["background"])
    return system
