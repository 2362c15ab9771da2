"""GPU test: the reference-authored config-4 file drives a full solve
(config loading -> on-GPU shell assembly -> device-resident GMRES)."""

import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

HERE = os.path.dirname(os.path.abspath(__file__))


def test_config4_from_reference_config(hip_lib_path):
    from skellysim_amd.config import load_config, build_system
    from skellysim_amd.system_fd import HipBackend

    cfg = load_config(os.path.join(HERE, "golden", "skelly_config_ellipsoid.toml"))
    sys_ = build_system(cfg, backend=HipBackend(),
                        shell_geometry=os.path.join(HERE, "golden",
                                                    "ellipsoid_8192_nodes.npz"),
                        dt=0.025)
    assert len(sys_.fibers) == 512 and sys_.shell.n_nodes == 8192
    assert sys_.eta == cfg["params"]["eta"]
    info = sys_.step(tol=1e-10, maxiter=300, restart=150)
    assert info["converged"], info
    xs = np.concatenate([f.x.reshape(-1) for f in sys_.fibers])
    assert np.isfinite(xs).all()
    # fibers stay inside the periphery after the step
    g = np.load(os.path.join(HERE, "golden", "ellipsoid_8192_nodes.npz"))
    a, b, c = float(g["a"]), float(g["b"]), float(g["c"])
    for f in sys_.fibers[::32]:
        lvl = (f.x[0] / a) ** 2 + (f.x[1] / b) ** 2 + (f.x[2] / c) ** 2
        assert np.all(lvl < 1.05)
