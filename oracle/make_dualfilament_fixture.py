"""Generate the dual-filament initial positions with the reference's OWN
perturbed_fiber_positions (src/skelly_sim/skelly_config.py:130-168), as the
reference's tests/combined/test_fiber_dualfilament.py does: fiber 0
cos-perturbed (amplitude 0.01, ortho +x), fiber 1 straight at x=1.
Build-container only (imports /root/reference with dependency shims)."""

import os
import sys
import types

import numpy as np


def install_config_shims():
    for name, mod in {
        "toml": dict(load=lambda f: {}, dump=lambda *a, **k: None,
                     dumps=lambda *a, **k: ""),
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
    fg.FunctionGenerator = type("FunctionGenerator", (),
                                {"__init__": lambda s, *a, **k: None})
    sys.modules.setdefault("function_generator", fg)


def main():
    install_config_shims()
    sys.path.insert(0, "/root/reference/src")
    from skelly_sim.skelly_config import perturbed_fiber_positions

    length, n_nodes = 2.0, 64
    x0 = perturbed_fiber_positions(0.01, length, np.array([0.0, 0.0, 0.0]),
                                   np.array([0.0, 0.0, 1.0]), n_nodes,
                                   np.array([1.0, 0.0, 0.0]))
    s = np.linspace(0, length, n_nodes)
    x1 = np.stack([np.full(n_nodes, 1.0), np.zeros(n_nodes), s], axis=1)
    # fill_node_positions(base=[1,0,0], normal=[0,0,1]) is the straight line
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = os.path.join(repo, "tests", "golden", "dualfilament_init.npz")
    np.savez(out, x0=x0, x1=x1)
    print("wrote", out, "fiber0 head", x0[0], "tail", x0[-1])


if __name__ == "__main__":
    main()
