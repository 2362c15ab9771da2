"""Generate the BASELINE config-4 TOML with the reference's OWN generator
(examples/ellipsoid/gen_config.py logic at BASELINE sizes: 512 fibers x 64
nodes, 8192-node ellipsoidal periphery) and commit it as a fixture so the
engine can be driven by a reference-authored config file. Build-container
only (imports the reference package under shims; the reference's toml.dump
is satisfied by the minimal writer below, validated by re-parsing with
tomli)."""

import os
import sys
import types

import numpy as np


def toml_escape(s):
    return '"' + s.replace("\\", "\\\\").replace('"', '\\"') + '"'


def fmt_value(v):
    if isinstance(v, bool):
        return "true" if v else "false"
    if isinstance(v, (int, np.integer)):
        return str(int(v))
    if isinstance(v, (float, np.floating)):
        return repr(float(v))
    if isinstance(v, str):
        return toml_escape(v)
    if isinstance(v, (list, tuple, np.ndarray)):
        return "[" + ", ".join(fmt_value(x) for x in v) + "]"
    raise TypeError(f"unsupported TOML value {type(v)}")


def dump_table(d, prefix, out):
    scalars = {k: v for k, v in d.items()
               if v is not None and not isinstance(v, dict)
               and not (isinstance(v, list) and v and isinstance(v[0], dict))}
    subtables = {k: v for k, v in d.items() if isinstance(v, dict)}
    table_arrays = {k: v for k, v in d.items()
                    if isinstance(v, list) and v and isinstance(v[0], dict)}
    for k, v in scalars.items():
        out.append(f"{k} = {fmt_value(v)}")
    for k, v in subtables.items():
        name = f"{prefix}{k}"
        out.append(f"\n[{name}]")
        dump_table(v, name + ".", out)
    for k, lst in table_arrays.items():
        name = f"{prefix}{k}"
        for item in lst:
            out.append(f"\n[[{name}]]")
            dump_table(item, name + ".", out)


def toml_dumps(d):
    out = []
    dump_table(d, "", out)
    return "\n".join(out) + "\n"


def install_shims():
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from make_periphery_fixture import install_shims as base
    base()
    toml = types.ModuleType("toml")
    import tomli

    toml.load = lambda f: tomli.loads(f.read() if hasattr(f, "read") else open(f).read())
    toml.dump = lambda d, f: f.write(toml_dumps(d))
    toml.dumps = toml_dumps
    sys.modules.setdefault("toml", toml)
    du = types.ModuleType("dataclass_utils")
    du.check_type = lambda *a, **k: None
    sys.modules.setdefault("dataclass_utils", du)
    npt = types.ModuleType("nptyping")

    class _Sub:
        def __class_getitem__(cls, item):
            return np.ndarray

    npt.NDArray = _Sub
    npt.Shape = _Sub
    npt.Float64 = float
    sys.modules.setdefault("nptyping", npt)


def main():
    install_shims()
    sys.path.insert(0, "/root/reference/src")
    from skelly_sim.skelly_config import ConfigEllipsoidal, Fiber

    np.random.seed(100)  # examples/ellipsoid/gen_config.py:19
    n_fibers = 512       # BASELINE config 4 (the example default is 2000)

    config = ConfigEllipsoidal()
    config.params.dt_write = 0.1
    config.params.dt_initial = 8e-3
    config.params.dt_max = 8e-3
    config.fibers = [
        Fiber(length=1.0, bending_rigidity=2.5e-3, parent_body=-1,
              force_scale=-0.05, minus_clamped=True, n_nodes=64)
        for _ in range(n_fibers)
    ]
    config.periphery.n_nodes = 8192  # "8k-node" periphery at our fixture size
    config.periphery.move_fibers_to_surface(config.fibers, ds_min=0.1)

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = os.path.join(repo, "tests", "golden", "skelly_config_ellipsoid.toml")
    config.save(out)

    # validate the minimal TOML writer output with a strict parser
    import tomli
    d = tomli.loads(open(out).read())
    assert len(d["fibers"]) == n_fibers
    assert d["fibers"][0]["n_nodes"] == 64 and d["fibers"][0]["minus_clamped"]
    assert d["periphery"]["n_nodes"] == 8192
    print("wrote", out, os.path.getsize(out) / 1e6, "MB;",
          len(d["fibers"]), "fibers; eta =", d["params"]["eta"])


if __name__ == "__main__":
    main()
