"""GPU test of the listener service with the product (HIP) compute path."""

import io
import struct

import msgpack
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_listener_velocity_field_hip(tmp_path, hip_lib_path):
    from skellysim_amd.system_fd import HipBackend
    from skellysim_amd.listener import Trajectory, serve, eigen_decode
    from test_listener import _write_traj, _request_bytes
    import oracle

    path, sys_ = _write_traj(tmp_path)
    targets = np.random.default_rng(0).uniform(-1, 1, (257, 3))
    traj = Trajectory(path)
    stdin = io.BytesIO(_request_bytes(1, targets) + struct.pack("<Q", 0))
    stdout = io.BytesIO()
    serve(stdin, stdout, traj, HipBackend(), eta=1.0)
    stdout.seek(0)
    (size,) = struct.unpack("<Q", stdout.read(8))
    res = msgpack.unpackb(stdout.read(size), raw=False)
    u = eigen_decode(res["velocity_field"])

    r_src, wf = [], []
    for f in sys_.fibers:
        sol = np.concatenate([f.x.reshape(-1), f.tension])
        ff = f.force_operator @ sol
        fn = np.stack([ff[i * f.n_nodes:(i + 1) * f.n_nodes] for i in range(3)],
                      axis=1)
        r_src.append(f.x.T)
        wf.append(fn * f.quadrature_weights()[:, None])
    ref = oracle.stokeslet(np.concatenate(r_src), np.concatenate(wf), targets, 1.0)
    rel = np.linalg.norm(u - ref) / np.linalg.norm(ref)
    assert rel < 1e-10, rel
