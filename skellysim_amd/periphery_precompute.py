"""On-GPU periphery operator assembly — the reference's shell precompute
(src/skelly_sim/precompute.py::precompute_periphery, lines 104-135) rebuilt
with the engine's own dense/contraction kernels so the 3N x 3N operators are
assembled and inverted on device in seconds (the reference assembles them in
numba/numpy and inverts with scipy on the host — minutes-to-hours at 8k
nodes).

Given nodes, inward normals and quadrature weights (geometry + weights stay
host-side tooling; see oracle/make_periphery_fixture.py and
oracle/make_ellipsoid_nodes.py):

  S  = stresslet_kernel_times_normal (dense, precompute.py:114)
  ex/ey/ez = singularity-subtraction vectors (periphery.py:72-91; three
       stresslet(normal x density) contractions with e_k * w densities)
  S += -I_blocks - diag(1/w)   (precompute.py:116-126), where I_blocks has
       3x3 diagonal blocks [ex_i | ey_i | ez_i] / w_i
  C  = outer(normals_flat, normals_flat)  (kernels.py::complementary_kernel)
  A  = S + C                    (stresslet_plus_complementary)
  M_inv = A^-1                  (precompute.py:135; rocSOLVER here)

Validated bit-level against the reference-generated 192-node fixture
(tests/test_gpu_periphery_solve.py)."""

import torch

from .evaluator import stresslet_times_normal_device, stresslet_normal_density_device


def assemble_shell_operator(nodes, normals, weights, eta=1.0, want_inverse=True):
    """nodes, normals: (N, 3) CUDA fp64; weights: (N,) CUDA fp64.
    Returns (A, M_inv) device tensors (M_inv None if want_inverse=False)."""
    N = nodes.shape[0]
    dev = nodes.device
    weights = weights.reshape(-1)  # the RBF tooling emits (N, 1)

    S = stresslet_times_normal_device(nodes, normals)  # (3N, 3N)

    # singularity-subtraction vectors (periphery.py:72-91); eta enters the
    # reference call but the kernel's factor is eta-independent
    # (kernels.cpp:311), matching the reference's eta=1 usage.
    evecs = []
    for k in range(3):
        e = torch.zeros((N, 3), dtype=torch.float64, device=dev)
        e[:, k] = weights
        evecs.append(stresslet_normal_density_device(nodes, normals, e).reshape(-1))
    ex, ey, ez = evecs

    # I correction (precompute.py:116-126): block-diagonal 3x3 blocks
    # [ex_i | ey_i | ez_i] / w_i, plus diag(1/w) per component.
    idx = torch.arange(3 * N, device=dev)
    blk_row = idx                                   # 3i+a
    i_node = idx // 3
    for j, e in enumerate((ex, ey, ez)):
        col = 3 * i_node + j
        S[blk_row, col] -= e / weights[i_node]
    S[idx, idx] -= 1.0 / weights[i_node]

    # complementary kernel (kernels.py:769-777)
    nflat = normals.reshape(-1)
    A = S + torch.outer(nflat, nflat)

    from .batched import robust_inv
    M_inv = robust_inv(A) if want_inverse else None
    return A, M_inv
