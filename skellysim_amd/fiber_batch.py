"""Batched (vectorized-over-fibers) assembly of the finite-difference fiber
operators for UNIFORM discretizations — numerically identical to the
per-fiber methods in fiber_fd.py (same formulas, fiber_finite_difference.cpp
citations there), but each of the 16 operator blocks, the RHS and the
boundary-condition rows are built as one broadcasted numpy expression over
the whole fiber population, and the BC downsample is one batched GEMM.
Validated block-for-block against the per-fiber path (tests/test_fiber_fd.py).
"""

import numpy as np

from .fiber_fd import BC_VELOCITY


def assemble_uniform(fibers, dt, eta, flow=None, f_external=None, bc_force=None):
    """update_linear_operator + update_RHS + apply_bc_rectangular for a
    uniform-fiber population. flow/f_external/bc_force: (nf, 3, n) or None
    (bc_force is the f_on_fiber the reference passes to apply_bcs — the
    EXTERNAL forces only, system.cpp:453). Installs A/RHS on each fiber
    (adopt_operator)."""
    nf = len(fibers)
    f0 = fibers[0]
    n = f0.n_nodes
    m = f0.mats
    for f in fibers:
        if f.n_nodes != n:
            raise ValueError("assemble_uniform requires uniform n_nodes")

    # per-fiber scalars (broadcast shapes (nf, 1, 1) / (nf, 1))
    c0 = np.array([f.c0 for f in fibers])[:, None, None]
    c1 = np.array([f.c1 for f in fibers])[:, None, None]
    E = np.array([f.bending_rigidity for f in fibers])[:, None, None]
    beta_dt = np.array([f.beta_tstep for f in fibers])[:, None, None] / dt
    pen = np.array([f.penalty_param for f in fibers])[:, None, None]
    L = np.array([f.length for f in fibers])

    # shared derivative matrices scaled per fiber (uniform length in the
    # configs, but keep the general per-fiber scale)
    s = (2.0 / L)[:, None, None]
    D1 = s * m["D_1_0"].T[None]
    D2 = s ** 2 * m["D_2_0"].T[None]
    D3 = s ** 3 * m["D_3_0"].T[None]
    D4 = s ** 4 * m["D_4_0"].T[None]
    D1pre = np.transpose(D1, (0, 2, 1))

    xs = np.stack([f.xs for f in fibers])      # (nf, 3, n)
    xss = np.stack([f.xss for f in fibers])
    xsss = np.stack([f.xsss for f in fibers])
    x = np.stack([f.x for f in fibers])

    I = np.eye(n)[None]
    A = np.zeros((nf, 4 * n, 4 * n))

    def blk(i, j):
        return (slice(None), slice(i * n, (i + 1) * n), slice(j * n, (j + 1) * n))

    for i in range(3):
        for j in range(3):
            if i == j:
                A[blk(i, i)] = (beta_dt * I
                                + E * c0 * ((1.0 + xs[:, i] ** 2)[:, :, None] * D4)
                                + E * c1 * ((1.0 - xs[:, i] ** 2)[:, :, None] * D4))
            elif j > i:
                A[blk(i, j)] = E * (c0 - c1) * ((xs[:, i] * xs[:, j])[:, :, None] * D4)
            else:
                A[blk(i, j)] = A[blk(j, i)]
        AiT = -(2.0 * c0) * (xs[:, i][:, :, None] * D1)
        AiT[:, np.arange(n), np.arange(n)] -= (c0[:, :, 0] + c1[:, :, 0]) * xss[:, i]
        A[blk(i, 3)] = AiT
        A[blk(3, i)] = (-(c1 + 7.0 * c0) * E * (xss[:, i][:, :, None] * D4)
                        - 6.0 * c0 * E * (xsss[:, i][:, :, None] * D3)
                        - pen * (xs[:, i][:, :, None] * D1))
    ATT = -2.0 * c0 * D2
    ATT[:, np.arange(n), np.arange(n)] += (c0[:, :, 0] + c1[:, :, 0]) * \
        (xss[:, 0] ** 2 + xss[:, 1] ** 2 + xss[:, 2] ** 2)
    A[blk(3, 3)] = ATT

    # ---- RHS (update_RHS) ----
    alpha = m["alpha"]
    vg = np.array([f.v_growth for f in fibers])[:, None]
    s_dot = (1.0 + alpha)[None] * (0.5 * vg)
    RHS = np.zeros((nf, 4 * n))
    for i in range(3):
        RHS[:, i * n: (i + 1) * n] = x[:, i] / dt + s_dot * xs[:, i]
    RHS[:, 3 * n:] = -pen[:, :, 0]
    if flow is not None:
        fD = np.einsum("fin,fnm->fim", flow, D1pre)
        for i in range(3):
            RHS[:, i * n: (i + 1) * n] += flow[:, i]
        RHS[:, 3 * n:] += (xs * fD).sum(axis=1)
    if f_external is not None:
        fs = np.einsum("fin,fnm->fim", f_external, D1pre)
        for i in range(3):
            acc = np.zeros((nf, n))
            for j in range(3):
                delta = 1.0 if i == j else 0.0
                acc += c0[:, :, 0] * ((delta + xs[:, i] * xs[:, j]) * f_external[:, j])
                acc += c1[:, :, 0] * ((delta - xs[:, i] * xs[:, j]) * f_external[:, j])
            RHS[:, i * n: (i + 1) * n] += acc
        RHS[:, 3 * n:] += 2 * c0[:, :, 0] * (xs * fs).sum(axis=1)
        RHS[:, 3 * n:] += (c0[:, :, 0] - c1[:, :, 0]) * (xss * f_external).sum(axis=1)

    # ---- force operator (update_force_operator) ----
    F = np.zeros((nf, 3 * n, 4 * n))
    D4preT = np.transpose(s ** 4 * m["D_4_0"][None], (0, 2, 1))
    for i in range(3):
        F[:, i * n: (i + 1) * n, i * n: (i + 1) * n] = -E * D4preT
        T = np.transpose(D1pre * xs[:, i][:, None, :], (0, 2, 1)).copy()
        T[:, np.arange(n), np.arange(n)] += xss[:, i]
        F[:, i * n: (i + 1) * n, 3 * n: 4 * n] = T

    # ---- boundary conditions (apply_bc_rectangular) ----
    P = m["P_downsample_bc"]
    A[:, : 4 * n - 14, :] = np.matmul(P[None], A)
    RHS[:, : 4 * n - 14] = RHS @ P.T
    B = np.zeros((nf, 14, 4 * n))
    B_RHS = np.zeros((nf, 14))
    von = flow  # v_on_fiber (prep passes it)
    fon = bc_force

    # group fibers by BC signature (vectorize within each group)
    minus_vel = np.array([f.bc_minus[0] == BC_VELOCITY for f in fibers])
    plus_vel = np.array([f.bc_plus[0] == BC_VELOCITY for f in fibers])
    c0f = c0[:, 0, 0]
    Ef = E[:, 0, 0]
    bdtf = beta_dt[:, 0, 0]

    idx = np.where(minus_vel)[0]
    if len(idx):
        for col, comp in ((0, 0), (1, 1), (2, 2)):
            B[idx, col, comp * n] = bdtf[idx]
        for i in range(3):
            B[idx, 3, i * n: (i + 1) * n] = \
                (6.0 * Ef[idx] * c0f[idx] * xss[idx, i, 0])[:, None] * D3[idx, 0]
        B[idx, 3, 3 * n:] = (2.0 * c0f[idx])[:, None] * D1[idx, 0]
        B_RHS[idx, 0:3] = x[idx, :, 0] / dt
        if von is not None:
            B_RHS[idx, 3] -= (xs[idx, :, 0] * von[idx, :, 0]).sum(axis=1)
        if fon is not None:
            B_RHS[idx, 3] -= 2 * c0f[idx] * (xs[idx, :, 0] * fon[idx, :, 0]).sum(axis=1)
    idx = np.where(~minus_vel)[0]
    if len(idx):
        for i in range(3):
            B[idx, i, i * n: (i + 1) * n] = Ef[idx][:, None] * D3[idx, 0]
            B[idx, i, 3 * n] = -xs[idx, i, 0]
            B[idx, 3, i * n: (i + 1) * n] = \
                (-Ef[idx] * xss[idx, i, 0])[:, None] * D2[idx, 0]
        B[idx, 3, 3 * n] = -1.0
        if fon is not None:
            B_RHS[idx, 0:3] = fon[idx, :, 0]
            B_RHS[idx, 3] = (fon[idx, :, 0] * xs[idx, :, 0]).sum(axis=1)
    # minus second BC
    idx = np.where(minus_vel)[0]  # AngularVelocity pairs with Velocity here
    if len(idx):
        for i in range(3):
            B[idx, 4 + i, i * n: (i + 1) * n] = bdtf[idx][:, None] * D1[idx, 0]
        B_RHS[idx, 4:7] = xs[idx, :, 0] / dt
    idx = np.where(~minus_vel)[0]  # Torque
    if len(idx):
        for i in range(3):
            B[idx, 4 + i, i * n: (i + 1) * n] = D2[idx, 0]
    # plus first BC
    idx = np.where(plus_vel)[0]
    if len(idx):
        B[idx, 7, 1 * n - 1] = bdtf[idx]
        B[idx, 8, 2 * n - 1] = bdtf[idx]
        B[idx, 9, 3 * n - 1] = bdtf[idx]
        for i in range(3):
            B[idx, 10, i * n: (i + 1) * n] = \
                (6.0 * Ef[idx] * c0f[idx] * xss[idx, i, -1])[:, None] * D3[idx, -1]
        B[idx, 10, 3 * n:] = (2.0 * c0f[idx])[:, None] * D1[idx, -1]
        B_RHS[idx, 7:10] = x[idx, :, -1] / dt
        if von is not None:
            B_RHS[idx, 10] -= (xs[idx, :, -1] * von[idx, :, -1]).sum(axis=1)
        if fon is not None:
            B_RHS[idx, 10] -= 2 * c0f[idx] * (xs[idx, :, -1] * fon[idx, :, -1]).sum(axis=1)
    idx = np.where(~plus_vel)[0]  # Force
    if len(idx):
        for i in range(3):
            B[idx, 7 + i, i * n: (i + 1) * n] = -Ef[idx][:, None] * D3[idx, -1]
            B[idx, 7 + i, 4 * n - 1] = xs[idx, i, -1]
            B[idx, 10, i * n: (i + 1) * n] = \
                (Ef[idx] * xss[idx, i, -1])[:, None] * D2[idx, -1]
        B[idx, 10, 4 * n - 1] = 1.0
        if fon is not None:
            B_RHS[idx, 7:10] = fon[idx, :, -1]
            B_RHS[idx, 10] = (fon[idx, :, -1] * xs[idx, :, -1]).sum(axis=1)
    # plus second BC: Torque for all supported configurations
    for i in range(3):
        B[:, 11 + i, i * n: (i + 1) * n] = D2[:, -1]

    A[:, 4 * n - 14:, :] = B
    RHS[:, 4 * n - 14:] = B_RHS

    for k, f in enumerate(fibers):
        f.adopt_operator(A[k], RHS[k])
        f.force_operator = F[k]
    return A, RHS, F


def assemble_uniform_t(fibers, dt, eta, flow=None, f_external=None,
                       bc_force=None, device=None):
    """Torch twin of assemble_uniform: the identical block/RHS/BC formulas
    evaluated as device-tensor expressions, so the per-timestep operator
    assembly (the dominant host cost of the solve prep) runs on the GPU.
    flow/f_external/bc_force: (nf, 3, n) numpy arrays or None. Returns
    (A, RHS, F) fp64 tensors on `device`; does NOT install anything on the
    fibers (the caller owns the host materialization). Exactness vs the
    numpy path is pinned by tests/test_fiber_fd.py."""
    import torch

    nf = len(fibers)
    f0 = fibers[0]
    n = f0.n_nodes
    m = f0.mats
    for f in fibers:
        if f.n_nodes != n:
            raise ValueError("assemble_uniform_t requires uniform n_nodes")
    dev = torch.device(device if device is not None else "cpu")
    T = lambda a: torch.from_numpy(np.ascontiguousarray(np.asarray(a, float))).to(dev)
    ar = torch.arange(n, device=dev)

    c0 = T([f.c0 for f in fibers])[:, None, None]
    c1 = T([f.c1 for f in fibers])[:, None, None]
    E = T([f.bending_rigidity for f in fibers])[:, None, None]
    beta_dt = T([f.beta_tstep for f in fibers])[:, None, None] / dt
    pen = T([f.penalty_param for f in fibers])[:, None, None]
    L = T([f.length for f in fibers])

    s = (2.0 / L)[:, None, None]
    D1 = s * T(m["D_1_0"].T)[None]
    D2 = s ** 2 * T(m["D_2_0"].T)[None]
    D3 = s ** 3 * T(m["D_3_0"].T)[None]
    D4 = s ** 4 * T(m["D_4_0"].T)[None]
    D1pre = D1.transpose(1, 2)

    xs = T(np.stack([f.xs for f in fibers]))       # (nf, 3, n)
    xss = T(np.stack([f.xss for f in fibers]))
    xsss = T(np.stack([f.xsss for f in fibers]))
    x = T(np.stack([f.x for f in fibers]))

    I = torch.eye(n, dtype=torch.float64, device=dev)[None]
    A = torch.zeros((nf, 4 * n, 4 * n), dtype=torch.float64, device=dev)

    def blk(i, j):
        return (slice(None), slice(i * n, (i + 1) * n), slice(j * n, (j + 1) * n))

    for i in range(3):
        for j in range(3):
            if i == j:
                A[blk(i, i)] = (beta_dt * I
                                + E * c0 * ((1.0 + xs[:, i] ** 2)[:, :, None] * D4)
                                + E * c1 * ((1.0 - xs[:, i] ** 2)[:, :, None] * D4))
            elif j > i:
                A[blk(i, j)] = E * (c0 - c1) * ((xs[:, i] * xs[:, j])[:, :, None] * D4)
            else:
                A[blk(i, j)] = A[blk(j, i)]
        AiT = -(2.0 * c0) * (xs[:, i][:, :, None] * D1)
        AiT[:, ar, ar] -= (c0[:, :, 0] + c1[:, :, 0]) * xss[:, i]
        A[blk(i, 3)] = AiT
        A[blk(3, i)] = (-(c1 + 7.0 * c0) * E * (xss[:, i][:, :, None] * D4)
                        - 6.0 * c0 * E * (xsss[:, i][:, :, None] * D3)
                        - pen * (xs[:, i][:, :, None] * D1))
    ATT = -2.0 * c0 * D2
    ATT[:, ar, ar] += (c0[:, :, 0] + c1[:, :, 0]) * \
        (xss[:, 0] ** 2 + xss[:, 1] ** 2 + xss[:, 2] ** 2)
    A[blk(3, 3)] = ATT

    # ---- RHS (update_RHS) ----
    alpha = T(m["alpha"])
    vg = T([f.v_growth for f in fibers])[:, None]
    s_dot = (1.0 + alpha)[None] * (0.5 * vg)
    RHS = torch.zeros((nf, 4 * n), dtype=torch.float64, device=dev)
    flow_t = T(flow) if flow is not None else None
    fe_t = T(f_external) if f_external is not None else None
    fon_t = T(bc_force) if bc_force is not None else None
    for i in range(3):
        RHS[:, i * n: (i + 1) * n] = x[:, i] / dt + s_dot * xs[:, i]
    RHS[:, 3 * n:] = -pen[:, :, 0]
    if flow_t is not None:
        fD = torch.einsum("fin,fnm->fim", flow_t, D1pre)
        for i in range(3):
            RHS[:, i * n: (i + 1) * n] += flow_t[:, i]
        RHS[:, 3 * n:] += (xs * fD).sum(dim=1)
    if fe_t is not None:
        fs = torch.einsum("fin,fnm->fim", fe_t, D1pre)
        for i in range(3):
            acc = torch.zeros((nf, n), dtype=torch.float64, device=dev)
            for j in range(3):
                delta = 1.0 if i == j else 0.0
                acc += c0[:, :, 0] * ((delta + xs[:, i] * xs[:, j]) * fe_t[:, j])
                acc += c1[:, :, 0] * ((delta - xs[:, i] * xs[:, j]) * fe_t[:, j])
            RHS[:, i * n: (i + 1) * n] += acc
        RHS[:, 3 * n:] += 2 * c0[:, :, 0] * (xs * fs).sum(dim=1)
        RHS[:, 3 * n:] += (c0[:, :, 0] - c1[:, :, 0]) * (xss * fe_t).sum(dim=1)

    # ---- force operator (update_force_operator) ----
    F = torch.zeros((nf, 3 * n, 4 * n), dtype=torch.float64, device=dev)
    D4preT = (s ** 4 * T(m["D_4_0"])[None]).transpose(1, 2)
    for i in range(3):
        F[:, i * n: (i + 1) * n, i * n: (i + 1) * n] = -E * D4preT
        Tt = (D1pre * xs[:, i][:, None, :]).transpose(1, 2).clone()
        Tt[:, ar, ar] += xss[:, i]
        F[:, i * n: (i + 1) * n, 3 * n: 4 * n] = Tt

    # ---- boundary conditions (apply_bc_rectangular) ----
    P = T(m["P_downsample_bc"])
    A[:, : 4 * n - 14, :] = torch.matmul(P[None], A)
    RHS[:, : 4 * n - 14] = RHS @ P.T
    B = torch.zeros((nf, 14, 4 * n), dtype=torch.float64, device=dev)
    B_RHS = torch.zeros((nf, 14), dtype=torch.float64, device=dev)

    minus_vel = np.array([f.bc_minus[0] == BC_VELOCITY for f in fibers])
    plus_vel = np.array([f.bc_plus[0] == BC_VELOCITY for f in fibers])
    c0f = c0[:, 0, 0]
    Ef = E[:, 0, 0]
    bdtf = beta_dt[:, 0, 0]

    def _w(mask):
        return torch.from_numpy(np.where(mask)[0]).to(dev)

    idx = _w(minus_vel)
    if len(idx):
        for col, comp in ((0, 0), (1, 1), (2, 2)):
            B[idx, col, comp * n] = bdtf[idx]
        for i in range(3):
            B[idx, 3, i * n: (i + 1) * n] = \
                (6.0 * Ef[idx] * c0f[idx] * xss[idx, i, 0])[:, None] * D3[idx, 0]
        B[idx, 3, 3 * n:] = (2.0 * c0f[idx])[:, None] * D1[idx, 0]
        B_RHS[idx, 0:3] = x[idx, :, 0] / dt
        if flow_t is not None:
            B_RHS[idx, 3] -= (xs[idx, :, 0] * flow_t[idx, :, 0]).sum(dim=1)
        if fon_t is not None:
            B_RHS[idx, 3] -= 2 * c0f[idx] * (xs[idx, :, 0] * fon_t[idx, :, 0]).sum(dim=1)
    idx = _w(~minus_vel)
    if len(idx):
        for i in range(3):
            B[idx, i, i * n: (i + 1) * n] = Ef[idx][:, None] * D3[idx, 0]
            B[idx, i, 3 * n] = -xs[idx, i, 0]
            B[idx, 3, i * n: (i + 1) * n] = \
                (-Ef[idx] * xss[idx, i, 0])[:, None] * D2[idx, 0]
        B[idx, 3, 3 * n] = -1.0
        if fon_t is not None:
            B_RHS[idx, 0:3] = fon_t[idx, :, 0]
            B_RHS[idx, 3] = (fon_t[idx, :, 0] * xs[idx, :, 0]).sum(dim=1)
    idx = _w(minus_vel)  # AngularVelocity pairs with Velocity here
    if len(idx):
        for i in range(3):
            B[idx, 4 + i, i * n: (i + 1) * n] = bdtf[idx][:, None] * D1[idx, 0]
        B_RHS[idx, 4:7] = xs[idx, :, 0] / dt
    idx = _w(~minus_vel)  # Torque
    if len(idx):
        for i in range(3):
            B[idx, 4 + i, i * n: (i + 1) * n] = D2[idx, 0]
    idx = _w(plus_vel)
    if len(idx):
        B[idx, 7, 1 * n - 1] = bdtf[idx]
        B[idx, 8, 2 * n - 1] = bdtf[idx]
        B[idx, 9, 3 * n - 1] = bdtf[idx]
        for i in range(3):
            B[idx, 10, i * n: (i + 1) * n] = \
                (6.0 * Ef[idx] * c0f[idx] * xss[idx, i, -1])[:, None] * D3[idx, -1]
        B[idx, 10, 3 * n:] = (2.0 * c0f[idx])[:, None] * D1[idx, -1]
        B_RHS[idx, 7:10] = x[idx, :, -1] / dt
        if flow_t is not None:
            B_RHS[idx, 10] -= (xs[idx, :, -1] * flow_t[idx, :, -1]).sum(dim=1)
        if fon_t is not None:
            B_RHS[idx, 10] -= 2 * c0f[idx] * (xs[idx, :, -1] * fon_t[idx, :, -1]).sum(dim=1)
    idx = _w(~plus_vel)  # Force
    if len(idx):
        for i in range(3):
            B[idx, 7 + i, i * n: (i + 1) * n] = -Ef[idx][:, None] * D3[idx, -1]
            B[idx, 7 + i, 4 * n - 1] = xs[idx, i, -1]
            B[idx, 10, i * n: (i + 1) * n] = \
                (Ef[idx] * xss[idx, i, -1])[:, None] * D2[idx, -1]
        B[idx, 10, 4 * n - 1] = 1.0
        if fon_t is not None:
            B_RHS[idx, 7:10] = fon_t[idx, :, -1]
            B_RHS[idx, 10] = (fon_t[idx, :, -1] * xs[idx, :, -1]).sum(dim=1)
    # plus second BC: Torque for all supported configurations
    for i in range(3):
        B[:, 11 + i, i * n: (i + 1) * n] = D2[:, -1]

    A[:, 4 * n - 14:, :] = B
    RHS[:, 4 * n - 14:] = B_RHS
    return A, RHS, F
