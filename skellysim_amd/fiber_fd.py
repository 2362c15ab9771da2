"""Finite-difference fiber model — host-side restatement of the reference's
FiberFiniteDifference (src/core/fiber_finite_difference.cpp, .hpp).

This is the L5 per-fiber dense algebra of SURVEY.md §2: small (4n x 4n,
n <= 128) operator assembly on the host, with the O(N^2) pair interactions,
batched LU preconditioning and batched self-stokeslet builds routed to the
GPU at the container/system level (system_fd.py, flows.py, batched.py).

Numerics follow the reference line by line (citations per method):
  - Fornberg finite-difference matrices and barycentric resampling
    (src/core/utils.cpp:12-36, 48-105), 4th-order scheme
    (compute_matrices_finitediff(4), fiber_finite_difference.cpp:517-562);
  - operator A_ (update_linear_operator, cpp:95-190),
  - RHS (update_RHS, cpp:192-276),
  - rectangular boundary conditions (apply_bc_rectangular, cpp:345-514),
  - force operator (update_force_operator, cpp:317-337),
  - per-fiber matvec (cpp:278-315),
  - slender-body constants c_0, c_1 (fiber_finite_difference.hpp:140-144).
"""

import numpy as np


def finite_diff(s, M, n_s):
    """Fornberg finite-difference matrix (utils.cpp:48-105): derivative of
    order M at every grid point of s using n_s support points."""
    s = np.asarray(s, float)
    N = len(s) - 1
    D_s = np.zeros((N + 1, N + 1))
    n_s_half = (n_s - 1) // 2
    n_s = n_s - 1
    for xi in range(len(s)):
        si = s[xi]
        if xi < n_s_half:
            xlow, xhigh = 0, n_s + 1
        elif xi > len(s) - n_s_half - 2:
            xlow, xhigh = -n_s - 1, len(s)
        else:
            xlow, xhigh = xi - n_s_half, xi - n_s_half + n_s + 1
        if xlow < 0:
            xlow = len(s) + xlow
        x = s[xlow:xhigh]

        c1, c4 = 1.0, x[0] - si
        c = np.zeros((n_s + 1, M + 1))
        c[0, 0] = 1.0
        for i in range(1, n_s + 1):
            mn = min(i, M)
            c2, c5 = 1.0, c4
            c4 = x[i] - si
            for j in range(i):
                c3 = x[i] - x[j]
                c2 *= c3
                if j == i - 1:
                    for k in range(mn, 0, -1):
                        c[i, k] = c1 * (k * c[i - 1, k - 1] - c5 * c[i - 1, k]) / c2
                    c[i, 0] = -c1 * c5 * c[i - 1, 0] / c2
                for k in range(mn, 0, -1):
                    c[j, k] = (c4 * c[j, k] - k * c[j, k - 1]) / c3
                c[j, 0] = c4 * c[j, 0] / c3
            c1 = c2
        D_s[xi, xlow: xlow + n_s + 1] = c[:, M]
    return D_s


def barycentric_matrix(x, y):
    """Resampling matrix P_{N,-m} (utils.cpp:12-36)."""
    x = np.asarray(x, float)
    y = np.asarray(y, float)
    N, M = len(x), len(y)
    w = np.ones(N)
    w[1::2] = -1.0
    w[0] = 0.5
    w[N - 1] = -0.5 * (-1.0) ** N
    P = np.zeros((M, N))
    for j in range(M):
        S = np.sum(w / (y[j] - x))
        for k in range(N):
            if abs(y[j] - x[k]) > np.finfo(float).eps:
                P[j, k] = w[k] / (y[j] - x[k]) / S
            else:
                P[j, k] = 1.0
    return P


_MATS_CACHE = {}


def fib_mats(n_nodes, n_fd=4):
    """Per-n_nodes FD matrices (compute_matrices_finitediff,
    fiber_finite_difference.cpp:517-562). D_k_0 are PRE-TRANSPOSED like the
    reference's (so (3,n) positions right-multiply them)."""
    key = (n_nodes, n_fd)
    if key in _MATS_CACHE:
        return _MATS_CACHE[key]
    m = {}
    alpha = np.linspace(-1.0, 1.0, n_nodes)
    n_roots = n_nodes - 4
    alpha_roots = 2 * (0.5 + np.arange(n_roots)) / n_roots - 1
    n_tension = n_nodes - 2
    alpha_tension = 2 * (0.5 + np.arange(n_tension)) / n_tension - 1
    m["alpha"] = alpha
    m["D_1_0"] = finite_diff(alpha, 1, n_fd + 1).T
    m["D_2_0"] = finite_diff(alpha, 2, n_fd + 2).T
    m["D_3_0"] = finite_diff(alpha, 3, n_fd + 3).T
    m["D_4_0"] = finite_diff(alpha, 4, n_fd + 4).T
    m["P_X"] = barycentric_matrix(alpha, alpha_roots)
    m["P_T"] = barycentric_matrix(alpha, alpha_tension)
    w0 = np.full(n_nodes, 2.0)
    w0[0] = w0[-1] = 1.0
    w0 /= n_nodes - 1
    m["weights_0"] = w0
    np_ = n_nodes
    P = np.zeros((4 * np_ - 14, 4 * np_))
    P[0 * (np_ - 4): 1 * (np_ - 4), 0 * np_: 1 * np_] = m["P_X"]
    P[1 * (np_ - 4): 2 * (np_ - 4), 1 * np_: 2 * np_] = m["P_X"]
    P[2 * (np_ - 4): 3 * (np_ - 4), 2 * np_: 3 * np_] = m["P_X"]
    P[3 * (np_ - 4): 3 * (np_ - 4) + np_ - 2, 3 * np_: 4 * np_] = m["P_T"]
    m["P_downsample_bc"] = P
    _MATS_CACHE[key] = m
    return m


# boundary-condition kinds (fiber_finite_difference.cpp:23-24)
BC_FORCE = "Force"
BC_TORQUE = "Torque"
BC_VELOCITY = "Velocity"
BC_ANGULAR_VELOCITY = "AngularVelocity"


def points_collide(pc, periphery_shape, threshold=0.0):
    """Point cloud (3, n) vs periphery collision
    (SphericalPeriphery::check_collision periphery.cpp:126-133 /
    EllipsoidalPeriphery periphery.cpp:203-224). periphery_shape: dict with
    kind="sphere"|"ellipsoid" and radius=/abc=."""
    if periphery_shape["kind"] == "sphere":
        r2 = (periphery_shape["radius"] - threshold) ** 2
        return bool(np.any((pc ** 2).sum(axis=0) >= r2))
    a, b, c = periphery_shape["abc"]
    r_s = pc / np.array([a, b, c])[:, None]
    r_s_mag = np.linalg.norm(r_s, axis=0)
    phi = np.arctan2(r_s[1], r_s[0] + 1e-12)
    theta = np.arccos(r_s[2] / (1e-12 + r_s_mag))
    st = np.sin(theta)
    rc = np.stack([(a - threshold) * st * np.cos(phi),
                   (b - threshold) * st * np.sin(phi),
                   (c - threshold) * np.cos(theta)])
    return bool(np.any((pc ** 2).sum(axis=0) >= (rc ** 2).sum(axis=0)))


class FiberFD:
    """One finite-difference fiber (reference FiberFiniteDifference)."""

    def __init__(self, x, length, bending_rigidity, eta, radius=0.0125,
                 force_scale=0.0, minus_clamped=False, penalty_param=500.0,
                 beta_tstep=1.0, layout=None):
        x = np.asarray(x, float)
        if x.shape == (3, 3) and layout is None:
            raise ValueError(
                "ambiguous (3, 3) node array: pass layout='coords_major' "
                "for (3, n) input or layout='nodes_major' for (n, 3)")
        if layout == "nodes_major" or (layout is None and x.shape[0] != 3):
            x = x.T
        self.x = np.ascontiguousarray(x, dtype=float)
        if self.x.shape[0] != 3:
            raise ValueError("x must be (3, n) or (n, 3)")
        self.n_nodes = self.x.shape[1]
        self.length = float(length)
        self.length_prev = float(length)
        self.bending_rigidity = float(bending_rigidity)
        self.radius = float(radius)
        self.force_scale = float(force_scale)
        self.minus_clamped = bool(minus_clamped)
        self.penalty_param = float(penalty_param)
        self.beta_tstep = float(beta_tstep)
        self.v_growth = 0.0
        self.binding_site = (-1, -1)  # (i_body, i_site); body.hpp:31-34
        self.tension = np.zeros(self.n_nodes)
        self.mats = fib_mats(self.n_nodes)
        # free ends by default; update_boundary_conditions may change
        self.bc_minus = (BC_VELOCITY, BC_ANGULAR_VELOCITY) if minus_clamped \
            else (BC_FORCE, BC_TORQUE)
        self.bc_plus = (BC_FORCE, BC_TORQUE)
        self.update_constants(eta)

    # hpp:140-144
    def update_constants(self, eta):
        self.epsilon = self.radius / self.length
        self.c0 = -np.log(np.e * self.epsilon ** 2) / (8 * np.pi * eta)
        self.c1 = 2.0 / (8.0 * np.pi * eta)

    # cpp:62-68
    def update_derivatives(self):
        m = self.mats
        s = 2.0 / self.length_prev
        self.xs = s * self.x @ m["D_1_0"]
        self.xss = s ** 2 * self.x @ m["D_2_0"]
        self.xsss = s ** 3 * self.x @ m["D_3_0"]
        self.xssss = s ** 4 * self.x @ m["D_4_0"]

    # cpp:95-190
    def update_linear_operator(self, dt, eta):
        np_ = self.n_nodes
        m = self.mats
        E = self.bending_rigidity
        c0, c1 = self.c0, self.c1
        # D_k = mats.D_k_0.transpose() * scale^k — back to "rows are nodes"
        D1 = m["D_1_0"].T * (2.0 / self.length) ** 1
        D2 = m["D_2_0"].T * (2.0 / self.length) ** 2
        D3 = m["D_3_0"].T * (2.0 / self.length) ** 3
        D4 = m["D_4_0"].T * (2.0 / self.length) ** 4
        I = np.eye(np_)
        xs, xss, xsss = self.xs, self.xss, self.xsss

        A = np.zeros((4 * np_, 4 * np_))

        def blk(i, j):
            return (slice(i * np_, (i + 1) * np_), slice(j * np_, (j + 1) * np_))

        for i in range(3):
            for j in range(3):
                if i == j:
                    A[blk(i, i)] = (self.beta_tstep / dt * I
                                    + E * c0 * ((1.0 + xs[i] ** 2)[:, None] * D4)
                                    + E * c1 * ((1.0 - xs[i] ** 2)[:, None] * D4))
                elif j > i:
                    A[blk(i, j)] = E * (c0 - c1) * ((xs[i] * xs[j])[:, None] * D4)
                else:
                    A[blk(i, j)] = A[blk(j, i)]
            # A_iT (cpp:165-173)
            A[blk(i, 3)] = -(2.0 * c0) * (xs[i][:, None] * D1) \
                - (c0 + c1) * np.diag(self.xss[i])
            # A_Ti (cpp:178-190)
            A[blk(3, i)] = (-(c1 + 7.0 * c0) * E * (xss[i][:, None] * D4)
                            - 6.0 * c0 * E * (xsss[i][:, None] * D3)
                            - self.penalty_param * (xs[i][:, None] * D1))
        A[blk(3, 3)] = -2.0 * c0 * D2 + np.diag(
            (c0 + c1) * (xss[0] ** 2 + xss[1] ** 2 + xss[2] ** 2))
        self.A = A

    # cpp:192-276
    def update_RHS(self, dt, flow=None, f_external=None):
        np_ = self.n_nodes
        m = self.mats
        D1pre = m["D_1_0"] * (2.0 / self.length)  # pre-transposed, cpp:201
        xs = self.xs
        alpha = m["alpha"]
        s_dot = (1.0 + alpha) * (0.5 * self.v_growth)
        RHS = np.zeros(4 * np_)
        for i in range(3):
            RHS[i * np_: (i + 1) * np_] = self.x[i] / dt + s_dot * xs[i]
        RHS[3 * np_:] = -self.penalty_param

        if flow is not None and flow.size:
            for i in range(3):
                RHS[i * np_: (i + 1) * np_] += flow[i]
            RHS[3 * np_:] += (xs[0] * (flow[0] @ D1pre) + xs[1] * (flow[1] @ D1pre)
                              + xs[2] * (flow[2] @ D1pre))
        if f_external is not None and f_external.size:
            c0, c1 = self.c0, self.c1
            fs = f_external @ D1pre
            f = f_external
            for i in range(3):
                acc = np.zeros(np_)
                for j in range(3):
                    delta = 1.0 if i == j else 0.0
                    acc += c0 * ((delta + xs[i] * xs[j]) * f[j])
                    acc += c1 * ((delta - xs[i] * xs[j]) * f[j])
                RHS[i * np_: (i + 1) * np_] += acc
            RHS[3 * np_:] += 2 * c0 * (xs[0] * fs[0] + xs[1] * fs[1] + xs[2] * fs[2])
            RHS[3 * np_:] += (c0 - c1) * (self.xss[0] * f[0] + self.xss[1] * f[1]
                                          + self.xss[2] * f[2])
        self.RHS = RHS

    # cpp:317-337
    def update_force_operator(self):
        np_ = self.n_nodes
        m = self.mats
        D1pre = m["D_1_0"] * (2.0 / self.length)
        D4pre = m["D_4_0"] * (2.0 / self.length) ** 4
        F = np.zeros((3 * np_, 4 * np_))
        for i in range(3):
            F[i * np_: (i + 1) * np_, i * np_: (i + 1) * np_] = \
                -self.bending_rigidity * D4pre.T
            T = np.diag(self.xss[i]) + (D1pre * self.xs[i][None, :]).T
            F[i * np_: (i + 1) * np_, 3 * np_: 4 * np_] = T
        self.force_operator = F

    # cpp:278-315
    def matvec(self, x, v, v_boundary=None):
        np_ = self.n_nodes
        m = self.mats
        bc_start = 4 * np_ - 14
        D1pre = m["D_1_0"] * (2.0 / self.length_prev)
        vT = np.zeros(4 * np_)
        vT[0 * np_: 1 * np_] = v[0]
        vT[1 * np_: 2 * np_] = v[1]
        vT[2 * np_: 3 * np_] = v[2]
        # xsDs = (D_1.colwise()*xs_i).transpose() = D1pre^T @ diag(xs_i)
        for i in range(3):
            vT[3 * np_:] += D1pre.T @ (self.xs[i] * v[i])
        vT_in = np.zeros(4 * np_)
        vT_in[:bc_start] = m["P_downsample_bc"] @ vT

        xs_vT = np.zeros(4 * np_)
        xs_vT[bc_start + 3] = v[:, 0] @ self.xs[:, 0]
        if self.bc_plus[0] == BC_VELOCITY:
            xs_vT[bc_start + 10] = v[:, -1] @ self.xs[:, -1]

        y_BC = np.zeros(4 * np_)
        if v_boundary is not None and len(v_boundary):
            y_BC[bc_start: bc_start + 7] = v_boundary

        return self.A @ x - vT_in + xs_vT + y_BC

    # cpp:345-514
    def apply_bc_rectangular(self, dt, v_on_fiber=None, f_on_fiber=None):
        np_ = self.n_nodes
        m = self.mats
        E = self.bending_rigidity
        c0 = self.c0
        s = 2.0 / self.length
        D1 = m["D_1_0"].T * s
        D2 = m["D_2_0"].T * s ** 2
        D3 = m["D_3_0"].T * s ** 3
        xs, xss = self.xs, self.xss

        # downsample A and RHS, leaving last 14 rows for the BCs (cpp:353-360)
        P = m["P_downsample_bc"]
        self.A[: 4 * np_ - 14, :] = P @ self.A
        self.RHS[: 4 * np_ - 14] = P @ self.RHS
        B = np.zeros((14, 4 * np_))
        B_RHS = np.zeros(14)

        def seg(i):
            return slice(i * np_, (i + 1) * np_)

        if self.bc_minus[0] == BC_VELOCITY:  # clamped (cpp:365-389)
            B[0, 0 * np_] = self.beta_tstep / dt
            B[1, 1 * np_] = self.beta_tstep / dt
            B[2, 2 * np_] = self.beta_tstep / dt
            for i in range(3):
                B[3, seg(i)] = (6.0 * E * c0) * xss[i, 0] * D3[0]
            B[3, seg(3)] = (2.0 * c0) * D1[0]
            B_RHS[0:3] = self.x[:, 0] / dt
            B_RHS[3] = 0.0
            if v_on_fiber is not None and v_on_fiber.size:
                B_RHS[3] -= self.xs[:, 0] @ v_on_fiber[:, 0]
            if f_on_fiber is not None and f_on_fiber.size:
                B_RHS[3] -= 2 * c0 * (self.xs[:, 0] @ f_on_fiber[:, 0])
        elif self.bc_minus[0] == BC_FORCE:  # free (cpp:390-407)
            for i in range(3):
                B[i, seg(i)] = E * D3[0]
                B[i, 3 * np_] = -xs[i, 0]
                B[3, seg(i)] = -E * D2[0] * xss[i, 0]
            B[3, 3 * np_] = -1.0
            f0 = np.zeros(3)
            if f_on_fiber is not None and f_on_fiber.size:
                f0 = f_on_fiber[:, 0]
            B_RHS[0:3] = f0
            B_RHS[3] = f0 @ xs[:, 0]
        else:
            raise NotImplementedError(self.bc_minus)

        if self.bc_minus[1] == BC_ANGULAR_VELOCITY:  # cpp:416-428
            for i in range(3):
                B[4 + i, seg(i)] = (self.beta_tstep / dt) * D1[0]
            B_RHS[4:7] = xs[:, 0] / dt
        elif self.bc_minus[1] == BC_TORQUE:  # cpp:429-437
            for i in range(3):
                B[4 + i, seg(i)] = D2[0]
        else:
            raise NotImplementedError(self.bc_minus)

        if self.bc_plus[0] == BC_VELOCITY:  # hinged at cortex (cpp:444-467)
            B[7, 1 * np_ - 1] = self.beta_tstep / dt
            B[8, 2 * np_ - 1] = self.beta_tstep / dt
            B[9, 3 * np_ - 1] = self.beta_tstep / dt
            for i in range(3):
                B[10, seg(i)] = (6.0 * E * c0) * D3[-1] * xss[i, -1]
            B[10, seg(3)] = (2.0 * c0) * D1[-1]
            B_RHS[7:10] = self.x[:, -1] / dt
            B_RHS[10] = 0.0
            if v_on_fiber is not None and v_on_fiber.size:
                B_RHS[10] -= self.xs[:, -1] @ v_on_fiber[:, -1]
            if f_on_fiber is not None and f_on_fiber.size:
                B_RHS[10] -= 2 * c0 * (self.xs[:, -1] @ f_on_fiber[:, -1])
        elif self.bc_plus[0] == BC_FORCE:  # free (cpp:468-490)
            for i in range(3):
                B[7 + i, seg(i)] = -E * D3[-1]
                B[7 + i, 4 * np_ - 1] = xs[i, -1]
                B[10, seg(i)] = E * D2[-1] * xss[i, -1]
            B[10, 4 * np_ - 1] = 1.0
            fend = np.zeros(3)
            if f_on_fiber is not None and f_on_fiber.size:
                fend = f_on_fiber[:, -1]
            B_RHS[7:10] = fend
            B_RHS[10] = fend @ xs[:, -1]
        else:
            raise NotImplementedError(self.bc_plus)

        if self.bc_plus[1] == BC_TORQUE:  # cpp:498-506
            for i in range(3):
                B[11 + i, seg(i)] = D2[-1]
        else:
            raise NotImplementedError(self.bc_plus)

        self.A[4 * np_ - 14:, :] = B
        self.RHS[4 * np_ - 14:] = B_RHS

    def update_boundary_conditions(self, periphery_shape=None,
                                   periphery_binding=None):
        """FiberFiniteDifference::update_boundary_conditions
        (fiber_finite_difference.cpp:74-91): minus end clamped when attached
        (to a body or by the flag), plus end hinged at the cortex
        (Velocity, Torque) when periphery binding is active, the plus end's
        polar angle is inside [polar_angle_start, polar_angle_end], and the
        fiber collides with the shell within `threshold`."""
        attached = self.minus_clamped or self.binding_site[0] >= 0
        self.bc_minus = (BC_VELOCITY, BC_ANGULAR_VELOCITY) if attached \
            else (BC_FORCE, BC_TORQUE)
        near = False
        pb = periphery_binding
        if pb and pb.get("active", False) and periphery_shape is not None:
            plus = self.x[:, -1]
            angle = float(np.arccos(plus[2] / np.linalg.norm(plus)))
            if pb.get("polar_angle_start", 0.0) <= angle \
                    <= pb.get("polar_angle_end", 0.5 * np.pi):
                near = points_collide(self.x, periphery_shape,
                                      pb.get("threshold", 0.75))
        self.bc_plus = (BC_VELOCITY, BC_TORQUE) if near \
            else (BC_FORCE, BC_TORQUE)

    def periphery_repulsion(self, kind, f_0=20.0, l_0=0.05, radius=None,
                            abc=None):
        """Steric repulsion from the periphery on this fiber's nodes
        (SphericalPeriphery::fiber_interaction, periphery.cpp:140-162;
        EllipsoidalPeriphery::fiber_interaction, periphery.cpp:232-263;
        defaults f_0=20, l_0=0.05, params.hpp:46-47). Returns (3, n);
        minus-clamped fibers skip node 0 (periphery.cpp:148)."""
        pc = self.x
        f = np.zeros_like(pc)
        start = 1 if self.minus_clamped else 0
        for i in range(start, pc.shape[1]):
            p = pc[:, i]
            r_mag = np.linalg.norm(p)
            if kind == "sphere":
                if r_mag >= radius or r_mag == 0.0:
                    # >= radius: collision, force routine leaves zero
                    # (periphery.cpp:152); == 0: the reference would divide
                    # 0/0 — a node exactly at the center gets no force here
                    continue
                u_hat = p / r_mag
                dr = p - u_hat * radius
                gap = radius - r_mag
            elif kind == "ellipsoid":
                a, b, c = abc
                r_s = p / np.array([a, b, c])
                r_s_mag = np.linalg.norm(r_s)
                phi = np.arctan2(r_s[1], r_s[0] + 1e-12)
                theta = np.arccos(r_s[2] / (1e-12 + r_s_mag))
                st = np.sin(theta)
                r_cortex = np.array([a * st * np.cos(phi), b * st * np.sin(phi),
                                     c * np.cos(theta)])
                r_cortex_mag = np.linalg.norm(r_cortex)
                if r_mag >= r_cortex_mag:
                    continue
                dr = p - r_cortex
                gap = r_cortex_mag - r_mag
            else:
                raise ValueError(kind)
            d = np.linalg.norm(dr)
            f[:, i] = f_0 * dr / d * np.exp(-gap / l_0)
        return f

    def adopt_operator(self, A, RHS):
        """Install externally (batch-)assembled operator state."""
        self.A = A
        self.RHS = RHS

    def quadrature_weights(self):
        """0.5 * length * weights_0 (fiber_container_finite_difference.cpp:186)."""
        return 0.5 * self.length * self.mats["weights_0"]

    def step(self, sol):
        """Adopt a 4n solution slice (fiber_container_finite_difference.cpp:292-303)."""
        np_ = self.n_nodes
        for i in range(3):
            self.x[i] = sol[i * np_: (i + 1) * np_]
        self.tension = sol[3 * np_:].copy()
