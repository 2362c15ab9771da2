"""Rigid spherical bodies (MTOCs/centrosomes) — the reference's SphericalBody
(src/core/body_spherical.cpp, include/body.hpp / body_spherical.hpp),
restated for this engine's (n, 3) point-major conventions.

Model (completed double-layer formulation): each body carries surface
densities d (3 per node) plus rigid unknowns [U(3), omega(3)]:
  node rows: M d - K [U, w] = -v_on_body      (body_spherical.cpp:39-62,
             with M applied as flow + the diagonal singularity subtraction)
  com rows:  -K^T lambda + [U, w] = 0         (body_spherical.cpp:61)
The dense preconditioner block is the exact per-body operator
(body_spherical.cpp:104-127), so isolated bodies converge in O(1) GMRES
iterations. Bodies generate flow on everything through the double layer of
their densities plus a center Stokeslet (link/external forces) and a center
rotlet (torques) — body_container.cpp:269-337.
"""

import numpy as np


# ---- quaternions (Eigen convention: stored/printed (w, x, y, z)) ---------

def quat_mult(a, b):
    """Hamilton product a*b, each (w, x, y, z)."""
    aw, ax, ay, az = a
    bw, bx, by, bz = b
    return np.array([
        aw * bw - ax * bx - ay * by - az * bz,
        aw * bx + ax * bw + ay * bz - az * by,
        aw * by - ax * bz + ay * bw + az * bx,
        aw * bz + ax * by - ay * bx + az * bw,
    ])


def quat_to_rot(q):
    w, x, y, z = np.asarray(q, float) / np.linalg.norm(q)
    return np.array([
        [1 - 2 * (y * y + z * z), 2 * (x * y - w * z), 2 * (x * z + w * y)],
        [2 * (x * y + w * z), 1 - 2 * (x * x + z * z), 2 * (y * z - w * x)],
        [2 * (x * z - w * y), 2 * (y * z + w * x), 1 - 2 * (x * x + y * y)],
    ])


class SphericalBody:
    """One rigid spherical body.

    nodes_ref/normals_ref: (n, 3) surface geometry in the body frame
    (normals OUTWARD — the opposite of the periphery convention);
    weights: (n,) quadrature weights; nucleation_sites_ref: (m, 3) fiber
    attachment points in the body frame (body.hpp:31-34). Attachment sites
    must sit OFF the quadrature surface (the reference's config radius is
    the attachment radius, 'the hydrodynamic radius is a bit smaller',
    skelly_config.py:733-734) — a site coinciding with a quadrature node
    makes the attached fiber's base stokeslet near-singular."""

    def __init__(self, nodes_ref, normals_ref, weights, radius,
                 position=(0.0, 0.0, 0.0), orientation=(1.0, 0.0, 0.0, 0.0),
                 nucleation_sites_ref=None, external_force=(0.0, 0.0, 0.0),
                 external_torque=(0.0, 0.0, 0.0),
                 external_force_type="Linear",
                 oscillation_amplitude=0.0, oscillation_omega=0.0,
                 oscillation_phase=0.0):
        self.nodes_ref = np.asarray(nodes_ref, float)
        self.normals_ref = np.asarray(normals_ref, float)
        self.weights = np.asarray(weights, float).reshape(-1)
        self.n_nodes = len(self.nodes_ref)
        self.radius = float(radius)
        self.nucleation_sites_ref = (np.asarray(nucleation_sites_ref, float)
                                     if nucleation_sites_ref is not None
                                     else np.zeros((0, 3)))
        self.external_force = np.asarray(external_force, float)
        self.external_torque = np.asarray(external_torque, float)
        # Linear: constant force; Oscillatory: amplitude*sin(omega*t - phase)
        # along external_force (body_container.cpp:419-426,
        # body_spherical.cpp:243-256)
        self.external_force_type = external_force_type
        self.oscillation_amplitude = float(oscillation_amplitude)
        self.oscillation_omega = float(oscillation_omega)
        self.oscillation_phase = float(oscillation_phase)
        self.velocity = np.zeros(3)
        self.angular_velocity = np.zeros(3)
        self.solution_vec = np.zeros(self.solution_size)
        self.place(np.asarray(position, float), np.asarray(orientation, float))

    @property
    def solution_size(self):
        return 3 * self.n_nodes + 6

    def external_force_at(self, time):
        """BodyContainer::calculate_external_forces_torques
        (body_container.cpp:419-426)."""
        if self.external_force_type == "Oscillatory":
            return self.oscillation_amplitude * np.sin(
                self.oscillation_omega * time - self.oscillation_phase) \
                * self.external_force
        return self.external_force

    def place(self, position, orientation):
        """Move to position/orientation; refresh lab-frame geometry
        (body_spherical.cpp:146-159)."""
        self.position = np.asarray(position, float)
        self.orientation = np.asarray(orientation, float)
        rot = quat_to_rot(self.orientation)
        self.nodes = self.position[None, :] + self.nodes_ref @ rot.T
        self.normals = self.normals_ref @ rot.T
        self.nucleation_sites = self.position[None, :] + \
            self.nucleation_sites_ref @ rot.T

    # ---- per-configuration cache (body_spherical.cpp:94-127) -------------
    def update_cache(self, eta, backend):
        """Singularity-subtraction vectors, K matrix, dense preconditioner.

        ex/ey/ez: columns of stresslet_times_normal_times_density with unit
        densities weighted by the quadrature (body_spherical.cpp:168-181);
        here (n, 3) arrays ex[i] = velocity at node i."""
        n = self.n_nodes
        self.e_sub = []
        for k in range(3):
            dens = np.zeros((n, 3))
            dens[:, k] = self.weights
            self.e_sub.append(backend.stresslet_normal_density(
                self.nodes, self.normals, dens, self.nodes, eta))

        # K (3n, 6): rigid motion at each node (body_spherical.cpp:74-86)
        K = np.zeros((3 * n, 6))
        vec = self.nodes - self.position[None, :]
        for i in range(n):
            K[3 * i: 3 * i + 3, 0:3] = np.eye(3)
            vx, vy, vz = vec[i]
            K[3 * i + 0, 3:6] = [0.0, vz, -vy]
            K[3 * i + 1, 3:6] = [-vz, 0.0, vx]
            K[3 * i + 2, 3:6] = [vy, -vx, 0.0]
        self.K = K

        # dense exact operator = preconditioner (body_spherical.cpp:104-127)
        A = np.zeros((3 * n + 6, 3 * n + 6))
        A[: 3 * n, : 3 * n] = backend.stresslet_times_normal(
            self.nodes, self.normals, eta)
        for i in range(n):
            for k in range(3):
                A[3 * i: 3 * i + 3, 3 * i + k] -= self.e_sub[k][i] / self.weights[i]
        A[: 3 * n, 3 * n:] = -K
        A[3 * n:, : 3 * n] = -K.T
        A[3 * n:, 3 * n:] = np.eye(6)
        import scipy.linalg as scla
        self._A_dense = A
        self._A_lu = scla.lu_factor(A)

    # ---- solver pieces ---------------------------------------------------
    def matvec(self, v_nodes, x):
        """body_spherical.cpp:39-62. v_nodes (n, 3): flow at this body's
        nodes from ALL sources (including every body's double layer);
        x (3n+6): [densities point-major | U | omega]."""
        n = self.n_nodes
        d = x[: 3 * n].reshape(n, 3)
        U = x[3 * n:]
        res = np.empty(3 * n + 6)
        sub = (d[:, 0:1] * self.e_sub[0] + d[:, 1:2] * self.e_sub[1]
               + d[:, 2:3] * self.e_sub[2]) / self.weights[:, None]
        res[: 3 * n] = (-sub).reshape(-1) - self.K @ U + v_nodes.reshape(-1)
        res[3 * n:] = -self.K.T @ x[: 3 * n] + U
        return res

    def apply_preconditioner(self, x):
        import scipy.linalg as scla
        return scla.lu_solve(self._A_lu, x)

    def update_RHS(self, v_nodes):
        """[-v_on_body; 0] (body_spherical.cpp:134-138)."""
        return np.concatenate([-v_nodes.reshape(-1), np.zeros(6)])

    def step(self, dt, sol):
        """Adopt [densities | U | omega]; advance position and orientation
        (body_spherical.cpp:13-35 — with the position update applied even
        for zero angular velocity, where the reference skips place()
        entirely inside its `if (phi_norm)` branch)."""
        n3 = 3 * self.n_nodes
        self.velocity = sol[n3: n3 + 3].copy()
        self.angular_velocity = sol[n3 + 3: n3 + 6].copy()
        self.solution_vec = np.asarray(sol, float).copy()
        x_new = self.position + dt * self.velocity
        phi = dt * self.angular_velocity
        phi_norm = np.linalg.norm(phi)
        q = self.orientation
        if phi_norm > 0:
            s = np.cos(0.5 * phi_norm)
            p = np.sin(0.5 * phi_norm) * phi / phi_norm
            q = quat_mult(np.array([s, p[0], p[1], p[2]]), self.orientation)
        self.place(x_new, q)

    def check_collision(self, other, threshold=0.0):
        """Sphere-sphere (body_spherical.cpp:304-307)."""
        dr2 = np.sum((self.position - other.position) ** 2)
        return dr2 < (self.radius + other.radius + threshold) ** 2


class EllipsoidalBody(SphericalBody):
    """Rigid ellipsoidal body (src/core/body_ellipsoidal.cpp): the solver
    machinery is IDENTICAL to the spherical body's (matvec/preconditioner/
    RHS/step/K are line-for-line the same in the reference; only the
    config radius becomes the 3-vector axis_length and the collision
    checks differ — the reference stubs most of those, lines 283-331).
    The ellipsoidal surface itself enters through the geometry arrays."""

    def __init__(self, nodes_ref, normals_ref, weights, axis_length, **kw):
        self.axis_length = np.asarray(axis_length, float).reshape(3)
        super().__init__(nodes_ref, normals_ref, weights,
                         radius=float(np.max(self.axis_length)), **kw)

    def check_collision(self, other, threshold=0.0):
        """Not implemented in the reference for ellipsoids
        (body_ellipsoidal.cpp returns false with a warning); mirrored."""
        return False


def calculate_link_conditions(fibers, x_fib, body_velocities, bodies):
    """Fiber<->body attachment coupling
    (body_container.cpp:171-268): returns
      velocities_on_fiber (n_fibers, 7): [v(3), tension_cond, w(3)] rows fed
        into each fiber's BC rows of the matvec (zero for unattached fibers),
      body_forces_torques (n_bodies, 6): force/torque each body receives
        from its attached fibers' candidate solution.
    x_fib: the fiber block of the solution vector; body_velocities
    (n_bodies, 6): each body's [U, omega] slice of the candidate solution.
    (The reference advances its fiber solution offset only for attached
    fibers, body_container.cpp:265-266 — an inconsistency for mixed
    populations; here the offset always advances and the velocity rows are
    per-fiber, matching f_c_fd.cpp:216-234's per-fiber consumption.)"""
    n_fib = len(fibers)
    vel_on_fiber = np.zeros((n_fib, 7))
    body_ft = np.zeros((len(bodies), 6))
    off = 0
    for i_fib, fib in enumerate(fibers):
        n = fib.n_nodes
        i_body, i_site = getattr(fib, "binding_site", (-1, -1))
        if i_body < 0:
            off += 4 * n
            continue
        body = bodies[i_body]
        site_pos = body.nucleation_sites[i_site] - body.position
        x_new = x_fib[off: off + 3 * n].reshape(3, n)
        T0 = x_fib[off + 3 * n]
        m = fib.mats
        L = fib.length
        xs0 = fib.xs[:, 0]
        xss_new0 = (2.0 / L) ** 2 * (x_new @ m["D_2_0"])[:, 0]
        xsss_new0 = (2.0 / L) ** 3 * (x_new @ m["D_3_0"])[:, 0]

        E = fib.bending_rigidity
        F_body = -E * xsss_new0 + xs0 * T0
        L_body = (-E * np.cross(site_pos, xsss_new0)
                  + np.cross(site_pos, xs0) * T0
                  + E * np.cross(xs0, xss_new0))
        body_ft[i_body, 0:3] += F_body
        body_ft[i_body, 3:6] += L_body

        U = body_velocities[i_body, 0:3]
        w = body_velocities[i_body, 3:6]
        vel_on_fiber[i_fib, 0:3] = -U - np.cross(w, site_pos)
        vel_on_fiber[i_fib, 3] = -xs0 @ U + np.cross(xs0, site_pos) @ w
        site_hat = site_pos / np.linalg.norm(site_pos)
        vel_on_fiber[i_fib, 4:7] = np.cross(site_hat, w)
        off += 4 * n
    return vel_on_fiber, body_ft
