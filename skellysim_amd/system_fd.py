"""System orchestrator for fiber + periphery solves — the thin host harness
around the accelerated evaluators ("host control flow; kept as
thin harness around the accelerated evaluator for configs 4-5").

Mirrors the reference solve pipeline for finite-difference fibers and a
periphery, without bodies (src/core/system.cpp):
  prep_state_for_solver  (system.cpp:396-459)
  apply_matvec           (system.cpp:269-324)
  apply_preconditioner   (system.cpp:248-263)
  solve / step           (system.cpp:464-493) — backward-Euler via the
                         beta_tstep/dt terms inside the fiber operator.

Pair interactions, batched LU and the shell GEMVs run through a pluggable
backend: HipBackend (the product path — device kernels, fails loudly with
no GPU) or any object with the same methods (tests inject the CPU oracle to
validate the orchestration end-to-end off-GPU).
"""

import numpy as np

from .fiber_fd import FiberFD


class HipBackend:
    """Product backend: pair kernels + batched algebra on the MI355X."""

    def __init__(self, device="cuda:0"):
        import torch
        self.torch = torch
        self.dev = torch.device(device)

    def _t(self, a):
        return self.torch.from_numpy(np.ascontiguousarray(a)).to(self.dev)

    def stokeslet(self, r_src, f_src, r_trg, eta):
        from .evaluator import stokeslet_device
        u = stokeslet_device(self._t(r_src), self._t(f_src), self._t(r_trg), eta)
        self.torch.cuda.synchronize()
        return u.cpu().numpy()

    def stresslet_normal_density(self, r_src, normals, density, r_trg, eta):
        """Shell/body double layer: f_dl = 2*eta*n (x) d then stresslet/eta
        (periphery.cpp:68-74)."""
        from .evaluator import stresslet_device
        f_dl = 2.0 * eta * np.einsum("ni,nj->nij", normals, density).reshape(-1, 9)
        u = stresslet_device(self._t(r_src), self._t(f_dl), self._t(r_trg), eta)
        self.torch.cuda.synchronize()
        return u.cpu().numpy()

    def self_stokeslet_batch(self, pts, eta):
        """(nf, n, 3) -> (nf, 3n, 3n) oseen_tensor_direct per fiber."""
        from .evaluator import oseen_tensor_batched_device
        G = oseen_tensor_batched_device(self._t(pts), eta=eta)
        self.torch.cuda.synchronize()
        return G.cpu().numpy()

    def batched_lu(self, A_batch):
        """Factor (nf, m, m); returns solve(rhs_batch (nf, m)) -> (nf, m)."""
        from .batched import BatchedLU
        lu = BatchedLU(self._t(A_batch))

        def solve(rhs):
            x = lu.solve(self._t(rhs))
            self.torch.cuda.synchronize()
            return x.cpu().numpy()

        return solve

    def shell_ops(self, A, M_inv):
        At, Mt = self._t(A), self._t(M_inv)

        def matvec(x):
            return (At @ self._t(x)).cpu().numpy()

        def precond(x):
            return (Mt @ self._t(x)).cpu().numpy()

        return matvec, precond


class OracleBackend:
    """TEST-ONLY backend over the CPU oracle (oracle/ is test infrastructure;
    this class exists so the orchestration logic is CPU-testable and must
    never be used outside tests)."""

    def __init__(self):
        import oracle
        self.oracle = oracle

    def stokeslet(self, r_src, f_src, r_trg, eta):
        return self.oracle.stokeslet(r_src, f_src, r_trg, eta)

    def stresslet_normal_density(self, r_src, normals, density, r_trg, eta):
        f_dl = 2.0 * eta * np.einsum("ni,nj->nij", normals, density).reshape(-1, 9)
        return self.oracle.stresslet(r_src, f_dl, r_trg, eta)

    def self_stokeslet_batch(self, pts, eta):
        return np.stack([self.oracle.oseen_tensor(p, eta) for p in pts])

    def batched_lu(self, A_batch):
        import scipy.linalg as scla
        lus = [scla.lu_factor(A) for A in A_batch]

        def solve(rhs):
            return np.stack([scla.lu_solve(lu, r) for lu, r in zip(lus, rhs)])

        return solve

    def shell_ops(self, A, M_inv):
        return (lambda x: A @ x), (lambda x: M_inv @ x)


class Shell:
    """Periphery state: nodes/normals (N, 3) + the dense operators
    (stresslet_plus_complementary, M_inv)."""

    def __init__(self, nodes, normals, A, M_inv):
        self.nodes = np.asarray(nodes, float)
        self.normals = np.asarray(normals, float)
        self.A = A
        self.M_inv = M_inv
        self.n_nodes = len(self.nodes)


class SystemFD:
    def __init__(self, fibers, eta, dt, shell=None, background_flow=None, backend=None):
        self.fibers = list(fibers)
        self.eta = float(eta)
        self.dt = float(dt)
        self.shell = shell
        self.background_flow = background_flow  # fn: (n,3) -> (n,3)
        self.backend = backend if backend is not None else HipBackend()
        self._uniform = all(f.n_nodes == self.fibers[0].n_nodes for f in self.fibers) \
            if self.fibers else True
        if shell is not None:
            self._shell_matvec, self._shell_precond = \
                self.backend.shell_ops(shell.A, shell.M_inv)

    # ---- layout helpers -------------------------------------------------
    @property
    def fiber_node_count(self):
        return sum(f.n_nodes for f in self.fibers)

    @property
    def fiber_sol_size(self):
        return sum(4 * f.n_nodes for f in self.fibers)

    @property
    def shell_sol_size(self):
        return 3 * self.shell.n_nodes if self.shell else 0

    def fiber_nodes(self):
        return np.concatenate([f.x.T for f in self.fibers], axis=0) \
            if self.fibers else np.zeros((0, 3))

    def all_nodes(self):
        parts = [self.fiber_nodes()]
        if self.shell:
            parts.append(self.shell.nodes)
        return np.concatenate(parts, axis=0)

    def _fiber_slices(self):
        out, off = [], 0
        for f in self.fibers:
            out.append((f, off, off + 4 * f.n_nodes))
            off += 4 * f.n_nodes
        return out

    def _fiber_node_slices(self):
        out, off = [], 0
        for f in self.fibers:
            out.append((f, off, off + f.n_nodes))
            off += f.n_nodes
        return out

    # ---- fiber container operations ------------------------------------
    def _fiber_flow(self, r_trg, fib_forces):
        """FiberContainerFiniteDifference::flow (f_c_fd.cpp:172-214):
        quadrature-weighted stokeslet of fiber forces at r_trg, minus the
        per-fiber self term on the fiber's own slice of the targets."""
        if not self.fibers:
            return np.zeros_like(r_trg)
        w = np.concatenate([f.quadrature_weights() for f in self.fibers])
        wf = fib_forces * w[:, None]
        vel = self.backend.stokeslet(self.fiber_nodes(), wf, r_trg, self.eta)
        for f, a, b in self._fiber_node_slices():
            # stokeslet_ and wf are point-major interleaved (xyz per node),
            # matching the reference's VectorMap flattening (f_c_fd.cpp:204-208)
            vel[a:b] -= (f.stokeslet @ wf[a:b].reshape(-1)).reshape(f.n_nodes, 3)
        return vel

    def _apply_fiber_force(self, x_fib):
        """force_operator per fiber -> (n_fib_nodes, 3) (f_c_fd.cpp:272-287)."""
        fw = np.zeros((self.fiber_node_count, 3))
        node_off = 0
        for f, a, b in self._fiber_slices():
            ff = f.force_operator @ x_fib[a:b]
            np_ = f.n_nodes
            for i in range(3):
                fw[node_off: node_off + np_, i] = ff[i * np_: (i + 1) * np_]
            node_off += np_
        return fw

    # ---- solver pipeline ------------------------------------------------
    def prep_state_for_solver(self):
        """system.cpp:396-459 (no bodies, no dynamic instability, no
        fiber-periphery repulsion)."""
        dt, eta = self.dt, self.eta
        for f in self.fibers:
            f.update_constants(eta)
            f.update_derivatives()
            f.update_linear_operator(dt, eta)
            f.update_force_operator()

        # self-stokeslets (fiber_finite_difference.cpp:56), batched on device
        if self.fibers:
            if self._uniform:
                pts = np.stack([f.x.T for f in self.fibers])
                G = self.backend.self_stokeslet_batch(pts, eta)
                for f, g in zip(self.fibers, G):
                    f.stokeslet = g
            else:
                for f in self.fibers:
                    f.stokeslet = self.backend.self_stokeslet_batch(
                        f.x.T[None], eta)[0]

        r_all = self.all_nodes()
        nf_nodes = self.fiber_node_count

        # motor force (generate_constant_force, f_c_fd.cpp:160-169)
        motor = np.zeros((nf_nodes, 3))
        for f, a, b in self._fiber_node_slices():
            motor[a:b] = (f.force_scale * f.xs).T

        # v_all: background flow only (no point/body sources here;
        # external/periphery-interaction forces are zero, so fc_->flow
        # contributes nothing — system.cpp:425)
        v_all = np.zeros_like(r_all)
        if self.background_flow is not None:
            v_all += self.background_flow(r_all)

        v_fib = v_all[:nf_nodes]
        for f, a, b in self._fiber_node_slices():
            f.update_RHS(dt, v_fib[a:b].T, motor[a:b].T)
            f.apply_bc_rectangular(dt, v_fib[a:b].T, None)

        # preconditioner: batched LU of the (BC-applied) fiber operators
        if self.fibers and self._uniform:
            A_batch = np.stack([f.A for f in self.fibers])
            self._fiber_lu_solve = self.backend.batched_lu(A_batch)
        elif self.fibers:
            solves = [self.backend.batched_lu(f.A[None]) for f in self.fibers]
            self._fiber_lu_solve = None
            self._fiber_lu_solves = solves

        rhs_parts = [f.RHS for f in self.fibers]
        if self.shell:
            v_shell = v_all[nf_nodes:]
            rhs_parts.append(-v_shell.reshape(-1))  # update_RHS, periphery.cpp:86
        self.RHS = np.concatenate(rhs_parts) if rhs_parts else np.zeros(0)
        return self.RHS

    def apply_matvec(self, x):
        """system.cpp:269-324 (fibers + shell)."""
        nf_nodes = self.fiber_node_count
        x_fib = x[: self.fiber_sol_size]
        x_shell = x[self.fiber_sol_size:]
        r_all = self.all_nodes()

        fw = self._apply_fiber_force(x_fib)
        v_all = self._fiber_flow(r_all, fw)

        if self.shell:
            dens = x_shell.reshape(-1, 3)
            v_shell2fib = self.backend.stresslet_normal_density(
                self.shell.nodes, self.shell.normals, dens,
                r_all[:nf_nodes], self.eta) if nf_nodes else np.zeros((0, 3))
            v_all[:nf_nodes] += v_shell2fib

        res = np.zeros_like(x)
        v_fib = v_all[:nf_nodes]
        for (f, a, b), (_, na, nb) in zip(self._fiber_slices(),
                                          self._fiber_node_slices()):
            res[a:b] = f.matvec(x_fib[a:b], v_fib[na:nb].T, None)
        if self.shell:
            v_shell = v_all[nf_nodes:]
            res[self.fiber_sol_size:] = \
                self._shell_matvec(x_shell) + v_shell.reshape(-1)
        return res

    def apply_preconditioner(self, x):
        """system.cpp:248-263."""
        res = np.zeros_like(x)
        x_fib = x[: self.fiber_sol_size]
        if self.fibers:
            if self._uniform:
                m = 4 * self.fibers[0].n_nodes
                sol = self._fiber_lu_solve(x_fib.reshape(len(self.fibers), m))
                res[: self.fiber_sol_size] = sol.reshape(-1)
            else:
                for (f, a, b), solve in zip(self._fiber_slices(),
                                            self._fiber_lu_solves):
                    res[a:b] = solve(x_fib[a:b][None])[0]
        if self.shell:
            res[self.fiber_sol_size:] = self._shell_precond(x[self.fiber_sol_size:])
        return res

    def solve(self, tol=1e-10, maxiter=200, restart=None):
        """system.cpp:464-478 via the engine GMRES (right-preconditioned,
        ICGS — solver_hydro.cpp:64-87)."""
        import torch
        from .gmres import gmres

        rhs = self.prep_state_for_solver()
        b = torch.from_numpy(rhs)
        mv = lambda v: torch.from_numpy(self.apply_matvec(v.numpy()))
        pc = lambda v: torch.from_numpy(self.apply_preconditioner(v.numpy()))
        if restart is None:
            restart = min(200, maxiter)
        x, info = gmres(mv, b, precond=pc, tol=tol, maxiter=maxiter, restart=restart)
        self.solution = x.numpy()
        return info

    def step(self, tol=1e-10, maxiter=200):
        """system.cpp:482-493 (no bodies): solve then adopt positions."""
        info = self.solve(tol=tol, maxiter=maxiter)
        for f, a, b in self._fiber_slices():
            f.step(self.solution[a:b])
        return info
