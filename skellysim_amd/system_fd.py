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


class HipBackend:
    """Product backend: pair kernels + batched algebra on the MI355X."""

    def __init__(self, device="cuda:0"):
        import torch
        self.torch = torch
        self.dev = torch.device(device)

    def _t(self, a):
        if self.torch.is_tensor(a):
            return a.to(self.dev)
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

    def rotlet(self, centers, torques, r_trg, eta):
        from .evaluator import rotlet_device
        u = rotlet_device(self._t(centers), self._t(r_trg), self._t(torques), eta)
        self.torch.cuda.synchronize()
        return u.cpu().numpy()

    def oseen_contract(self, r_src, r_trg, density, eta):
        from .evaluator import oseen_contract_device
        u = oseen_contract_device(self._t(r_src), self._t(r_trg),
                                  self._t(density), eta)
        self.torch.cuda.synchronize()
        return u.cpu().numpy()

    def stresslet_times_normal(self, nodes, normals, eta):
        """Dense (3n, 3n) operator (kernels.cpp:264-287; eta-independent
        factor) — the M block of the body preconditioner."""
        from .evaluator import stresslet_times_normal_device
        S = stresslet_times_normal_device(self._t(nodes), self._t(normals))
        self.torch.cuda.synchronize()
        return S.cpu().numpy()

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
    def __init__(self, fibers, eta, dt, shell=None, background_flow=None, backend=None,
                 periphery_interaction=None, bodies=None, periphery_shape=None,
                 periphery_binding=None, dynamic_instability=None, seed=130319):
        self.fibers = list(fibers)
        self.eta = float(eta)
        self.dt = float(dt)
        self.shell = shell
        self.bodies = list(bodies) if bodies is not None else []
        # periphery_shape: dict(kind=, radius=|abc=) for collision/binding
        # geometry; periphery_binding: dict(active, polar_angle_start/_end,
        # threshold) (skelly_config.py:352-372); dynamic_instability: dict
        # with skellysim_amd.instability.DEFAULTS keys (0 n_nodes = off)
        self.periphery_shape = periphery_shape
        self.periphery_binding = periphery_binding
        self.dynamic_instability = dynamic_instability
        # fiber-fiber steric repulsion (ENGINE EXTENSION, default off):
        # dict(f_0=, l_0=, d_max=) — see _fiber_fiber_repulsion
        self.steric_interaction = None
        self.rng = np.random.default_rng(seed)  # params.seed default 130319
        # point/background sources (sources.py); simulation clock for their
        # lifetimes (properties.time — advanced by run())
        self.point_sources = None
        self.background_source = None
        self.time = 0.0
        # motor forces held at zero until the clock passes this
        # (params.implicit_motor_activation_delay, system.cpp:417-419)
        self.motor_activation_delay = 0.0
        self.background_flow = background_flow  # fn: (n,3) -> (n,3)
        # steric fiber-periphery repulsion (system.cpp:421, params.cpp:18):
        # dict(kind="sphere"|"ellipsoid", f_0=, l_0=, radius=|abc=) or None
        self.periphery_interaction = periphery_interaction
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

    @property
    def body_sol_size(self):
        return sum(b.solution_size for b in self.bodies)

    @property
    def body_node_count(self):
        return sum(b.n_nodes for b in self.bodies)

    def fiber_nodes(self):
        return np.concatenate([f.x.T for f in self.fibers], axis=0) \
            if self.fibers else np.zeros((0, 3))

    def body_nodes(self):
        return np.concatenate([b.nodes for b in self.bodies], axis=0) \
            if self.bodies else np.zeros((0, 3))

    def body_normals(self):
        return np.concatenate([b.normals for b in self.bodies], axis=0) \
            if self.bodies else np.zeros((0, 3))

    def all_nodes(self):
        """Node order [fibers | shell | bodies] (system.cpp:408-418)."""
        parts = [self.fiber_nodes()]
        if self.shell:
            parts.append(self.shell.nodes)
        if self.bodies:
            parts.append(self.body_nodes())
        return np.concatenate(parts, axis=0)

    def _body_sol_slices(self):
        out, off = [], self.fiber_sol_size + self.shell_sol_size
        for b in self.bodies:
            out.append((b, off, off + b.solution_size))
            off += b.solution_size
        return out

    def _body_node_slices(self):
        """Slices into the [fibers | shell | bodies] node array."""
        out, off = [], self.fiber_node_count + \
            (self.shell.n_nodes if self.shell else 0)
        for b in self.bodies:
            out.append((b, off, off + b.n_nodes))
            off += b.n_nodes
        return out

    def _body_flow(self, r_trg, x_bodies, forces_torques):
        """BodyContainer::flow_spherical (body_container.cpp:269-337):
        double layer of the bodies' surface densities + center Stokeslet of
        the forces + center rotlet of the torques. x_bodies: the body block
        of a solution vector (densities per body, point-major, then U/w);
        forces_torques: (n_bodies, 6) from link conditions (matvec) or the
        external forces (prep)."""
        if not self.bodies:
            return np.zeros_like(r_trg)
        dens_parts, off = [], 0
        for b in self.bodies:
            dens_parts.append(
                x_bodies[off: off + 3 * b.n_nodes].reshape(b.n_nodes, 3))
            off += b.solution_size
        dens = np.concatenate(dens_parts)
        v = self.backend.stresslet_normal_density(
            self.body_nodes(), self.body_normals(), dens, r_trg, self.eta)
        centers = np.stack([b.position for b in self.bodies])
        v += self.backend.stokeslet(centers, forces_torques[:, 0:3], r_trg,
                                    self.eta)
        v += self.backend.rotlet(centers, forces_torques[:, 3:6], r_trg,
                                 self.eta)
        return v

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
        if self.fibers[0].stokeslet is None:
            if self._uniform:
                pts = np.stack([f.x.T for f in self.fibers])
                G = self.backend.self_stokeslet_batch(pts, self.eta)
                for f, g in zip(self.fibers, G):
                    f.stokeslet = g
            else:
                for f in self.fibers:
                    f.stokeslet = self.backend.self_stokeslet_batch(
                        f.x.T[None], self.eta)[0]
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
        """system.cpp:396-459."""
        dt, eta = self.dt, self.eta
        # dynamic instability runs first (system.cpp:403) and can change
        # the fiber population
        if self.dynamic_instability:
            from .instability import dynamic_instability
            dynamic_instability(self, self.dynamic_instability, self.rng)
            self._uniform = all(f.n_nodes == self.fibers[0].n_nodes
                                for f in self.fibers) if self.fibers else True
        for f in self.fibers:
            f.update_constants(eta)
            f.update_derivatives()
        # periphery binding re-evaluates the plus-end BCs every prep
        # (fc_->update_boundary_conditions, system.cpp:453)
        if self.periphery_binding and self.periphery_binding.get("active"):
            for f in self.fibers:
                f.update_boundary_conditions(self.periphery_shape,
                                             self.periphery_binding)
        if not (self._uniform and self.fibers):
            for f in self.fibers:
                f.update_linear_operator(dt, eta)
                f.update_force_operator()

        # self-stokeslets (fiber_finite_difference.cpp:56) are built lazily:
        # the device-resident solve path builds its own resident stack, and
        # the host path materializes them on first _fiber_flow use
        for f in self.fibers:
            f.stokeslet = None

        r_all = self.all_nodes()
        nf_nodes = self.fiber_node_count

        # motor force (generate_constant_force, f_c_fd.cpp:160-169); held
        # at zero until the activation delay passes (system.cpp:417-419)
        motor = np.zeros((nf_nodes, 3))
        if not (self.motor_activation_delay > self.time):
            for f, a, b in self._fiber_node_slices():
                motor[a:b] = (f.force_scale * f.xs).T

        # fiber-periphery steric repulsion (fc_->periphery_force,
        # system.cpp:421; per-fiber force periphery.cpp:140-162,232-263)
        ext = np.zeros((nf_nodes, 3))
        if self.periphery_interaction is not None:
            pi = self.periphery_interaction
            for f, a, b in self._fiber_node_slices():
                ext[a:b] = f.periphery_repulsion(**pi).T
        # fiber-fiber steric repulsion (ENGINE EXTENSION, opt-in; see
        # _fiber_fiber_repulsion)
        if self.steric_interaction is not None and nf_nodes:
            ext += self._fiber_fiber_repulsion(**self.steric_interaction)

        # v_all: flow induced by the external (periphery) forces
        # (fc_->flow(r_all, external_force_fibers), system.cpp:425)
        # + background flow; motor forces enter only the RHS
        v_all = np.zeros_like(r_all)
        if nf_nodes and (self.periphery_interaction is not None
                         or self.steric_interaction is not None):
            v_all += self._fiber_flow(r_all, ext)
        if self.background_flow is not None:
            v_all += self.background_flow(r_all)
        # point + background sources (psc_.flow / bs_.flow, system.cpp:445-446)
        if self.point_sources is not None:
            v_all += self.point_sources.flow(r_all, eta, self.time,
                                             self.backend)
        if self.background_source is not None and \
                self.background_source.is_active():
            v_all += self.background_source.flow(r_all, eta)

        # body caches + external body force/torque flow (system.cpp:427-443)
        if self.bodies:
            for b in self.bodies:
                b.update_cache(eta, self.backend)
            ext_ft = np.stack([np.concatenate([b.external_force_at(self.time),
                                               b.external_torque])
                               for b in self.bodies])
            if np.any(ext_ft):
                # global-sized zero densities (body_sol_size is rank-local
                # in the distributed subclass; the flow's sources are the
                # replicated global bodies)
                v_all += self._body_flow(
                    r_all, np.zeros(sum(b.solution_size
                                        for b in self.bodies)), ext_ft)

        motor = motor + ext  # total_force_fibers (system.cpp:450)
        v_fib = v_all[:nf_nodes]
        self._assembled_dev = None
        if self._uniform and self.fibers:
            # batched assembly (operator + RHS + BCs + force operator) —
            # numerically identical to the per-fiber loop (fiber_batch.py)
            from .fiber_batch import assemble_uniform, assemble_uniform_t
            n = self.fibers[0].n_nodes
            nf = len(self.fibers)
            flow_b = v_fib.reshape(nf, n, 3).transpose(0, 2, 1)
            motor_b = motor.reshape(nf, n, 3).transpose(0, 2, 1)
            ext_b = ext.reshape(nf, n, 3).transpose(0, 2, 1) \
                if self.periphery_interaction is not None else None
            dev = getattr(self.backend, "dev", None)
            if dev is not None:
                # assemble on device (the dominant prep cost); materialize
                # host copies so the host matvec/precond paths stay valid,
                # and keep the resident tensors for _build_device_operators
                A_t, RHS_t, F_t = assemble_uniform_t(
                    self.fibers, dt, eta, flow=flow_b, f_external=motor_b,
                    bc_force=ext_b, device=dev)
                A_h, R_h, F_h = (A_t.cpu().numpy(), RHS_t.cpu().numpy(),
                                 F_t.cpu().numpy())
                for k, f in enumerate(self.fibers):
                    f.adopt_operator(A_h[k], R_h[k])
                    f.force_operator = F_h[k]
                self._assembled_dev = (A_t, F_t)
            else:
                assemble_uniform(self.fibers, dt, eta, flow=flow_b,
                                 f_external=motor_b, bc_force=ext_b)
        else:
            for f, a, b in self._fiber_node_slices():
                f.update_RHS(dt, v_fib[a:b].T, motor[a:b].T)
                f.apply_bc_rectangular(dt, v_fib[a:b].T, ext[a:b].T
                                       if self.periphery_interaction is not None
                                       else None)

        # preconditioner: batched LU of the (BC-applied) fiber operators
        # (lazy — the device-resident solve path factors its own resident
        # copy in _build_device_operators)
        if self.fibers and self._uniform:
            self._fiber_lu_solve = None
        elif self.fibers:
            solves = [self.backend.batched_lu(f.A[None]) for f in self.fibers]
            self._fiber_lu_solve = None
            self._fiber_lu_solves = solves

        rhs_parts = [f.RHS for f in self.fibers]
        sh_nodes = self.shell.n_nodes if self.shell else 0
        if self.shell:
            v_shell = v_all[nf_nodes: nf_nodes + sh_nodes]
            rhs_parts.append(-v_shell.reshape(-1))  # update_RHS, periphery.cpp:86
        for b, a, bb in self._body_node_slices():
            rhs_parts.append(b.update_RHS(v_all[a:bb]))
        self.RHS = np.concatenate(rhs_parts) if rhs_parts else np.zeros(0)
        return self.RHS

    def apply_matvec(self, x):
        """system.cpp:269-324 (fibers + shell + bodies)."""
        nf_nodes = self.fiber_node_count
        sh_nodes = self.shell.n_nodes if self.shell else 0
        x_fib = x[: self.fiber_sol_size]
        x_shell = x[self.fiber_sol_size: self.fiber_sol_size + self.shell_sol_size]
        x_bodies = x[self.fiber_sol_size + self.shell_sol_size:]
        r_all = self.all_nodes()

        fw = self._apply_fiber_force(x_fib)
        v_all = self._fiber_flow(r_all, fw)

        if self.shell:
            # shell double layer flows to fibers AND bodies, not to itself
            # (system.cpp:302-305,314-316)
            dens = x_shell.reshape(-1, 3)
            trg = np.concatenate([r_all[:nf_nodes],
                                  r_all[nf_nodes + sh_nodes:]])
            if len(trg):
                v = self.backend.stresslet_normal_density(
                    self.shell.nodes, self.shell.normals, dens, trg, self.eta)
                v_all[:nf_nodes] += v[:nf_nodes]
                v_all[nf_nodes + sh_nodes:] += v[nf_nodes:]

        vel_on_fiber = None
        if self.bodies:
            # fiber<->body link conditions + body flow (system.cpp:308-317)
            from .body import calculate_link_conditions
            body_vels = np.stack([
                x[a + 3 * b.n_nodes: bb] for b, a, bb in self._body_sol_slices()])
            vel_on_fiber, body_ft = calculate_link_conditions(
                self.fibers, x_fib, body_vels, self.bodies)
            v_all += self._body_flow(r_all, x_bodies, body_ft)

        res = np.zeros_like(x)
        v_fib = v_all[:nf_nodes]
        for i, ((f, a, b), (_, na, nb)) in enumerate(
                zip(self._fiber_slices(), self._fiber_node_slices())):
            vb = vel_on_fiber[i] if vel_on_fiber is not None else None
            res[a:b] = f.matvec(x_fib[a:b], v_fib[na:nb].T, vb)
        if self.shell:
            v_shell = v_all[nf_nodes: nf_nodes + sh_nodes]
            res[self.fiber_sol_size: self.fiber_sol_size + self.shell_sol_size] = \
                self._shell_matvec(x_shell) + v_shell.reshape(-1)
        for (b, a, bb), (_, na, nb) in zip(self._body_sol_slices(),
                                           self._body_node_slices()):
            res[a:bb] = b.matvec(v_all[na:nb], x[a:bb])
        return res

    def apply_preconditioner(self, x):
        """system.cpp:248-263."""
        res = np.zeros_like(x)
        x_fib = x[: self.fiber_sol_size]
        if self.fibers:
            if self._uniform:
                if self._fiber_lu_solve is None:
                    A_batch = np.stack([f.A for f in self.fibers])
                    self._fiber_lu_solve = self.backend.batched_lu(A_batch)
                m = 4 * self.fibers[0].n_nodes
                sol = self._fiber_lu_solve(x_fib.reshape(len(self.fibers), m))
                res[: self.fiber_sol_size] = sol.reshape(-1)
            else:
                for (f, a, b), solve in zip(self._fiber_slices(),
                                            self._fiber_lu_solves):
                    res[a:b] = solve(x_fib[a:b][None])[0]
        if self.shell:
            sh = slice(self.fiber_sol_size,
                       self.fiber_sol_size + self.shell_sol_size)
            res[sh] = self._shell_precond(x[sh])
        for b, a, bb in self._body_sol_slices():
            res[a:bb] = b.apply_preconditioner(x[a:bb])
        return res

    # ---- device-resident iteration (uniform fibers + HipBackend) --------
    def _build_device_operators(self):
        """Stack the per-fiber operators in HBM so the entire GMRES iteration
        (matvec + preconditioner) runs on device with zero host traffic —
        SURVEY.md §8f row 3 in full."""
        import torch
        dev = self.backend.dev
        T = lambda a: torch.from_numpy(np.ascontiguousarray(a)).to(dev)

        nf = len(self.fibers)
        f0 = self.fibers[0]
        n = f0.n_nodes
        self._dev = dict(nf=nf, n=n)
        d = self._dev
        if getattr(self, "_assembled_dev", None) is not None:
            d["A"], d["F"] = self._assembled_dev  # already resident from prep
        else:
            d["A"] = T(np.stack([f.A for f in self.fibers]))
            d["F"] = T(np.stack([f.force_operator for f in self.fibers]))
        # self-stokeslets built directly on device (kernels.cpp:146-195)
        from .evaluator import oseen_tensor_batched_device
        pts = T(np.stack([f.x.T for f in self.fibers]))
        d["G"] = oseen_tensor_batched_device(pts, eta=self.eta)
        d["xs"] = T(np.stack([f.xs for f in self.fibers]))          # (nf, 3, n)
        d["w"] = T(np.concatenate([f.quadrature_weights() for f in self.fibers]))
        d["P_ds"] = T(f0.mats["P_downsample_bc"])                    # (4n-14, 4n)
        d["D1preT"] = T((f0.mats["D_1_0"] * (2.0 / f0.length_prev)).T)
        d["r_fib"] = T(self.fiber_nodes())
        d["r_all"] = T(self.all_nodes())
        d["plus_vel"] = T(np.array(
            [1.0 if f.bc_plus[0] == "Velocity" else 0.0 for f in self.fibers]))
        if self.shell:
            d["sh_nodes"] = T(self.shell.nodes)
            d["sh_normals"] = T(self.shell.normals)
            d["sh_A"] = self.shell.A if torch.is_tensor(self.shell.A) \
                else T(self.shell.A)
            d["sh_Minv"] = self.shell.M_inv if torch.is_tensor(self.shell.M_inv) \
                else T(self.shell.M_inv)
            # transposed copies: rocBLAS dgemv's reduction (trans) path is
            # ~1.5x the axpy path on these square operators (measured
            # 0.96 vs 1.43 ms at 24576^2, tools/prof_iter.py), so store
            # X^T contiguous and apply as mv(X_T.t(), v). HBM cost is one
            # extra copy of each operator — trivial in 288 GB.
            if getattr(self.shell, "_A_T", None) is None:
                self.shell._A_T = d["sh_A"].t().contiguous()
                self.shell._Minv_T = d["sh_Minv"].t().contiguous()
            d["sh_A_T"] = self.shell._A_T
            d["sh_Minv_T"] = self.shell._Minv_T
        from .batched import BatchedLU
        d["lu"] = BatchedLU(d["A"])

        if self.bodies:
            # per-body blocks + fiber<->body link statics (body.py)
            blocks = []
            from .batched import BatchedLU
            for b in self.bodies:
                # BatchedLU(batch of 1) so the per-iteration solves take the
                # trsm path — magma's lu_solve corrupts under deep stream
                # queues (profiles/cadence_matrix_r02.md) and the device
                # preconditioner runs at sync cadence 8 by default
                blocks.append(dict(n=b.n_nodes,
                                   e=T(np.stack(b.e_sub)),     # (3, n, 3)
                                   w=T(b.weights), K=T(b.K),
                                   lu=BatchedLU(T(b._A_dense)[None])))
            d["bodies"] = blocks
            d["body_nodes"] = T(self.body_nodes())
            d["body_normals"] = T(self.body_normals())
            d["centers"] = T(np.stack([b.position for b in self.bodies]))
            att = [(i, f.binding_site) for i, f in enumerate(self.fibers)
                   if f.binding_site[0] >= 0]
            d["att_idx"] = torch.tensor([i for i, _ in att], dtype=torch.long,
                                        device=dev)
            d["att_body"] = torch.tensor([bs[0] for _, bs in att],
                                         dtype=torch.long, device=dev)
            site = np.stack([self.bodies[bs[0]].nucleation_sites[bs[1]]
                             - self.bodies[bs[0]].position
                             for _, bs in att]) if att else np.zeros((0, 3))
            d["att_site"] = T(site)
            d["att_site_hat"] = T(site / np.linalg.norm(site, axis=1,
                                                        keepdims=True)) \
                if len(att) else T(site)
            d["d2c0"] = T(f0.mats["D_2_0"][:, 0])
            d["d3c0"] = T(f0.mats["D_3_0"][:, 0])
            d["E_fib"] = T(np.array([f.bending_rigidity for f in self.fibers]))
            d["s2"] = T(np.array([(2.0 / f.length) ** 2 for f in self.fibers]))
            d["s3"] = T(np.array([(2.0 / f.length) ** 3 for f in self.fibers]))

    def _apply_matvec_device(self, x):
        """apply_matvec entirely on device (torch fp64 CUDA vector in/out);
        covers fibers + shell + bodies (node order [fib | shell | body],
        system.cpp:269-324)."""
        import torch
        from .evaluator import stokeslet_device, stresslet_device
        d = self._dev
        nf, n = d["nf"], d["n"]
        eta = self.eta
        nf_nodes = nf * n
        sh_size = self.shell_sol_size
        sh_nodes_n = sh_size // 3
        x_fib = x[: 4 * nf_nodes].reshape(nf, 4 * n)
        x_shell = x[4 * nf_nodes: 4 * nf_nodes + sh_size]
        x_bodies = x[4 * nf_nodes + sh_size:]

        # forces: F @ x (component-major) -> node-major (nf*n, 3)
        fw = torch.bmm(d["F"], x_fib.unsqueeze(-1)).squeeze(-1)     # (nf, 3n)
        fw_nodes = fw.reshape(nf, 3, n).permute(0, 2, 1).reshape(nf_nodes, 3)
        wf = (fw_nodes * d["w"][:, None]).contiguous()

        v_all = stokeslet_device(d["r_fib"], wf, d["r_all"], eta)
        # per-fiber self subtraction (point-major, f_c_fd.cpp:203-210)
        wf_pm = wf.reshape(nf, 3 * n, 1)
        corr = torch.bmm(d["G"], wf_pm).reshape(nf_nodes, 3)
        v_all[:nf_nodes] -= corr

        if self.shell:
            # shell double layer flows to fibers and bodies, not to itself
            dens = x_shell.reshape(-1, 3)
            f_dl = 2.0 * eta * torch.einsum("ni,nj->nij", d["sh_normals"],
                                            dens).reshape(-1, 9).contiguous()
            v_all[:nf_nodes] += stresslet_device(d["sh_nodes"], f_dl,
                                                 d["r_fib"], eta)
            if self.bodies:
                v_all[nf_nodes + sh_nodes_n:] += stresslet_device(
                    d["sh_nodes"], f_dl, d["body_nodes"], eta)

        vel7 = None
        if self.bodies:
            from .evaluator import rotlet_device
            nb = len(self.bodies)
            # per-body [U, w] slices of the body block
            body_vels = torch.zeros((nb, 6), dtype=x.dtype, device=x.device)
            dens_parts, off = [], 0
            for bi, bd in enumerate(d["bodies"]):
                nn = bd["n"]
                dens_parts.append(x_bodies[off: off + 3 * nn].reshape(nn, 3))
                body_vels[bi] = x_bodies[off + 3 * nn: off + 3 * nn + 6]
                off += 3 * nn + 6
            bdens = torch.cat(dens_parts)

            # link conditions (body_container.cpp:171-268) in torch
            body_ft = torch.zeros((nb, 6), dtype=x.dtype, device=x.device)
            ai = d["att_idx"]
            if len(ai):
                xa = x_fib[ai]
                x_new = xa[:, : 3 * n].reshape(-1, 3, n)
                T0 = xa[:, 3 * n]
                xss0 = torch.einsum("ain,n->ai", x_new, d["d2c0"]) \
                    * d["s2"][ai][:, None]
                xsss0 = torch.einsum("ain,n->ai", x_new, d["d3c0"]) \
                    * d["s3"][ai][:, None]
                xs0 = d["xs"][ai][:, :, 0]
                E = d["E_fib"][ai][:, None]
                site = d["att_site"]
                F_b = -E * xsss0 + xs0 * T0[:, None]
                L_b = (-E * torch.cross(site, xsss0, dim=1)
                       + torch.cross(site, xs0, dim=1) * T0[:, None]
                       + E * torch.cross(xs0, xss0, dim=1))
                body_ft.index_add_(0, d["att_body"],
                                   torch.cat([F_b, L_b], dim=1))
                U_att = body_vels[d["att_body"], 0:3]
                w_att = body_vels[d["att_body"], 3:6]
                v_f = -U_att - torch.cross(w_att, site, dim=1)
                tc = -(xs0 * U_att).sum(dim=1) \
                    + (torch.cross(xs0, site, dim=1) * w_att).sum(dim=1)
                w_f = torch.cross(d["att_site_hat"], w_att, dim=1)
                vel7 = torch.zeros((nf, 7), dtype=x.dtype, device=x.device)
                vel7[ai] = torch.cat([v_f, tc[:, None], w_f], dim=1)

            # body flow: double layer + center stokeslet/rotlet of link F/T
            f_dl_b = 2.0 * eta * torch.einsum(
                "ni,nj->nij", d["body_normals"], bdens).reshape(-1, 9).contiguous()
            v_all += stresslet_device(d["body_nodes"], f_dl_b, d["r_all"], eta)
            v_all += stokeslet_device(d["centers"],
                                      body_ft[:, 0:3].contiguous(),
                                      d["r_all"], eta)
            v_all += rotlet_device(d["centers"], d["r_all"],
                                   body_ft[:, 3:6].contiguous(), eta)

        res = torch.empty_like(x)
        res[: 4 * nf_nodes] = self._fiber_block_device(
            x_fib, v_all[:nf_nodes], vel7).reshape(-1)
        if self.shell:
            v_shell = v_all[nf_nodes: nf_nodes + sh_nodes_n].reshape(-1)
            res[4 * nf_nodes: 4 * nf_nodes + sh_size] = \
                torch.addmv(v_shell, d["sh_A_T"].t(), x_shell)
        if self.bodies:
            off = 0
            off_node = nf_nodes + sh_nodes_n
            base = 4 * nf_nodes + sh_size
            for bd in d["bodies"]:
                nn = bd["n"]
                xb = x_bodies[off: off + 3 * nn + 6]
                dloc = xb[: 3 * nn].reshape(nn, 3)
                U = xb[3 * nn:]
                sub = (dloc[:, 0:1] * bd["e"][0] + dloc[:, 1:2] * bd["e"][1]
                       + dloc[:, 2:3] * bd["e"][2]) / bd["w"][:, None]
                v_nodes = v_all[off_node: off_node + nn]
                res[base + off: base + off + 3 * nn] = \
                    (-sub + v_nodes).reshape(-1) - bd["K"] @ U
                res[base + off + 3 * nn: base + off + 3 * nn + 6] = \
                    -bd["K"].T @ xb[: 3 * nn] + U
                off += 3 * nn + 6
                off_node += nn
        return res

    def _fiber_block_device(self, x_fib, v_fib_nodes, vel7=None):
        """Per-fiber operator block: A x - vT_in + the two BC velocity
        corrections (fiber_fd.matvec, fiber_finite_difference.cpp:278-315),
        plus the y_BC link-condition rows when vel7 (nf, 7) is given
        (fiber matvec's v_boundary, f_c_fd.cpp:226).
        x_fib (nf, 4n), v_fib_nodes (nf*n, 3) node-major -> (nf, 4n)."""
        import torch
        d = self._dev
        nf, n = d["nf"], d["n"]
        v_fib = v_fib_nodes.reshape(nf, n, 3).permute(0, 2, 1)       # (nf, 3, n)
        vT = torch.zeros((nf, 4 * n), dtype=x_fib.dtype, device=x_fib.device)
        vT[:, : 3 * n] = v_fib.reshape(nf, 3 * n)
        tens = torch.einsum("ts,fis->fit", d["D1preT"],
                            d["xs"] * v_fib).sum(dim=1)
        vT[:, 3 * n:] = tens
        vT_in = torch.zeros_like(vT)
        vT_in[:, : 4 * n - 14] = vT @ d["P_ds"].T
        res_fib = torch.bmm(d["A"], x_fib.unsqueeze(-1)).squeeze(-1) - vT_in
        bc_start = 4 * n - 14
        res_fib[:, bc_start + 3] += (v_fib[:, :, 0] * d["xs"][:, :, 0]).sum(dim=1)
        res_fib[:, bc_start + 10] += d["plus_vel"] * \
            (v_fib[:, :, -1] * d["xs"][:, :, -1]).sum(dim=1)
        if vel7 is not None:
            res_fib[:, bc_start: bc_start + 7] += vel7
        return res_fib

    def _apply_precond_device(self, x):
        import torch
        d = self._dev
        nf, n = d["nf"], d["n"]
        sh_size = self.shell_sol_size
        res = torch.empty_like(x)
        res[: 4 * nf * n] = d["lu"].solve(x[: 4 * nf * n].reshape(nf, 4 * n)).reshape(-1)
        if self.shell:
            res[4 * nf * n: 4 * nf * n + sh_size] = torch.mv(
                d["sh_Minv_T"].t(), x[4 * nf * n: 4 * nf * n + sh_size])
        if self.bodies:
            off = 4 * nf * n + sh_size
            for bd in d["bodies"]:
                m = 3 * bd["n"] + 6
                res[off: off + m] = bd["lu"].solve(
                    x[off: off + m].unsqueeze(0)).squeeze(0)
                off += m
        return res

    def solve(self, tol=1e-10, maxiter=200, restart=None, device_mode=None,
              warm_start=None):
        """system.cpp:464-478 via the engine GMRES (right-preconditioned,
        ICGS — solver_hydro.cpp:64-87). With uniform fibers on the HIP
        backend the whole iteration runs device-resident, INCLUDING body
        blocks (torch link conditions + body flow + body rows — default
        since the round-2 GPU validation, tests/test_gpu_body.py::
        test_device_resident_bodies_match_host_path). device_mode=False
        forces the host matvec loop.

        warm_start=True seeds GMRES with the PREVIOUS timestep's solution
        (the state changes O(dt) per step, so the initial residual starts
        small) — an engine capability beyond the reference, which
        constructs a zero-initialized Tpetra X_ every solve
        (include/solver.hpp:25). Default off (reference semantics;
        SKELLY_WARM_START=1 flips it): the converged answer agrees to the
        GMRES tolerance either way, but iterate-level trajectories differ,
        so the reference-pinned regression tests keep cold starts."""
        import torch
        from .gmres import gmres

        rhs = self.prep_state_for_solver()
        if restart is None:
            restart = min(200, maxiter)
        if warm_start is None:
            import os
            warm_start = os.environ.get("SKELLY_WARM_START", "0") == "1"
        prev = getattr(self, "solution", None)
        x0_np = prev if (warm_start and prev is not None
                         and prev.shape == rhs.shape) else None

        if device_mode is None:
            device_mode = (bool(self.fibers) and self._uniform
                           and isinstance(self.backend, HipBackend))
        if device_mode:
            self._build_device_operators()
            b = self.backend._t(rhs)
            x0 = self.backend._t(x0_np) if x0_np is not None else None
            x, info = gmres(self._apply_matvec_device, b,
                            precond=self._apply_precond_device,
                            tol=tol, maxiter=maxiter, restart=restart, x0=x0)
            if b.is_cuda:
                self.backend.torch.cuda.synchronize()
            self.solution = x.cpu().numpy()
            return info

        b = torch.from_numpy(rhs)
        x0 = torch.from_numpy(x0_np) if x0_np is not None else None
        mv = lambda v: torch.from_numpy(self.apply_matvec(v.numpy()))
        pc = lambda v: torch.from_numpy(self.apply_preconditioner(v.numpy()))
        x, info = gmres(mv, b, precond=pc, tol=tol, maxiter=maxiter,
                        restart=restart, x0=x0)
        self.solution = x.numpy()
        return info

    def step(self, tol=1e-10, maxiter=200, restart=None):
        """system.cpp:482-493: solve then adopt positions, then repin
        body-attached fibers to their (moved) nucleation sites
        (FiberContainerFiniteDifference::repin_to_bodies,
        fiber_container_finite_difference.cpp:308-316 via system.cpp:488)."""
        info = self.solve(tol=tol, maxiter=maxiter, restart=restart)
        for f, a, b in self._fiber_slices():
            f.step(self.solution[a:b])
        for b, a, bb in self._body_sol_slices():
            b.step(self.dt, self.solution[a:bb])
        self.repin_to_bodies()
        return info

    def _fiber_fiber_repulsion(self, f_0=20.0, l_0=0.05, d_max=None,
                               fiber_radius=0.0125):
        """Pairwise fiber-fiber steric repulsion — an ENGINE EXTENSION,
        default OFF (set system.steric_interaction = dict(f_0=, l_0=,
        d_max=)). The reference implements steric interaction only
        against the periphery (periphery.cpp:140-162); at its own dense
        oocyte packing (examples/oocyte, ~0.1 fiber spacing) near-contact
        fiber pairs develop quadrature-singular velocities that reject
        timesteps at any dt (profiles/oocyte_r02.md). This applies the
        SAME exponential force law pairwise between nodes of different
        fibers: f = f_0 * exp(-gap/l_0) * unit(dx), equal and opposite,
        gap = center distance - 2*fiber_radius (clamped >= 0), cutoff
        d_max (default 5*l_0 + 2*fiber_radius)."""
        from scipy.spatial import cKDTree
        import torch.distributed as dist
        if dist.is_available() and dist.is_initialized() \
                and dist.get_world_size() > 1:
            raise NotImplementedError(
                "fiber-fiber steric repulsion needs a global neighbor "
                "search; not wired into the distributed prep yet")
        pts = self.fiber_nodes()
        out = np.zeros_like(pts)
        if d_max is None:
            d_max = 5.0 * l_0 + 2.0 * fiber_radius
        fid = np.concatenate([np.full(f.n_nodes, i)
                              for i, f in enumerate(self.fibers)])
        pairs = cKDTree(pts).query_pairs(r=d_max, output_type="ndarray")
        if len(pairs) == 0:
            return out
        i, j = pairs[:, 0], pairs[:, 1]
        keep = fid[i] != fid[j]
        i, j = i[keep], j[keep]
        if len(i) == 0:
            return out
        dx = pts[i] - pts[j]
        d = np.linalg.norm(dx, axis=1)
        pos = d > 0
        i, j, dx, d = i[pos], j[pos], dx[pos], d[pos]
        gap = np.maximum(d - 2.0 * fiber_radius, 0.0)
        mag = f_0 * np.exp(-gap / l_0) / d
        fvec = mag[:, None] * dx
        np.add.at(out, i, fvec)
        np.add.at(out, j, -fvec)
        return out

    def repin_to_bodies(self):
        """Translate every body-attached fiber so its minus end coincides
        with the body's nucleation site again — fiber and body do not move
        exactly together under finite dt
        (fiber_container_finite_difference.cpp:308-316)."""
        for f in self.fibers:
            ib, js = getattr(f, "binding_site", (-1, -1))
            if ib >= 0:
                delta = self.bodies[ib].nucleation_sites[js] - f.x[:, 0]
                f.x += delta[:, None]

    # ---- adaptive time-stepping driver (System::run, system.cpp:516-570) --
    def fiber_error(self):
        """max | |x_s| - 1 | over all fiber nodes
        (FiberContainerFiniteDifference::fiber_error_local, f_c_fd.cpp:79-89)."""
        err = 0.0
        for f in self.fibers:
            m = f.mats
            xs = (2.0 / f.length) * f.x @ m["D_1_0"]
            err = max(err, float(np.abs(np.linalg.norm(xs, axis=0) - 1.0).max()))
        return err

    def check_collision(self, periphery_shape=None, threshold=0.0):
        """System::check_collision (system.cpp:576-595): fiber-periphery
        (f_c_fd.cpp:39-55; minus-clamped fibers skip node 0),
        body-periphery (spherical shell only: |pos| + R > R_shell -
        threshold, periphery.cpp:94-97; ellipsoidal shells are stubbed in
        the reference) and body-body pairs. periphery_shape: dict like
        periphery_interaction."""
        from .fiber_fd import points_collide
        from .body import EllipsoidalBody
        if periphery_shape is not None:
            for f in self.fibers:
                pc = f.x[:, 1:] if f.minus_clamped else f.x
                if points_collide(pc, periphery_shape, threshold):
                    return True
            if periphery_shape["kind"] == "sphere":
                for b in self.bodies:
                    if np.linalg.norm(b.position) + b.radius > \
                            periphery_shape["radius"] - threshold:
                        return True
        for i, b1 in enumerate(self.bodies):
            for b2 in self.bodies[i + 1:]:
                # any ellipsoid pairing is stubbed False in the reference
                # (body_spherical.cpp:328-331, body_ellipsoidal.cpp)
                if isinstance(b1, EllipsoidalBody) or \
                        isinstance(b2, EllipsoidalBody):
                    continue
                if b1.check_collision(b2, threshold):
                    return True
        return False

    def backup(self):
        """System::backup (system.cpp:495-505)."""
        self._bak = [(f.x.copy(), f.tension.copy()) for f in self.fibers]
        self._bak_bodies = [(b.position.copy(), b.orientation.copy(),
                             b.solution_vec.copy()) for b in self.bodies]

    def restore(self):
        for f, (x, t) in zip(self.fibers, self._bak):
            f.x = x
            f.tension = t
        for b, (p, q, s) in zip(self.bodies, self._bak_bodies):
            b.place(p, q)
            b.solution_vec = s

    def run(self, t_final, adaptive=True, dt_min=1e-4, dt_max=None,
            beta_up=1.2, beta_down=0.5, fiber_error_tol=0.1,
            periphery_shape=None, tol=1e-10, maxiter=300, restart=None,
            on_accept=None):
        """The reference timestep loop, replicated EXACTLY
        (System::run, system.cpp:516-570): backup -> step -> accept if
        converged and fiber error within tolerance (grow dt when comfortably
        inside, params.cpp:9-10 defaults beta_up=1.2, beta_down=0.5),
        reject+restore+shrink otherwise; abort below dt_min; collision
        rejects with dt/2. Including the reference's time accounting quirk:
        properties.dt is updated to dt_new BEFORE time advances
        (system.cpp:554-558), so an accepted step advances the clock by the
        NEW dt although the state moved by the old one — kept verbatim so
        the reference's pinned end-to-end tests (e.g. the clamped-buckling
        peak values) reproduce. on_accept(system, time) is the
        trajectory-write hook (fires on every accepted step; callers apply
        the reference's dt_write crossing test, system.cpp:560-561)."""
        dt_max = dt_max if dt_max is not None else self.dt
        history = []
        while self.time < t_final:
            self.backup()
            info = self.step(tol=tol, maxiter=maxiter, restart=restart)
            err = self.fiber_error()
            dt_new = self.dt
            accept = not adaptive
            if adaptive:
                if info["converged"] and err <= fiber_error_tol:
                    accept = True
                    if err <= 0.9 * fiber_error_tol:
                        dt_new = min(dt_max, self.dt * beta_up)
                else:
                    dt_new = self.dt * beta_down
                    accept = False
                if info["converged"] and self.check_collision(periphery_shape):
                    dt_new = self.dt * 0.5
                    accept = False
                if dt_new < dt_min:
                    raise RuntimeError("Timestep smaller than dt_min")
                self.dt = dt_new  # BEFORE the clock advance (system.cpp:554)
            if accept:
                self.time += self.dt
                history.append(dict(time=self.time, dt=self.dt,
                                    iters=info["iters"], fiber_error=err))
                if on_accept is not None:
                    on_accept(self, self.time)
            else:
                self.restore()
        return history
