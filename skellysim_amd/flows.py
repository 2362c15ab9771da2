"""Container-level flow operators of the hot path
and the periphery dense operators (§8f next-row 1), on-device.

Mirrors:
  periphery_flow      — Periphery::flow (reference src/core/periphery.cpp:55-79):
                        f_dl(i*3+j) = 2*eta*normal_i*density_j per node, then
                        the stresslet evaluator (already /eta).
  fiber_flow          — FiberContainerFiniteDifference::flow
                        (src/core/fiber_container_finite_difference.cpp:172-214):
                        quadrature-weighted forces -> stokeslet evaluation ->
                        optional per-fiber self-interaction subtraction
                        vel_fib -= stokeslet_fib @ wf (lines 203-210).
  ShellOperator       — Periphery::apply_preconditioner / matvec
                        (src/core/periphery.cpp:21-47): the two dense
                        per-GMRES-iteration GEMVs (M_inv_ and
                        stresslet_plus_complementary_) with HBM-resident
                        matrices, rocBLAS dgemv via torch.mv; in a
                        distributed world the rows are block-sharded and the
                        solution all-gathered per apply (the reference's
                        MPI_Allgatherv, periphery.cpp:26,44 -> RCCL).

All tensors are torch fp64; the pair-kernel legs require CUDA tensors (the
product path; no CPU fallback).
"""

import torch

from .evaluator import stokeslet_device, stresslet_device
from .sharded import allgather_rows


def periphery_flow(node_pos, node_normal, density, r_trg, eta):
    """Velocity at r_trg due to the shell's double-layer density.

    node_pos, node_normal, density: (n_nodes, 3); r_trg: (n_trg, 3).
    f_dl(node)[i*3+j] = 2*eta*normal_i*density_j (periphery.cpp:68-71);
    the stresslet evaluator divides by eta (periphery.cpp:74).
    """
    if node_pos.shape[0] == 0:
        return torch.zeros_like(r_trg)
    f_dl = 2.0 * eta * torch.einsum("ni,nj->nij", node_normal, density).reshape(-1, 9)
    return stresslet_device(node_pos, f_dl.contiguous(), r_trg, eta)


def fiber_flow(r_src, fib_forces, weights, r_trg, eta, fiber_sizes=None,
               self_stokeslets=None):
    """Velocity at r_trg due to fiber forces.

    r_src, fib_forces: (n_fib_nodes, 3); weights: (n_fib_nodes,) — the
    per-node quadrature weights 0.5 * length * weights_0
    (fiber_container_finite_difference.cpp:186). If self_stokeslets is given
    (list/stacked tensor of per-fiber (3n, 3n) matrices, with fiber_sizes the
    per-fiber node counts), the per-fiber self term stokeslet_ @ wf is
    subtracted from that fiber's slice of the result (lines 203-210) — this
    requires r_trg to start with the fiber nodes in order, as in the
    reference's matvec (system.cpp:298-299).
    """
    if r_src.shape[0] == 0:
        return torch.zeros_like(r_trg)
    wf = fib_forces * weights[:, None]
    vel = stokeslet_device(r_src, wf.contiguous(), r_trg, eta)
    if self_stokeslets is not None:
        if fiber_sizes is None:
            raise ValueError("fiber_sizes required with self_stokeslets")
        if all(n == fiber_sizes[0] for n in fiber_sizes):
            # uniform fibers: one batched GEMV (rocBLAS batched under torch.bmm)
            n = fiber_sizes[0]
            nf = len(fiber_sizes)
            S = self_stokeslets if torch.is_tensor(self_stokeslets) \
                else torch.stack(list(self_stokeslets))
            wf_flat = wf.reshape(nf, 3 * n, 1)
            corr = torch.bmm(S, wf_flat).reshape(nf * n, 3)
            vel[: nf * n] -= corr
        else:
            off = 0
            for S, n in zip(self_stokeslets, fiber_sizes):
                wf_flat = wf[off: off + n].reshape(-1)
                vel[off: off + n] -= (S @ wf_flat).reshape(n, 3)
                off += n
    return vel


def body_flow(node_pos, node_normals, densities, centers, forces, torques, r_trg, eta,
              reg=5e-3, epsilon_distance=1e-5):
    """Velocity at r_trg due to rigid bodies (BodyContainer::flow_spherical /
    _ellipsoidal, src/core/body_container.cpp:269-410): a stresslet from the
    body quadrature nodes with f_dl = 2*eta*n_i*d_j (lines 299-302), plus a
    stokeslet from the body centers with the link forces (line 327), plus a
    rotlet from the centers with the torques (line 335; always direct)."""
    v = periphery_flow(node_pos, node_normals, densities, r_trg, eta)
    if centers.shape[0]:
        v = v + stokeslet_device(centers, forces, r_trg, eta)
        from .evaluator import rotlet_device
        v = v + rotlet_device(centers, r_trg, torques, eta, reg, epsilon_distance)
    return v


def velocity_at_targets(r_trg, eta, fiber=None, shell=None, bodies=None):
    """Composite free-space velocity at arbitrary targets — the kernel
    portion of System::velocity_at_targets (src/core/system.cpp:330-384,
    lines 355-359: fiber flow + body flow + shell flow summed; the reference
    additionally overwrites points inside bodies with rigid-body velocity,
    which is host-side post-processing outside this engine).

    fiber  = dict(r_src, forces, weights[, fiber_sizes, self_stokeslets])
    shell  = dict(node_pos, node_normal, density)
    bodies = dict(node_pos, node_normals, densities, centers, forces, torques)
    """
    u = torch.zeros_like(r_trg)
    if fiber is not None:
        u = u + fiber_flow(fiber["r_src"], fiber["forces"], fiber["weights"], r_trg, eta,
                           fiber_sizes=fiber.get("fiber_sizes"),
                           self_stokeslets=fiber.get("self_stokeslets"))
    if shell is not None:
        u = u + periphery_flow(shell["node_pos"], shell["node_normal"], shell["density"],
                               r_trg, eta)
    if bodies is not None:
        u = u + body_flow(bodies["node_pos"], bodies["node_normals"], bodies["densities"],
                          bodies["centers"], bodies["forces"], bodies["torques"], r_trg, eta)
    return u


class ShellOperator:
    """HBM-resident periphery dense operators (M_inv, stresslet+complementary).

    Single-GPU: matrices are the full (3N, 3N); distributed: each rank holds
    its contiguous row block (3n_local, 3N) — the reference's row
    distribution (periphery.cpp:422-442) — and x is all-gathered per apply.
    The matrices stay resident in HBM across GMRES iterations (4.6 GB at an
    8k-node shell — trivial in 288 GB).
    """

    def __init__(self, M_inv_rows, stresslet_plus_complementary_rows, distributed=False):
        self.M_inv = M_inv_rows.contiguous()
        self.SPC = stresslet_plus_complementary_rows.contiguous()
        self.distributed = distributed
        if self.M_inv.dtype != torch.float64:
            raise TypeError("ShellOperator expects fp64")

    def _gather(self, x_local):
        if self.distributed:
            return allgather_rows(x_local.reshape(-1, 1)).reshape(-1)
        return x_local

    def apply_preconditioner(self, x_local):
        """M_inv_ @ allgather(x) (periphery.cpp:21-29)."""
        x = self._gather(x_local)
        return torch.mv(self.M_inv, x)

    def matvec(self, x_local, v_local):
        """stresslet_plus_complementary_ @ allgather(x) + v (periphery.cpp:34-47)."""
        x = self._gather(x_local)
        return torch.addmv(v_local.reshape(-1), self.SPC, x)
