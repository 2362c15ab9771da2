"""Test-only backend over the CPU oracle for SystemFD orchestration tests.
Lives under tests/ (not the product package) so nothing on the product path
can import the oracle (oracle/ is test infrastructure; see oracle/__init__.py).
Method contract matches skellysim_amd.system_fd.HipBackend."""

import numpy as np


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

    def rotlet(self, centers, torques, r_trg, eta):
        return self.oracle.rotlet(centers, r_trg, torques, eta)

    def oseen_contract(self, r_src, r_trg, density, eta):
        return self.oracle.oseen_contract(r_src, r_trg, density, eta)

    def stresslet_times_normal(self, nodes, normals, eta):
        return self.oracle.np_stresslet_times_normal(nodes, normals)

    def shell_ops(self, A, M_inv):
        return (lambda x: A @ x), (lambda x: M_inv @ x)


class FakeDeviceBackend(OracleBackend):
    """TEST-ONLY: drives the device-resident code paths on CPU torch tensors
    (dev='cpu') so the device-mode orchestration — tensor layouts, gathers,
    batched fiber blocks, row-block GEMVs — is testable without a GPU. The
    pair-kernel device functions must be patched to oracle-backed fakes
    (see patch_device_kernels_with_oracle)."""

    def __init__(self):
        super().__init__()
        import torch
        self.torch = torch
        self.dev = torch.device("cpu")

    def _t(self, a):
        import torch
        if torch.is_tensor(a):
            return a.to(self.dev)
        return torch.from_numpy(np.ascontiguousarray(a)).to(self.dev)


def patch_device_kernels_with_oracle():
    """Replace skellysim_amd.evaluator's device kernels with oracle-backed
    CPU-tensor fakes (identical math by construction). TEST-ONLY; call in a
    worker subprocess (or restore() in-process). Returns a restore
    callable."""
    import torch
    import oracle
    import skellysim_amd.evaluator as ev

    saved = {k: getattr(ev, k) for k in
             ("stokeslet_device", "stresslet_device",
              "oseen_tensor_batched_device", "rotlet_device")}

    def restore():
        for k, v in saved.items():
            setattr(ev, k, v)

    def _np(t):
        return np.ascontiguousarray(t.detach().cpu().numpy())

    ev.stokeslet_device = lambda r, f, t, eta, out=None: torch.from_numpy(
        oracle.stokeslet(_np(r), _np(f), _np(t), eta))
    ev.stresslet_device = lambda r, f, t, eta, out=None: torch.from_numpy(
        oracle.stresslet(_np(r), _np(f), _np(t), eta))
    ev.oseen_tensor_batched_device = lambda pts, eta=1.0: torch.from_numpy(
        np.stack([oracle.oseen_tensor(p, eta) for p in _np(pts)]))
    ev.rotlet_device = lambda r, t, rho, eta=1.0, *a, **k: torch.from_numpy(
        oracle.rotlet(_np(r), _np(t), _np(rho), eta))
    return restore


