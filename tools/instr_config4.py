import sys, time
sys.path.insert(0, ".")
import numpy as np, torch
from skellysim_amd.fiber_fd import FiberFD
from skellysim_amd.system_fd import SystemFD, HipBackend, Shell
from skellysim_amd.periphery_precompute import assemble_shell_operator

fix = np.load("tests/golden/ellipsoid_8192_nodes.npz")
dev = torch.device("cuda:0")
A, M_inv = assemble_shell_operator(torch.from_numpy(fix["nodes"]).to(dev),
                                   torch.from_numpy(fix["normals"]).to(dev),
                                   torch.from_numpy(fix["quadrature_weights"]).to(dev))
shell = Shell(fix["nodes"], fix["normals"], A.cpu().numpy(), M_inv.cpu().numpy())
sel = np.linspace(0, 8191, 512).astype(int)
fibers = []
for i in sel:
    p = fix["nodes"][i]; n = fix["normals"][i]/np.linalg.norm(fix["normals"][i])
    s = np.linspace(0.02, 1.02, 64)
    fibers.append(FiberFD(p[None,:]+s[:,None]*n[None,:], length=1.0, bending_rigidity=2.5e-3,
                          eta=1.0, minus_clamped=True, force_scale=-0.05))
sys_ = SystemFD(fibers, eta=1.0, dt=0.025, shell=shell, backend=HipBackend())

nmv = [0, 0.0]; npc = [0, 0.0]
omv, opc = sys_.apply_matvec, sys_.apply_preconditioner
def mv(x):
    t0=time.perf_counter(); r = omv(x); nmv[0]+=1; nmv[1]+=time.perf_counter()-t0; return r
def pc(x):
    t0=time.perf_counter(); r = opc(x); npc[0]+=1; npc[1]+=time.perf_counter()-t0; return r
sys_.apply_matvec, sys_.apply_preconditioner = mv, pc
t0=time.perf_counter()
info = sys_.solve(tol=1e-10, maxiter=15, restart=15)
dt=time.perf_counter()-t0
it, cv = info["iters"], info["converged"]
print(f"iters={it} conv={cv} total={dt:.2f}s")
print(f"matvec: n={nmv[0]} total={nmv[1]:.2f}s avg={nmv[1]/max(1,nmv[0])*1e3:.0f}ms")
print(f"precond: n={npc[0]} total={npc[1]:.2f}s avg={npc[1]/max(1,npc[0])*1e3:.0f}ms")
print(f"other: {dt-nmv[1]-npc[1]:.2f}s")
