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
shell = Shell(fix["nodes"], fix["normals"], A, M_inv)   # keep on device
sel = np.linspace(0, 8191, 512).astype(int)
fibers = []
for i in sel:
    p = fix["nodes"][i]; n = fix["normals"][i]/np.linalg.norm(fix["normals"][i])
    s = np.linspace(0.02, 1.02, 64)
    fibers.append(FiberFD(p[None,:]+s[:,None]*n[None,:], length=1.0, bending_rigidity=2.5e-3,
                          eta=1.0, minus_clamped=True, force_scale=-0.05))
sys_ = SystemFD(fibers, eta=1.0, dt=0.025, shell=shell, backend=HipBackend())
sys_.prep_state_for_solver()
sys_._build_device_operators()
x = torch.from_numpy(np.random.default_rng(0).uniform(-1,1,sys_.fiber_sol_size+sys_.shell_sol_size)).to(dev)

def timeit(fn, iters=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters

print(f"matvec_device: {timeit(lambda: sys_._apply_matvec_device(x))*1e3:.2f} ms")
print(f"precond_device: {timeit(lambda: sys_._apply_precond_device(x))*1e3:.2f} ms")
# gmres end to end with null operators to isolate solver overhead
nullmv = lambda v: v * 1.0
from skellysim_amd.gmres import gmres
b = x.clone()
t0=time.perf_counter()
_, info = gmres(nullmv, b, tol=0, maxiter=50, restart=50)
torch.cuda.synchronize()
print(f"gmres-internal per iter (null op, device): {(time.perf_counter()-t0)/max(1,info['iters'])*1e3:.2f} ms  iters={info['iters']}")

import cProfile, pstats, io
pr = cProfile.Profile()
pr.enable()
info = sys_.solve(tol=1e-10, maxiter=40, restart=40)
pr.disable()
s = io.StringIO(); pstats.Stats(pr, stream=s).sort_stats("tottime").print_stats(16)
print("iters:", info["iters"])
print(s.getvalue())
