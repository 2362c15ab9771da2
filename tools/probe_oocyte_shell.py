import sys, time
sys.path.insert(0, ".")
import numpy as np, torch
from skellysim_amd.periphery_precompute import assemble_shell_operator
from skellysim_amd.flows import ShellOperator, periphery_flow
from skellysim_amd.gmres import gmres

fix = np.load("tests/golden/oocyte_nodes.npz")
dev = torch.device("cuda:0")
N = len(fix["nodes"])
A, M_inv = assemble_shell_operator(torch.from_numpy(fix["nodes"]).to(dev),
                                   torch.from_numpy(fix["normals"]).to(dev),
                                   torch.from_numpy(fix["quadrature_weights"]).to(dev))
op = ShellOperator(M_inv, A)
U = np.array([0.3, -0.2, 0.7])
rhs = torch.from_numpy(-np.tile(U, N)).to(dev)
v0 = torch.zeros_like(rhs)
q, info = gmres(lambda x: op.matvec(x, v0), rhs, precond=op.apply_preconditioner,
                tol=1e-10, maxiter=30, restart=30)
print("gmres:", info["converged"], info["iters"])
# interior points along the axis region
T, p1, p2, L, sf = (float(fix[k]) for k in ("envelope_T","envelope_p1","envelope_p2","envelope_length","scale_factor"))
rng = np.random.default_rng(5)
x = rng.uniform(-2.5, 2.5, 400)
h = 0.5*T*(1+2*x/L)**p1*(1-2*x/L)**p2*L
r = rng.uniform(0, 0.4, 400) * h
th = rng.uniform(0, 2*np.pi, 400)
pts = np.stack([x, r*np.cos(th), r*np.sin(th)], axis=1) * sf
pts = pts[:100]
u = periphery_flow(torch.from_numpy(fix["nodes"]).to(dev), torch.from_numpy(fix["normals"]).to(dev),
                   q.reshape(N,3), torch.from_numpy(pts).to(dev), 1.0)
torch.cuda.synchronize()
resid = np.abs(u.cpu().numpy() + U[None,:]).max()
print(f"interior |U + D[q]| max = {resid:.3e}")
# operator sanity
I_err = float(torch.norm(A @ M_inv[:, :100] - torch.eye(3*N, dtype=torch.float64, device=dev)[:, :100]))
print("A@Minv cols err:", I_err)
