import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from gcbfplus_amd.ops.qp import proxqp_solve

rng = np.random.default_rng(7)
N, nu = 8, 2
nv = N*nu+N; M = 64
H = np.tile(np.eye(nv, dtype=np.float32), (M,1,1)); H[:, N*nu:, N*nu:] *= 10.0
u_ref = rng.uniform(-1,1,size=(M,N*nu)).astype(np.float32)
g = np.concatenate([-u_ref, 1e3*np.ones((M,N),np.float32)], axis=1)
Lg = rng.normal(size=(M,N,N*nu)).astype(np.float32)
C = -np.concatenate([Lg, np.tile(np.eye(N,dtype=np.float32),(M,1,1))], axis=2)
b = (rng.normal(size=(M,N))*0.5).astype(np.float32)
l = np.concatenate([-np.ones((M,N*nu),np.float32), np.zeros((M,N),np.float32)],1)
u = np.concatenate([np.ones((M,N*nu),np.float32), np.full((M,N),np.inf,np.float32)],1)
ts = [torch.from_numpy(t) for t in (H,g,C,b,l,u)]
x_cpu = proxqp_solve(*ts, iters=100)
x_gpu = proxqp_solve(*[t.cuda() for t in ts], iters=100).cpu()
def obj(x): return 0.5*torch.einsum("mi,mij,mj->m", x, ts[0], x) + (ts[1]*x).sum(1)
og, oc = obj(x_gpu), obj(x_cpu)
viol_g = (torch.einsum("mkn,mn->mk", ts[2], x_gpu) - ts[3]).amax(1)
viol_c = (torch.einsum("mkn,mn->mk", ts[2], x_cpu) - ts[3]).amax(1)
boxv_g = torch.maximum(ts[4]-x_gpu, x_gpu-torch.nan_to_num(ts[5], posinf=1e30)).amax(1)
d = (og-oc)
order = d.abs().argsort(descending=True)
print("worst 8 problems:")
for i in order[:8].tolist():
    print(f"  i={i} obj_gpu={og[i]:.5f} obj_cpu={oc[i]:.5f} d={d[i]:+.2e} violG={viol_g[i]:.2e} violC={viol_c[i]:.2e} boxG={boxv_g[i]:.2e}")
print("summary: max|d|", d.abs().max().item(), "gpu better count", (d < -1e-6).sum().item(), "worse", (d > 1e-6).sum().item())
print("max viol gpu", viol_g.max().item(), "cpu", viol_c.max().item())
print("x diff max", (x_gpu-x_cpu).abs().max().item())
