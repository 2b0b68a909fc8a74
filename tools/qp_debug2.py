import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from gcbfplus_amd.ops.qp import proxqp_solve
from scipy.optimize import minimize

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
def obj(x): return 0.5*torch.einsum("mi,mij,mj->m", x, ts[0], x) + (ts[1]*x).sum(1)

def slsqp(i):
    Hn, gn, Cn, bn, ln, un = H[i],g[i],C[i],b[i],l[i],u[i]
    cons=[{"type":"ineq","fun":lambda x: bn - Cn@x, "jac":lambda x:-Cn}]
    bounds=[(ln[j], None if not np.isfinite(un[j]) else un[j]) for j in range(nv)]
    r=minimize(lambda x:0.5*x@Hn@x+gn@x, np.zeros(nv), jac=lambda x:Hn@x+gn,
               constraints=cons, bounds=bounds, method="SLSQP",
               options={"maxiter":500,"ftol":1e-12})
    return r.fun

for iters in (100, 400, 1500):
    x_cpu = proxqp_solve(*ts, iters=iters)
    x_gpu = proxqp_solve(*[t.cuda() for t in ts], iters=iters).cpu()
    og, oc = obj(x_gpu), obj(x_cpu)
    d = og-oc
    print(f"iters={iters}: max|d|={d.abs().max():.4f} gpu_worse={(d>1e-3).sum().item()} cpu_worse={(d<-1e-3).sum().item()}")
for i in (8, 0, 21, 12):
    fs = slsqp(i)
    x_cpu = proxqp_solve(*ts, iters=100)
    x_gpu = proxqp_solve(*[t.cuda() for t in ts], iters=100).cpu()
    print(f"i={i}: slsqp={fs:.5f} cpu={obj(x_cpu)[i]:.5f} gpu={obj(x_gpu)[i]:.5f}")
