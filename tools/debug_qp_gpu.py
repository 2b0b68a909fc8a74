"""Isolate the GPU-only ADMM divergence at nv=192: which ROCm linalg op is
wrong? Loads gpurun_out/qp_inputs_n64.pt and checks each primitive GPU vs
CPU, then runs the loop with per-iteration norms."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

d = torch.load("tools/data/qp_inputs_n64.pt", weights_only=False)
H, g, C, b, l, u = (d[k] for k in ("H", "g", "C", "b", "l", "u"))
M, n = g.shape
k = b.shape[1]
print(f"M={M} n={n} k={k}", flush=True)

for dt in (torch.float32, torch.float64):
    Hs = H.to(dt)
    eye_n = torch.eye(n, dtype=dt)
    A = torch.cat([C.to(dt), eye_n.expand(M, n, n)], dim=1)
    AtA = A.transpose(1, 2) @ A
    K = Hs + 1e-6 * eye_n + 0.1 * AtA
    Kg = K.cuda()
    # 1) cholesky
    Lc_cpu = torch.linalg.cholesky(K)
    Lc_gpu = torch.linalg.cholesky(Kg)
    err_f = (Lc_gpu.cpu() - Lc_cpu).abs().max().item()
    recon = (Lc_gpu @ Lc_gpu.transpose(1, 2) - Kg).abs().max().item()
    print(f"[{dt}] cholesky factor diff={err_f:.3e} recon_err={recon:.3e}", flush=True)
    # 2) cholesky_solve
    rhs = torch.randn(M, n, 1, dtype=dt)
    x_cpu = torch.cholesky_solve(rhs, Lc_cpu)
    x_gpu = torch.cholesky_solve(rhs.cuda(), Lc_gpu).cpu()
    resid_gpu = (K @ x_gpu - rhs).abs().max().item()
    resid_cpu = (K @ x_cpu - rhs).abs().max().item()
    print(f"[{dt}] cholesky_solve: diff={(x_gpu-x_cpu).abs().max().item():.3e} "
          f"resid gpu={resid_gpu:.3e} cpu={resid_cpu:.3e}", flush=True)
    # 3) solve with gpu-factored L against gpu rhs, einsum check
    e_cpu = torch.einsum("mkn,mn->mk", A, rhs.squeeze(-1))
    e_gpu = torch.einsum("mkn,mn->mk", A.cuda(), rhs.squeeze(-1).cuda()).cpu()
    print(f"[{dt}] einsum diff={(e_gpu-e_cpu).abs().max().item():.3e}", flush=True)

# full loop on GPU with iteration tracing (f64)
from gcbfplus_amd.ops import qp as qpmod

def traced_solve(dev):
    dt = torch.float64 if dev == "cuda" else torch.float32
    Ht, gt, Ct, bt, lt, ut = (t.to(dev) for t in (H, g, C, b, l, u))
    # replicate the first part of proxqp_solve with tracing
    Hs = Ht.to(dt); gs0 = gt.to(dt); Cs = Ct.to(dt); bs = bt.to(dt)
    eye_n = torch.eye(n, dtype=dt, device=dev)
    A = torch.cat([Cs, eye_n.expand(M, n, n)], dim=1)
    lo = torch.cat([torch.full_like(bs, -float("inf")), lt.to(dt)], dim=1)
    hi = torch.cat([bs, ut.to(dt)], dim=1)
    D = torch.ones(M, n, dtype=dt, device=dev)
    E = torch.ones(M, k + n, dtype=dt, device=dev)
    Hss, As = Hs, A
    for _ in range(5):
        col_h = Hss.abs().amax(dim=1)
        col_a = As.abs().amax(dim=1)
        dn = torch.clamp(torch.maximum(col_h, col_a), min=1e-8).rsqrt()
        de = torch.clamp(As.abs().amax(dim=2), min=1e-8).rsqrt()
        Hss = Hss * dn[:, :, None] * dn[:, None, :]
        As = As * de[:, :, None] * dn[:, None, :]
        D = D * dn; E = E * de
    gs = gs0 * D
    c = 1.0 / torch.clamp(torch.maximum(Hss.abs().amax(dim=1).mean(dim=1),
                                        gs.abs().amax(dim=1)), min=1.0)
    Hss = Hss * c[:, None, None]; gs = gs * c[:, None]
    los = lo * E; his = hi * E
    AtA = As.transpose(1, 2) @ As
    rho_v = torch.full((M, 1), 0.1, dtype=dt, device=dev)
    Lc = torch.linalg.cholesky(Hss + 1e-6 * eye_n + rho_v[:, :, None] * AtA)
    x = torch.zeros(M, n, dtype=dt, device=dev)
    z = torch.zeros(M, k + n, dtype=dt, device=dev)
    y = torch.zeros(M, k + n, dtype=dt, device=dev)
    alpha = 1.6
    for it in range(60):
        rhs = 1e-6 * x - gs + torch.einsum("mkn,mk->mn", As, rho_v * z - y)
        xt = torch.cholesky_solve(rhs.unsqueeze(-1), Lc).squeeze(-1)
        zt = torch.einsum("mkn,mn->mk", As, xt)
        x = alpha * xt + (1 - alpha) * x
        z_relax = alpha * zt + (1 - alpha) * z
        znew = torch.clamp(z_relax + y / rho_v, los, his)
        y = y + rho_v * (z_relax - znew)
        z = znew
        if it % 5 == 0 or it == 24:
            print(f"  [{dev}] it={it} |x|={x.abs().max().item():.3e} "
                  f"|y|={y.abs().max().item():.3e} |xt|={xt.abs().max().item():.3e} "
                  f"finite={torch.isfinite(x).all().item()}", flush=True)
    return x

print("CPU trace:", flush=True)
traced_solve("cpu")
print("GPU trace:", flush=True)
traced_solve("cuda")
print("DONE", flush=True)

# pinpoint: replay the CPU trajectory; per iteration compute every op on
# GPU from the CPU state and compare — first mismatch = the broken op.
def op_bisect():
    dt = torch.float64
    Hs0 = H.to(dt); gs0 = g.to(dt)
    eye_n = torch.eye(n, dtype=dt)
    A = torch.cat([C.to(dt), eye_n.expand(M, n, n)], dim=1)
    lo = torch.cat([torch.full_like(b.to(dt), -float("inf")), l.to(dt)], dim=1)
    hi = torch.cat([b.to(dt), u.to(dt)], dim=1)
    D = torch.ones(M, n, dtype=dt); E = torch.ones(M, k + n, dtype=dt)
    Hss, As = Hs0, A
    for _ in range(5):
        dn = torch.clamp(torch.maximum(Hss.abs().amax(dim=1), As.abs().amax(dim=1)), min=1e-8).rsqrt()
        de = torch.clamp(As.abs().amax(dim=2), min=1e-8).rsqrt()
        Hss = Hss * dn[:, :, None] * dn[:, None, :]
        As = As * de[:, :, None] * dn[:, None, :]
        D = D * dn; E = E * de
    gs = gs0 * D
    c = 1.0 / torch.clamp(torch.maximum(Hss.abs().amax(dim=1).mean(dim=1), gs.abs().amax(dim=1)), min=1.0)
    Hss = Hss * c[:, None, None]; gs = gs * c[:, None]
    los = lo * E; his = hi * E
    AtA = As.transpose(1, 2) @ As
    rho = torch.full((M, 1), 0.1, dtype=dt)
    Lc = torch.linalg.cholesky(Hss + 1e-6 * eye_n + rho[:, :, None] * AtA)
    Ag, Lcg = As.cuda(), Lc.cuda()
    losg, hisg, gsg = los.cuda(), his.cuda(), gs.cuda()
    rhog = rho.cuda()
    x = torch.zeros(M, n, dtype=dt); z = torch.zeros(M, k + n, dtype=dt)
    y = torch.zeros(M, k + n, dtype=dt)
    alpha = 1.6
    for it in range(12):
        rhs = 1e-6 * x - gs + torch.einsum("mkn,mk->mn", As, rho * z - y)
        rhs_g = (1e-6 * x.cuda() - gsg + torch.einsum("mkn,mk->mn", Ag, (rho * z - y).cuda())).cpu()
        xt = torch.cholesky_solve(rhs.unsqueeze(-1), Lc).squeeze(-1)
        xt_g = torch.cholesky_solve(rhs.cuda().unsqueeze(-1), Lcg).squeeze(-1).cpu()
        zt = torch.einsum("mkn,mn->mk", As, xt)
        zt_g = torch.einsum("mkn,mn->mk", Ag, xt.cuda()).cpu()
        x = alpha * xt + (1 - alpha) * x
        z_relax = alpha * zt + (1 - alpha) * z
        znew = torch.clamp(z_relax + y / rho, los, his)
        znew_g = torch.clamp((z_relax + y / rho).cuda(), losg, hisg).cpu()
        y = y + rho * (z_relax - znew)
        z = znew
        print(f"it {it}: rhs {(rhs_g-rhs).abs().max().item():.2e} "
              f"chol {(xt_g-xt).abs().max().item():.2e} "
              f"Ax {(zt_g-zt).abs().max().item():.2e} "
              f"clamp {(znew_g-znew).abs().max().item():.2e}", flush=True)

print("OP BISECT:", flush=True)
op_bisect()
print("DONE2", flush=True)


# prep-phase bisect: Ruiz loop, AtA (transposed bmm), first cholesky on GPU
def prep_bisect():
    dt = torch.float64
    for dev in ("cpu", "cuda"):
        Hs = H.to(dt).to(dev); gs0 = g.to(dt).to(dev)
        eye_n = torch.eye(n, dtype=dt, device=dev)
        A = torch.cat([C.to(dt).to(dev), eye_n.expand(M, n, n)], dim=1)
        D = torch.ones(M, n, dtype=dt, device=dev)
        E = torch.ones(M, k + n, dtype=dt, device=dev)
        Hss, As = Hs, A
        for _ in range(5):
            dn = torch.clamp(torch.maximum(Hss.abs().amax(dim=1), As.abs().amax(dim=1)), min=1e-8).rsqrt()
            de = torch.clamp(As.abs().amax(dim=2), min=1e-8).rsqrt()
            Hss = Hss * dn[:, :, None] * dn[:, None, :]
            As = As * de[:, :, None] * dn[:, None, :]
            D = D * dn; E = E * de
        AtA = As.transpose(1, 2) @ As
        AtA2 = torch.einsum("mki,mkj->mij", As, As)
        AtA3 = As.transpose(1, 2).contiguous() @ As
        Kmat = Hss + 1e-6 * eye_n + 0.1 * AtA
        Lc = torch.linalg.cholesky(Kmat)
        if dev == "cpu":
            ref = dict(As=As, Hss=Hss, AtA=AtA, Lc=Lc, D=D)
        else:
            for name, t in (("As", As), ("Hss", Hss), ("AtA", AtA),
                            ("AtA2", AtA2), ("AtA3", AtA3), ("Lc", Lc), ("D", D)):
                r = ref.get(name, ref.get("AtA") if name.startswith("AtA") else None)
                diff = (t.cpu() - r).abs().max().item()
                print(f"prep {name}: gpu-vs-cpu diff={diff:.3e}", flush=True)
            asym = (AtA - AtA.transpose(1, 2)).abs().max().item()
            print(f"prep AtA asymmetry on GPU: {asym:.3e}", flush=True)

print("PREP BISECT:", flush=True)
prep_bisect()
print("DONE3", flush=True)
