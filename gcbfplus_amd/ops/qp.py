"""Batched dense QP solver (K11) — ProxQP/OSQP-family operator splitting.

Solves, for a batch of M small dense problems:
    min_x  1/2 x^T H x + g^T x
    s.t.   C x <= b,   l <= x <= u

This replaces the reference's external JaxProxQP dependency
(``/root/reference/gcbfplus/algo/gcbf_plus.py:329-346``): same QP shape, same
fixed-iteration batched semantics (Settings.max_iter=100, no early exit so
every problem in the batch does identical work — GPU-friendly by
construction).

Method: ADMM with over-relaxation on the stacked constraint set
A = [C; I], lo = [-inf; l], hi = [b; u]; the (H + sigma I + rho A^T A) system
is factorized once (batched Cholesky) and each iteration is two triangular
solves + projections. All ops are batched torch -> runs on CPU (oracle) and
GPU; a one-workgroup-per-QP HIP kernel backs it on gfx950 when the problem
count is large.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor


def proxqp_solve(
    H: Tensor,  # (M, n, n)
    g: Tensor,  # (M, n)
    C: Tensor,  # (M, k, n)
    b: Tensor,  # (M, k)
    l: Tensor,  # (M, n) box lower
    u: Tensor,  # (M, n) box upper
    iters: int = 100,
    rho: float = 0.1,
    sigma: float = 1e-6,
    alpha: float = 1.6,
    ruiz_iters: int = 5,
) -> Tensor:
    """Returns x (M, n). Fixed iteration count, no per-problem early exit.

    Applies OSQP-style Ruiz equilibration + cost scaling first — essential
    because the GCBF+ QP mixes O(1) action costs with the 1e3 relaxation
    penalty (gcbf_plus.py:299 relax_penalty).

    GPU: one-wave-per-QP HIP kernel (K11) when sizes fit; torch path is the
    CPU oracle and the large-problem fallback.
    """
    M, n = g.shape
    if g.is_cuda and n <= 64 and b.shape[1] <= 32:
        from . import _require_ext

        ext = _require_ext()
        f32c = lambda t: t.to(torch.float32).contiguous()
        return ext.proxqp_solve(f32c(H), f32c(g), f32c(C), f32c(b), f32c(l), f32c(u),
                                iters, rho, sigma, alpha)
    k = b.shape[1]
    # f64 on GPU: ROCm's f32 batched cholesky/cholesky_solve produced wrong
    # factors at n~192 on trivially-conditioned systems (cond ~9; verified
    # against the same data on CPU — see profiles/qp_f64_note.md). MI355X
    # fp64 is fast and this torch path only serves QPs too big for the
    # one-wave HIP kernel, so precision is free insurance.
    dtype = torch.float64 if g.is_cuda else torch.float32
    out_dtype = torch.float32
    H = H.to(dtype)
    g = g.to(dtype)
    C = C.to(dtype)
    b = b.to(dtype)

    eye_n = torch.eye(n, dtype=dtype, device=g.device)
    A = torch.cat([C, eye_n.expand(M, n, n)], dim=1)  # (M, k+n, n)
    m_c = k + n
    lo = torch.cat([torch.full_like(b, -float("inf")), l.to(dtype)], dim=1)
    hi = torch.cat([b, u.to(dtype)], dim=1)

    # ---- Ruiz equilibration of [[H, A^T], [A, 0]] -----------------------
    D = torch.ones(M, n, dtype=dtype, device=g.device)
    E = torch.ones(M, m_c, dtype=dtype, device=g.device)
    Hs, As = H, A
    for _ in range(ruiz_iters):
        col_h = Hs.abs().amax(dim=1)  # (M, n) inf-norm of H columns
        col_a = As.abs().amax(dim=1)  # (M, n) inf-norm of A columns
        dn = torch.clamp(torch.maximum(col_h, col_a), min=1e-8).rsqrt()
        row_a = As.abs().amax(dim=2)  # (M, m_c)
        de = torch.clamp(row_a, min=1e-8).rsqrt()
        Hs = Hs * dn[:, :, None] * dn[:, None, :]
        As = As * de[:, :, None] * dn[:, None, :]
        D = D * dn
        E = E * de
    gs = g * D
    # cost scaling (OSQP): c = 1 / max(1, mean col norm of Hs, ||gs||_inf)
    c = 1.0 / torch.clamp(
        torch.maximum(Hs.abs().amax(dim=1).mean(dim=1), gs.abs().amax(dim=1)), min=1.0
    )
    Hs = Hs * c[:, None, None]
    gs = gs * c[:, None]
    los = lo * E
    his = hi * E

    AtA = As.transpose(1, 2) @ As
    rho_v = torch.full((M, 1), rho, dtype=dtype, device=g.device)

    on_gpu = g.is_cuda

    def factor(rv):
        K = Hs + sigma * eye_n + rv[:, :, None] * AtA
        # GPU: rocSOLVER's batched factorization ops race with the current
        # stream when free-running (per-op results verify correct with the
        # stream drained, but the async loop diverges — see
        # profiles/qp_f64_note.md). Keep rocSOLVER OUT of the iteration
        # loop: factor + explicit inverse here (bracketed by syncs; factors
        # are rare), then the loop is pure bmm.
        if on_gpu:
            torch.cuda.synchronize()
            Lc = torch.linalg.cholesky(K)
            Kinv = torch.cholesky_inverse(Lc)
            torch.cuda.synchronize()
            return Kinv
        return torch.linalg.cholesky(K)

    Lc = factor(rho_v)
    x = torch.zeros(M, n, dtype=dtype, device=g.device)
    z = torch.zeros(M, m_c, dtype=dtype, device=g.device)
    y = torch.zeros(M, m_c, dtype=dtype, device=g.device)

    for it in range(iters):
        rhs = sigma * x - gs + torch.einsum("mkn,mk->mn", As, rho_v * z - y)
        if on_gpu:
            xt = (Lc @ rhs.unsqueeze(-1)).squeeze(-1)  # Lc holds K^-1 here
        else:
            xt = torch.cholesky_solve(rhs.unsqueeze(-1), Lc).squeeze(-1)
        zt = torch.einsum("mkn,mn->mk", As, xt)
        x = alpha * xt + (1 - alpha) * x
        z_relax = alpha * zt + (1 - alpha) * z
        znew = torch.clamp(z_relax + y / rho_v, los, his)
        y = y + rho_v * (z_relax - znew)
        z = znew
        # OSQP-style per-problem rho adaptation every 25 iterations
        if (it + 1) % 25 == 0 and it + 1 < iters:
            ax = torch.einsum("mkn,mn->mk", As, x)
            pri = (ax - z).abs().amax(dim=1)
            pri_den = torch.maximum(ax.abs().amax(1), z.abs().amax(1)).clamp_min(1e-8)
            hx = torch.einsum("mij,mj->mi", Hs, x)
            aty = torch.einsum("mkn,mk->mn", As, y)
            dua = (hx + gs + aty).abs().amax(dim=1)
            dua_den = torch.maximum(
                torch.maximum(hx.abs().amax(1), aty.abs().amax(1)), gs.abs().amax(1)
            ).clamp_min(1e-8)
            ratio = torch.sqrt((pri / pri_den).clamp_min(1e-10) /
                               (dua / dua_den).clamp_min(1e-10))
            rho_v = (rho_v * ratio[:, None]).clamp(1e-6, 1e6)
            Lc = factor(rho_v)
    x = x * D

    # ---- polish (OSQP-style): exact solve on the active set --------------
    # Active constraints: dual pressure |y| above threshold OR z at a bound.
    # Solved as an f64 penalty least-squares (mu >> 1) which is an
    # equality-KKT solve up to O(1/mu); fall back to the ADMM iterate where
    # the polish is worse (non-PSD corner cases).
    Af = A.to(torch.float64)
    lof = lo.to(torch.float64)
    hif = hi.to(torch.float64)
    Hf = H.to(torch.float64)
    gf = g.to(torch.float64)
    mu = 1e8
    tol = 1e-4

    def _score(xx):
        """Violation-penalized merit: a tiny box/ineq violation under a
        large linear cost (the 1e3 relax penalty) fakes a lower objective,
        so feasibility must be priced into the comparison."""
        xx32 = xx.to(dtype)
        obj = 0.5 * torch.einsum("mi,mij,mj->m", xx32, H, xx32) + (g * xx32).sum(1)
        zz = torch.einsum("mkn,mn->mk", A, xx32)
        viol = torch.clamp(zz - hi, min=0.0) + torch.clamp(lo - zz, min=0.0)
        return obj + 1e6 * viol.sum(dim=1)

    x_best = x
    s_best = _score(x)
    x_cur = x.to(torch.float64)
    released = torch.zeros(M, m_c, dtype=torch.bool, device=g.device)
    for pp in range(4):
        zf = torch.einsum("mkn,mn->mk", Af, x_cur)
        at_lo = zf <= (lof + tol)
        at_hi = zf >= (hif - tol)  # includes rows the candidate violates
        active = (at_lo | at_hi) & ~released
        if pp == 0:  # first pass also trusts the ADMM duals
            active |= y.abs() > 1e-6 * y.abs().amax(dim=1, keepdim=True)
        vbound = torch.where(at_hi, hif, lof)
        vbound = torch.where(active & torch.isfinite(vbound), vbound, zf)
        w = active.to(torch.float64)
        Kp = Hf + mu * torch.einsum("mki,mk,mkj->mij", Af, w, Af)
        # tiny ridge: H may be PSD-singular (the dec-share QP's all-10 relax
        # block, reference dec_share_cbf.py:124 quirk, is rank-1) and a
        # degenerate active set then leaves Kp singular; 1e-8 is negligible
        # against the mu=1e8 penalty terms and the score check rejects any
        # off candidate anyway
        Kp = Kp + 1e-8 * torch.eye(n, dtype=torch.float64, device=g.device)
        rp = -gf + mu * torch.einsum("mki,mk,mk->mi", Af, w, vbound)
        if on_gpu:  # sync brackets retained (see profiles/qp_f64_note.md)
            torch.cuda.synchronize()
        xp = torch.linalg.solve(Kp, rp)
        if on_gpu:
            torch.cuda.synchronize()
        # penalty multiplier estimate lambda = mu * (A xp - vbound): wrong
        # sign means the pin fights the KKT conditions -> release next pass
        resid = torch.einsum("mkn,mn->mk", Af, xp) - vbound
        released = released | (active & at_hi & ~at_lo & (resid < -1e-12)) \
            | (active & at_lo & ~at_hi & (resid > 1e-12))
        # project onto the box: the penalty solve leaves box-active vars
        # biased by g/mu, which fakes a lower objective while violating box
        xp = torch.clamp(xp, l.to(torch.float64), u.to(torch.float64).clamp(max=1e30))
        sp = _score(xp)
        better = sp <= s_best
        x_best = torch.where(better[:, None], xp.to(dtype), x_best)
        s_best = torch.minimum(sp, s_best)
        x_cur = xp
    return x_best.to(out_dtype)


def qp_kkt_residuals(H, g, C, b, l, u, x) -> Tuple[Tensor, Tensor]:
    """(primal_infeas, stationarity) residual norms for testing."""
    ineq = torch.clamp(torch.einsum("mkn,mn->mk", C, x) - b, min=0.0)
    box = torch.clamp(l - x, min=0.0) + torch.clamp(x - u.clamp(max=1e30), min=0.0)
    primal = torch.cat([ineq, box], dim=1).abs().amax(dim=1)
    # stationarity is only checkable with duals; report gradient-projection residual
    grad = torch.einsum("mij,mj->mi", H, x) + g
    x_step = torch.clamp(x - grad, l, u.clamp(max=1e30))
    stat = (x - x_step).abs().amax(dim=1)
    return primal, stat
