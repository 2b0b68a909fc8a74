"""Kernel-level op surface for the MI355X build.

Every hot op has two backends:
  - CPU: plain PyTorch (fp32) — the numerics oracle and the no-GPU test path.
  - GPU (gfx950): hand-written HIP/CDNA4 kernels in gcbfplus_amd/ops/hip/,
    loaded as the in-tree extension ``gcbfplus_amd._C``.

On a CUDA/ROCm device these ops REQUIRE the extension: if it is missing we
raise instead of silently falling back to eager (per-project rule: the HIP
path must be the one that runs on GPU).

Op inventory (kernel numbering follows SURVEY.md §2.7):
  fused_linear        K3/K4: MFMA GEMM + bias + activation, with autograd
                      (backward: transposed-B dX, deterministic split-M dW,
                      upstream-activation fold, direct grad accumulation)
  masked_softmax_aggr K3/K4: per-receiver masked softmax + weighted message sum
  edge_msg_in         K2/K15: fused layer-0 GNN input build (+ analytic bwd)
  raytrace_rect       K1: 2D LiDAR fan vs rectangle set (no grad)
  gcbf_plus_loss      K10: all GCBF+ hinge losses + metrics in one kernel pair
  di_loss_prep        K10 prologue: u_ref + action clamp + euler + [cur; next]
  proxqp_solve        K11: batched dense QP (labels; no grad)
  (env_step.hip K5-K8 is launched via env._step_fused; optimizer.hip K13 via
  ops.optim.FusedAdamW)
"""
from __future__ import annotations

import os
from typing import Optional

import torch
from torch import Tensor

_EXT = None
_EXT_ERR: Optional[str] = None
_ZBIAS = {}


def _zero_bias(n: int, device) -> "Tensor":
    key = (n, str(device))
    z = _ZBIAS.get(key)
    if z is None:
        z = _ZBIAS[key] = torch.zeros(n, device=device)
    return z

ACT_NONE, ACT_RELU, ACT_TANH = 0, 1, 2


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from gcbfplus_amd import _C  # built in-tree by setup.py / __graft_entry__.build()

        _EXT = _C
    except ImportError as e:  # pragma: no cover
        _EXT_ERR = str(e)
    return _EXT


def hip_available() -> bool:
    return _load_ext() is not None


def _require_ext():
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "gcbfplus_amd._C HIP extension is required on GPU but could not be "
            f"imported: {_EXT_ERR}. Build it with `python setup.py build_ext --inplace` "
            "(or __graft_entry__.build())."
        )
    return ext


def _apply_act(y: Tensor, act: int) -> Tensor:
    if act == ACT_RELU:
        return torch.relu(y)
    if act == ACT_TANH:
        return torch.tanh(y)
    return y


# --------------------------------------------------------------------------
# fused_linear: y = act(x @ w + b)
# --------------------------------------------------------------------------
class _FusedLinearHIP(torch.autograd.Function):
    """GPU path. x: (M, K) bf16 (or f32, cast), w: (K, N) f32 master, b: (N,) f32.

    Forward runs the hand-written MFMA GEMM (bf16 in, f32 accumulate) with the
    bias+activation fused into the epilogue. Backward reuses the same GEMM for
    dx = dz @ w^T and a deterministic split-M reduction kernel for
    dw = x^T @ dz (+ db = colsum dz).
    """

    @staticmethod
    def forward(ctx, x: Tensor, w: Tensor, b: Optional[Tensor], act: int,
                row_gate: Optional[Tensor] = None):
        ext = _require_ext()
        x_bf = x if x.dtype == torch.bfloat16 else x.to(torch.bfloat16)
        # FusedAdamW maintains a bf16 shadow per param (p._bf, updated inside
        # the optimizer kernel) — skip the per-call cast when present
        w_bf = getattr(w, "_bf", None)
        if w_bf is None:
            w_bf = w.to(torch.bfloat16)
        bias = b if b is not None else _zero_bias(w.shape[1], w.device)
        y = ext.gemm_bias_act(x_bf.contiguous(), w_bf.contiguous(), bias.contiguous(), act)
        ctx.save_for_backward(x_bf, w_bf, y)
        ctx.act = act
        ctx.x_dtype = x.dtype
        ctx.has_bias = b is not None
        # direct-accumulate target: when w/b are leaf params whose .grad is a
        # live buffer (FusedAdamW flat-gradient views), backward writes dW/db
        # straight into it (+=) and returns None — skips the AccumulateGrad
        # add_ kernels (~28 per minibatch across the three networks)
        ctx.acc = None
        if (b is not None and w.is_leaf and b.is_leaf
                and w.grad is not None and b.grad is not None
                and w.grad.is_contiguous() and b.grad.is_contiguous()):
            ctx.acc = (w, b)
        # per-row dW/db stop-gradient mask (dX unaffected): GCBF+ unlabeled
        # h_dot rows — rows with gate=False contribute nothing to the params
        ctx.row_gate = row_gate
        return y

    @staticmethod
    def backward(ctx, dy: Tensor):
        ext = _require_ext()
        x_bf, w_bf, y = ctx.saved_tensors
        if y.dtype == torch.float32:  # small-N f32 head output path
            if ctx.act == ACT_RELU:
                dz_f = dy * (y > 0)
            elif ctx.act == ACT_TANH:
                dz_f = dy * (1.0 - y * y)
            else:
                dz_f = dy
            dz = dz_f.to(torch.bfloat16).contiguous()
            actin, yact = ACT_NONE, dz
        else:
            # activation backward dz = dy * act'(y) is FOLDED into the two
            # consumer GEMMs' operand stages (same f32 math, same bf16
            # rounding as a materialized act_bwd pass)
            dz = dy.contiguous().to(torch.bfloat16)
            actin, yact = ctx.act, y
            if actin != ACT_NONE and dz.shape[1] % 32 != 0:
                dz = ext.act_bwd(dz, y, actin)  # rare: narrow activated layer
                actin, yact = ACT_NONE, dz
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            if dz.shape[1] % 32 == 0:
                # dx = dz @ w^T via the transposed B-stage kernel: no
                # materialized w^T copy per backward
                dx = ext.gemm_bt(dz, w_bf, yact, actin)
            else:
                # small-N heads: pad path with an explicit transpose
                wt = w_bf.t().contiguous()
                dx = ext.gemm_bias_act(dz, wt, _zero_bias(wt.shape[1], wt.device), ACT_NONE)
            dx = dx.to(ctx.x_dtype)
        if ctx.needs_input_grad[1] or ctx.needs_input_grad[2]:
            rg = ctx.row_gate
            if ctx.acc is not None:
                wp, bp = ctx.acc
                ext.gemm_tn_acc(x_bf, dz, yact, actin, wp.grad, bp.grad, rg)
            else:
                dw, db = ext.gemm_tn(x_bf, dz, yact, actin, rg)  # f32 (K,N), (N,)
        if not ctx.has_bias:
            db = None
        return dx, dw, db, None, None


class _FusedLinearOneHotHIP(torch.autograd.Function):
    """y = act(x @ w[oh:] + (b + w[oh-1])) for the agents-only GNN update
    layer whose first `oh` input features are the constant agent one-hot
    [0,0,1]. Replaces the eager `w[3:]` slice + `b + w[2]` add whose autograd
    backward costs ~10 small kernels per call (zeros-pad of the sliced dW,
    scatter, AccumulateGrad adds): here dW accumulates straight into
    w.grad[oh:] and db into BOTH b.grad and w.grad[oh-1] inside the dW
    reduction (gemm_tn_acc2)."""

    @staticmethod
    def forward(ctx, x: Tensor, w: Tensor, b: Tensor, act: int, oh: int,
                row_gate: Optional[Tensor] = None):
        ext = _require_ext()
        x_bf = x if x.dtype == torch.bfloat16 else x.to(torch.bfloat16)
        w_bf = getattr(w, "_bf", None)
        if w_bf is None:
            w_bf = w.to(torch.bfloat16)
        w_bf3 = w_bf[oh:].contiguous() if not w_bf[oh:].is_contiguous() else w_bf[oh:]
        bias_eff = b + w[oh - 1]
        y = ext.gemm_bias_act(x_bf.contiguous(), w_bf3, bias_eff.contiguous(), act)
        ctx.save_for_backward(x_bf, w_bf3, y)
        ctx.act = act
        ctx.oh = oh
        ctx.x_dtype = x.dtype
        ctx.direct = (w.is_leaf and b.is_leaf and w.grad is not None
                      and b.grad is not None and w.grad.is_contiguous()
                      and b.grad.is_contiguous())
        ctx.params = (w, b) if ctx.direct else None
        ctx.row_gate = row_gate
        return y

    @staticmethod
    def backward(ctx, dy: Tensor):
        ext = _require_ext()
        x_bf, w_bf3, y = ctx.saved_tensors
        oh = ctx.oh
        if y.dtype == torch.float32:
            if ctx.act == ACT_RELU:
                dz_f = dy * (y > 0)
            elif ctx.act == ACT_TANH:
                dz_f = dy * (1.0 - y * y)
            else:
                dz_f = dy
            dz = dz_f.to(torch.bfloat16).contiguous()
            actin, yact = ACT_NONE, dz
        else:
            dz = dy.contiguous().to(torch.bfloat16)
            actin, yact = ctx.act, y
            if actin != ACT_NONE and dz.shape[1] % 32 != 0:
                dz = ext.act_bwd(dz, y, actin)
                actin, yact = ACT_NONE, dz
        dx = dw_full = db_out = None
        if ctx.needs_input_grad[0]:
            dx = ext.gemm_bt(dz, w_bf3, yact, actin).to(ctx.x_dtype)
        if ctx.direct:
            w, b = ctx.params
            ext.gemm_tn_acc2(x_bf, dz, yact, actin, w.grad[oh:], b.grad,
                             w.grad[oh - 1], ctx.row_gate)
        else:
            dw, db = ext.gemm_tn(x_bf, dz, yact, actin, ctx.row_gate)
            dw_full = torch.zeros(oh + dw.shape[0], dw.shape[1],
                                  device=dw.device, dtype=dw.dtype)
            dw_full[oh:] = dw
            dw_full[oh - 1] = db
            db_out = db
        return dx, dw_full, db_out, None, None, None


def fused_linear_onehot(x: Tensor, w: Tensor, b: Tensor, act: int, oh: int = 3,
                        row_gate: Optional[Tensor] = None) -> Tensor:
    """GPU-only one-hot fold (see _FusedLinearOneHotHIP). Callers fall back
    to fused_linear(x, w[oh:], b + w[oh-1], act) on CPU."""
    lead = x.shape[:-1]
    x2 = x.reshape(-1, x.shape[-1])
    y = _FusedLinearOneHotHIP.apply(x2, w, b, act, oh, row_gate)
    return y.reshape(*lead, w.shape[1])


def fused_linear(x: Tensor, w: Tensor, b: Optional[Tensor], act: int = ACT_NONE,
                 row_gate: Optional[Tensor] = None) -> Tensor:
    """act(x @ w + b). x: (..., K); w: (K, N) fp32 master weight; b: (N,) fp32.

    GPU: bf16 MFMA kernel. CPU: fp32 torch (autograd oracle).
    row_gate (flat rows,) bool: rows with False contribute nothing to dW/db
    (dX unaffected) — per-sample parameter stop-gradient.
    """
    lead = x.shape[:-1]
    x2 = x.reshape(-1, x.shape[-1])
    if x2.shape[-1] != w.shape[0]:
        if x2.shape[-1] < w.shape[0]:
            # weight padded at init (Dense pad_to) but input unpadded (eager
            # CPU edge path): the dropped rows are zero, so slicing is exact
            w = w.narrow(0, 0, x2.shape[-1])
        else:
            # zero-padded input (fused edge_msg_in emits K padded to 32): pad
            # W rows to match — zeros x zeros contribute nothing, and autograd
            # slices dW back to the master shape
            w = torch.cat([w, w.new_zeros(x2.shape[-1] - w.shape[0], w.shape[1])], dim=0)
    if x2.is_cuda:
        y = _FusedLinearHIP.apply(x2, w, b, act, row_gate)
    else:
        assert row_gate is None, "row_gate is a GPU-path feature"
        y = torch.addmm(b, x2, w) if b is not None else x2 @ w
        y = _apply_act(y, act)
    return y.reshape(*lead, w.shape[1])


# --------------------------------------------------------------------------
# masked_softmax_aggr: attention aggregate over the dense edge-slot axis
# --------------------------------------------------------------------------
class _SoftmaxAggrHIP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate: Tensor, msg: Tensor, mask: Tensor):
        ext = _require_ext()
        gate_f = gate.to(torch.float32).contiguous()
        msg_bf = msg.to(torch.bfloat16).contiguous()
        aggr, attn = ext.softmax_aggr_fwd(gate_f, msg_bf, mask.contiguous())
        ctx.save_for_backward(attn, msg_bf, mask)
        ctx.gate_dtype = gate.dtype
        ctx.msg_dtype = msg.dtype
        return aggr

    @staticmethod
    def backward(ctx, daggr: Tensor):
        ext = _require_ext()
        attn, msg_bf, mask = ctx.saved_tensors
        dgate, dmsg = ext.softmax_aggr_bwd(
            daggr.to(torch.bfloat16).contiguous(), attn, msg_bf, mask
        )
        return dgate.to(ctx.gate_dtype), dmsg.to(ctx.msg_dtype), None


def masked_softmax_aggr(gate: Tensor, msg: Tensor, mask: Tensor) -> Tensor:
    """Per-receiver attention aggregation (dense replacement for the
    reference's jraph.segment_softmax + segment_sum, nn/gnn.py:65-72).

    gate: (B, N, D) raw attention logits
    msg:  (B, N, D, C) messages
    mask: (B, N, D) bool; masked slots get zero attention.
    Returns aggr: (B, N, C) = sum_d softmax_d(gate | mask) * msg.
    Rows with no active slot return zeros.
    """
    if gate.is_cuda:
        return _SoftmaxAggrHIP.apply(gate, msg, mask)
    neg = torch.finfo(gate.dtype).min
    g = torch.where(mask, gate, torch.full_like(gate, neg))
    # numerically-stable masked softmax; all-masked rows -> zeros
    gmax = g.max(dim=-1, keepdim=True).values
    e = torch.exp(g - gmax) * mask.to(gate.dtype)
    denom = e.sum(dim=-1, keepdim=True).clamp_min(1e-20)
    attn = e / denom
    return torch.einsum("bnd,bndc->bnc", attn.to(msg.dtype), msg)


# --------------------------------------------------------------------------
# edge_msg_in: fused first-layer GNN input (K2/K15)
# --------------------------------------------------------------------------
class _EdgeMsgInHIP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, states: Tensor, n_agents: int, n_rays: int, pos_dim: int,
                k_pad: int, comm: float, mode: int):
        ext = _require_ext()
        st = states.contiguous()
        X = ext.edge_msg_in_fwd(st, n_agents, n_rays, pos_dim, k_pad, comm, mode)
        ctx.save_for_backward(st)
        ctx.meta = (n_agents, n_rays, pos_dim, comm, mode)
        return X

    @staticmethod
    def backward(ctx, dX: Tensor):
        ext = _require_ext()
        (st,) = ctx.saved_tensors
        n, r, pdim, comm, mode = ctx.meta
        dstates = ext.edge_msg_in_bwd(st, dX.contiguous().to(torch.bfloat16), n, r,
                                      pdim, comm, mode)
        return dstates, None, None, None, None, None, None


def edge_msg_in(states: Tensor, n_agents: int, n_rays: int, pos_dim: int,
                comm: float, k_pad: Optional[int] = None, mode: int = 0) -> Tensor:
    """Fused layer-0 GNN input: per edge slot
    [clip(recv - send) | sender one-hot | recv one-hot | 0 pad], (B, N, D, KP).

    Valid for the 'state-diff + position-clip' edge-feature family
    (SI/DI/LinearDrone; reference double_integrator.py:275-286 + one-hot
    node feats :288-295). GPU: bf16 kernel pair; CPU: composed fp32 torch.
    """
    B, V, S = states.shape
    K = S + 6
    if k_pad is None:
        k_pad = (K + 31) // 32 * 32
    if states.is_cuda:
        return _EdgeMsgInHIP.apply(states, n_agents, n_rays, pos_dim, k_pad, comm, mode)
    # CPU compose (fp32)
    if mode == 1:  # DubinsCar edge-state transform [x, y, v cos, v sin]
        th, v = states[..., 2], states[..., 3]
        states = torch.stack(
            [states[..., 0], states[..., 1], v * torch.cos(th), v * torch.sin(th)], dim=-1
        )
    n, r = n_agents, n_rays
    recv = states[:, :n, None, :]
    senders = torch.cat(
        [
            states[:, None, :n].expand(B, n, n, S),
            states[:, n : 2 * n, None, :],
            states[:, 2 * n :].reshape(B, n, r, S),
        ],
        dim=2,
    )
    e = recv - senders
    pos = e[..., :pos_dim]
    nrm = torch.sqrt(1e-6 + (pos * pos).sum(-1, keepdim=True))
    coef = torch.where(nrm > comm, comm / torch.clamp(nrm, min=comm), torch.ones_like(nrm))
    e = torch.cat([pos * coef, e[..., pos_dim:]], dim=-1)
    D = n + 1 + r
    oh = torch.zeros(n, D, 6, dtype=states.dtype, device=states.device)
    oh[:, :n, 2] = 1.0  # agent senders: 001
    oh[:, n, 1] = 1.0  # goal: 010
    oh[:, n + 1 :, 0] = 1.0  # obs: 100
    oh[:, :, 5] = 1.0  # receiver is always an agent: 001
    out = torch.cat([e, oh[None].expand(B, n, D, 6)], dim=-1)
    if k_pad > K:
        out = torch.cat([out, out.new_zeros(B, n, D, k_pad - K)], dim=-1)
    return out


# --------------------------------------------------------------------------
# raytrace (K1) — no autograd
# --------------------------------------------------------------------------
def raytrace_rect(pos: Tensor, points: Tensor, n_rays: int, sense_range: float) -> Tensor:
    """2D LiDAR fan: pos (B,N,2) origins, points (B,K,4,2) rectangle corners.
    Returns hits (B,N,R,2). GPU: HIP kernel; CPU: composed torch (obstacle.py)."""
    ext = _load_ext()
    if pos.is_cuda:
        _require_ext()
        return _EXT.raytrace_rect(pos.contiguous(), points.contiguous(), n_rays, sense_range)
    raise NotImplementedError("CPU path goes through env.get_lidar composition")


def raytrace_sphere_topk(pos: Tensor, centers: Tensor, radii: Tensor, n_beams: int,
                         topk: int, sense_range: float) -> Tensor:
    """3D LiDAR theta x phi fan + poles vs spheres, fused with stable top-k
    closest-hit selection (reference env/utils.py:49-79 + obstacle.py:237-270
    + the argsort top-k of utils.py:127-131). pos (B,N,3), centers (B,K,3),
    radii (B,K) -> hits (B,N,topk,3). GPU only (no grad)."""
    if pos.is_cuda:
        _require_ext()
        return _EXT.raytrace_sphere_topk(pos.contiguous(), centers.contiguous(),
                                         radii.contiguous(), n_beams, topk, sense_range)
    raise NotImplementedError("CPU path goes through env.get_lidar composition")


# --------------------------------------------------------------------------
# fused GCBF+ loss (K10)
# --------------------------------------------------------------------------
class _GCBFLossHIP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h, h_next, h_ng, action, u_qp, safe, unsafe, params):
        ext = _require_ext()
        dt, alpha, eps, c_act, c_unsafe, c_safe, c_hdot = params
        args = [t.contiguous() for t in (h, h_next, h_ng, action, u_qp)]
        out = ext.gcbf_loss_fwd(*args, safe.contiguous(), unsafe.contiguous(),
                                dt, alpha, eps, c_act, c_unsafe, c_safe, c_hdot)
        ctx.save_for_backward(*args, safe, unsafe, out)
        ctx.params = params
        parts = out[1:9]
        ctx.mark_non_differentiable(parts)
        return out[0], parts

    @staticmethod
    def backward(ctx, d_total, _d_parts):
        ext = _require_ext()
        h, h_next, h_ng, action, u_qp, safe, unsafe, out = ctx.saved_tensors
        dt, alpha, eps, c_act, c_unsafe, c_safe, c_hdot = ctx.params
        dh, dh_next, dh_ng, daction = ext.gcbf_loss_bwd(
            h, h_next, h_ng, action, u_qp, safe, unsafe, out,
            d_total.reshape(1).contiguous().float(),
            dt, alpha, eps, c_act, c_unsafe, c_safe, c_hdot,
        )
        return dh, dh_next, dh_ng, daction, None, None, None, None


def gcbf_plus_loss(h, h_next, h_ng, action, u_qp, safe, unsafe, dt, alpha, eps,
                   c_act, c_unsafe, c_safe, c_hdot):
    """Fused GCBF+ loss (reference gcbf_plus.py:364-431). Returns
    (total scalar, parts (8,) detached: [action, unsafe, safe, h_dot losses,
    acc_unsafe, acc_safe, acc_h_dot, unsafe_ratio])."""
    return _GCBFLossHIP.apply(h, h_next, h_ng, action, u_qp, safe, unsafe,
                              (dt, alpha, eps, c_act, c_unsafe, c_safe, c_hdot))


# --------------------------------------------------------------------------
# fused GCBF+ minibatch prologue for the DoubleIntegrator (K16)
# --------------------------------------------------------------------------
class _DILossPrepHIP(torch.autograd.Function):
    """u_ref (clipped-error LQR, reference env/double_integrator.py:332-338)
    -> action = clamp(2*raw + u_ref) -> euler next state (:112-143) ->
    big = [states; next_states]: one kernel instead of ~20 eager launches.
    Gradient flows to ``raw`` only (minibatch states are leaves)."""

    @staticmethod
    def forward(ctx, states, raw, K, n_agents, dt, inv_m, comm, vmax):
        ext = _require_ext()
        action, big = ext.di_loss_prep_fwd(
            states.contiguous(), raw.contiguous(), K.contiguous(),
            n_agents, dt, inv_m, comm, vmax)
        ctx.save_for_backward(states, raw, K, action)
        ctx.meta = (n_agents, dt, inv_m, comm, vmax)
        return action, big

    @staticmethod
    def backward(ctx, daction, dbig):
        ext = _require_ext()
        states, raw, K, action = ctx.saved_tensors
        n_agents, dt, inv_m, comm, vmax = ctx.meta
        if daction is None:
            daction = torch.zeros_like(action)
        if dbig is None:
            B, V, _ = states.shape
            dbig = states.new_zeros(2 * B, V, 4)
        draw = ext.di_loss_prep_bwd(
            states, raw, K, action, daction.contiguous(), dbig.contiguous(),
            n_agents, dt, inv_m, comm, vmax)
        return None, draw, None, None, None, None, None, None


def di_loss_prep(states: Tensor, raw: Tensor, K: Tensor, n_agents: int,
                 dt: float, inv_m: float, comm: float, vmax: float):
    """Returns (action (B,N,2), big_states (2B,V,4)). HIP on GPU; composed
    torch ops (same math, differentiable) on CPU."""
    if states.is_cuda and hip_available():
        return _DILossPrepHIP.apply(states, raw, K, n_agents, dt, inv_m, comm, vmax)
    N = n_agents
    agent = states[:, :N]
    goal = states[:, N:2 * N]
    err = goal - agent
    nrm = torch.linalg.vector_norm(err, dim=-1, keepdim=True).clamp_min(1e-9)
    emax = (err / nrm * comm).abs()
    uref = (torch.clamp(err, -emax, emax) @ K.t()).clamp(-1.0, 1.0)
    action = (2.0 * raw + uref).clamp(-1.0, 1.0)
    vel = (agent[..., 2:] + action * inv_m * dt).clamp(-vmax, vmax)
    pos = agent[..., :2] + agent[..., 2:] * dt
    nxt = torch.cat([torch.cat([pos, vel], -1), states[:, N:]], dim=1)
    return action, torch.cat([states, nxt], dim=0)
