"""GCBF+ — the headline algorithm (reference ``gcbfplus/algo/gcbf_plus.py``).

Differences vs GCBF (reference lines cited):
  - AdamW (wd=1e-3) + finite-guard (36-139)
  - target CBF net with polyak tau=0.5 (116, 188-191, 228)
  - safe labels from the horizon backprop of unsafe flags (160-174)
  - action label is the CBF-QP-rectified action u_qp computed from the
    TARGET CBF (193-211), solved by the batched ProxQP op (K11)
  - h_dot loss stop-gradient split for unlabeled samples (398-408)
  - act = 2*actor + u_ref (176-180)
"""
from __future__ import annotations

import copy
import os
from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor
from torch.func import functional_call

from .. import ops
from ..ops.qp import proxqp_solve
from ..parallel import dp
from ..trainer.data import FlatBatch, Rollout
from ..utils.graph import GraphBatch
from .gcbf import GCBF, _hinge_safe, _hinge_unsafe
from .utils import clip_grads_, horizon_safe_mask, polyak_, step_if_finite


class GCBFPlus(GCBF):
    def __init__(self, *args, horizon: int = 32, **kwargs):
        kwargs.setdefault("loss_h_dot_coef", 0.2)
        super().__init__(*args, **kwargs)
        self.horizon = horizon
        self.cbf_tgt = copy.deepcopy(self.cbf)
        for p in self.cbf_tgt.parameters():
            p.requires_grad_(False)
            if hasattr(p, "_bf"):  # never share the live net's bf16 shadow
                del p._bf
        self.qp_iters = 150
        self.qp_relax_penalty = 1e3

    def _create_optimizers(self):
        """AdamW wd=1e-3 (reference gcbf_plus.py:109,127)."""
        if self._use_fused_optim():
            self._make_fused_optimizers(weight_decay=1e-3)
        else:
            self.cbf_optim = torch.optim.AdamW(self.cbf.parameters(), lr=self.lr_cbf,
                                               weight_decay=1e-3)
            self.actor_optim = torch.optim.AdamW(self.actor.parameters(), lr=self.lr_actor,
                                                 weight_decay=1e-3)

    @property
    def config(self) -> dict:
        c = super().config
        c["horizon"] = self.horizon
        return c

    # ---- update ----------------------------------------------------------
    def update(self, rollout: Rollout, step: int) -> dict:
        env = self._env
        g = rollout.graph_at(env)
        b, T = rollout.rewards.shape[:2]
        unsafe = env.unsafe_mask(g).reshape(b, T, self.n_agents)
        safe = horizon_safe_mask(unsafe, self.horizon)
        batch = self._sample_batch(rollout, safe, unsafe)
        info = self._update_nets(batch)
        polyak_(self.cbf_tgt, self.cbf, 0.5)
        return info

    def _sample_batch(self, rollout: Rollout, safe: Tensor, unsafe: Tensor) -> FlatBatch:
        """Reference gcbf_plus.py:232-280 (memory sampled BEFORE appending)."""
        if self.buffer.n_data > self.batch_size:
            mem_r, mem_s, mem_u = self.buffer.sample(rollout.length, self.rng)
            if dp.all_agree(self.unsafe_buffer.length > 0):
                unsafe_flat = self.unsafe_buffer.sample(
                    rollout.length * rollout.time_horizon, self.rng
                )
            else:
                unsafe_flat = self._flat_from_rollout(mem_r, mem_s, mem_u)
            self._append_buffers(rollout, safe, unsafe)
            flat = FlatBatch.cat(
                [self._flat_from_rollout(mem_r, mem_s, mem_u),
                 self._flat_from_rollout(rollout, safe, unsafe)]
            )
            return FlatBatch.cat([unsafe_flat, flat])
        self._append_buffers(rollout, safe, unsafe)
        return self._flat_from_rollout(rollout, safe, unsafe)

    def _update_nets(self, batch: FlatBatch) -> dict:
        u_qp = self._get_b_u_qp(batch, n_chunks=8)
        batch = batch._replace(u_qp=u_qp)
        info = {}
        for ep in range(self.inner_epoch):
            perm = torch.from_numpy(self.rng.permutation(batch.n)).to(batch.states.device)
            n_mb = max(1, batch.n // self.batch_size)
            chunks = torch.chunk(perm, n_mb)
            for i, mb_idx in enumerate(chunks):
                last = ep == self.inner_epoch - 1 and i == len(chunks) - 1
                if not last and self._graphed_mb().run(batch, mb_idx):
                    continue
                info = self._update_minibatch(batch[mb_idx], want_info=last)
        # grad norms of the reported (last) minibatch (reference logs them,
        # gcbf_plus.py:439)
        for name, opt in (("cbf", self.cbf_optim), ("actor", self.actor_optim)):
            norm = getattr(opt, "last_norm", None)
            if norm is not None:
                info[f"grad_norm/{name}"] = float(norm)
        return info

    # ---- QP labels (reference :193-211, 299-352) ---------------------------
    def _get_b_u_qp(self, batch: FlatBatch, n_chunks: int = 8) -> Tensor:
        outs = []
        n = batch.n
        chunk = max(1, (n + n_chunks - 1) // n_chunks)
        for i in range(0, n, chunk):
            outs.append(self.get_qp_action(batch[i : i + chunk])[0])
        return torch.cat(outs, dim=0)

    def get_qp_action(self, batch_or_graph, relax_penalty: Optional[float] = None
                      ) -> Tuple[Tensor, Tensor]:
        """Batched CBF-QP: min ||u - u_ref||^2 + 10||r||^2 + penalty*r
        s.t. -Lg_h u - r <= Lf_h + alpha*0.1*h, u in box, r >= 0.
        Returns (u_opt (M, N, nu), r (M, N))."""
        env = self._env
        if isinstance(batch_or_graph, FlatBatch):
            graph = batch_or_graph.graph(env)
        else:
            graph = batch_or_graph
        relax = self.qp_relax_penalty if relax_penalty is None else relax_penalty
        M, N, nu = graph.batch_size, self.n_agents, self.action_dim

        h, h_x = self.cbf_and_jacobian(graph, self.cbf_tgt)  # (M,N), (M,N,N,S)
        agent = graph.agent_states
        f, gdyn = env.control_affine_dyn(agent)  # (M,N,S), (M,N,S,nu)
        Lf_h = torch.einsum("mijs,mjs->mi", h_x, f)
        Lg_h = torch.einsum("mijs,mjsu->miju", h_x, gdyn).reshape(M, N, N * nu)

        u_lb, u_ub = env.action_lim()
        dev = agent.device
        u_lb = u_lb.to(dev).repeat(N)
        u_ub = u_ub.to(dev).repeat(N)
        u_ref = env.u_ref(graph).reshape(M, N * nu)

        nv = N * nu + N
        H = torch.eye(nv, device=dev).expand(M, nv, nv).clone()
        H[:, N * nu :, N * nu :] *= 10.0
        gvec = torch.cat([-u_ref, relax * torch.ones(M, N, device=dev)], dim=1)
        eyeN = torch.eye(N, device=dev).expand(M, N, N)
        C = -torch.cat([Lg_h, eyeN], dim=2)  # (M, N, nv)
        bvec = Lf_h + self.alpha * 0.1 * h
        l_box = torch.cat([u_lb, torch.zeros(N, device=dev)]).expand(M, nv)
        u_box = torch.cat([u_ub, torch.full((N,), float("inf"), device=dev)]).expand(M, nv)

        x = proxqp_solve(H, gvec, C, bvec, l_box, u_box, iters=self.qp_iters)
        u_opt = x[:, : N * nu].reshape(M, N, nu)
        r = x[:, N * nu :]
        return u_opt, r

    def cbf_and_jacobian(self, graph: GraphBatch, cbf_net) -> Tuple[Tensor, Tensor]:
        """h (M, N) and dh_i/dx_j (M, N, N, S) w.r.t. agent states.

        Fast path (gnn_layers == 1): with one message-passing round, h_i
        depends only on edge slots of receiver i, so ONE backward pass w.r.t.
        the dense edge features yields every per-agent gradient; the
        edge-build jacobian (state diff + position clip) is applied
        analytically (reference does N reverse passes via jax.jacobian,
        gcbf_plus.py:310-317).

        General path (gnn_layers > 1): ONE backward over an N-replicated
        batch — replica (m, i) selects output h_i, so the whole (N, N)
        jacobian block falls out of a single fused fwd+bwd instead of N
        sequential reverse passes (the reference pays N passes via
        jax.jacobian, gcbf_plus.py:310-317; callers chunk M, so the N-fold
        activation memory is bounded).
        """
        env = self._env
        M, N, S = graph.batch_size, self.n_agents, graph.state_dim
        states = graph.states.detach()
        with torch.enable_grad():
            if self.gnn_layers == 1 and getattr(env, "analytic_edge_jac", True):
                e = env.edge_feats(graph, states).detach().requires_grad_(True)
                h = cbf_net(graph, e).squeeze(-1)  # (M, N)
                (ge,) = torch.autograd.grad(h.sum(), e)  # (M, N, D, E)
                h_x = env.edge_grad_to_state_jac(graph, states, ge)
                return h.detach(), h_x
            # batched-replica path: (M*N, V, S), replica k = m*N + i
            st = states.repeat_interleave(N, dim=0).requires_grad_(True)
            g_rep = GraphBatch(states=st,
                               mask=graph.mask.repeat_interleave(N, dim=0),
                               n_agents=N, n_rays=graph.n_rays)
            e = env.edge_feats(g_rep, st)
            h_rep = cbf_net(g_rep, e).squeeze(-1)  # (M*N, N)
            row = torch.arange(N, device=st.device).repeat(M)
            sel = h_rep.gather(1, row[:, None]).sum()
            (gs,) = torch.autograd.grad(sel, st)
            h_x = gs[:, :N].reshape(M, N, N, S)
            h = h_rep.detach().reshape(M, N, N)[:, 0]
        return h, h_x

    # ---- loss (reference :354-431) ----------------------------------------
    def _loss(self, mb: FlatBatch, want_info: bool = True) -> Tuple[Tensor, dict]:
        env = self._env
        g = mb.graph(env)
        B = g.batch_size
        e, mi = self._net_inputs(g)
        safe_m = mb.safe.reshape(-1)
        unsafe_m = mb.unsafe.reshape(-1)

        # action = 2*actor + u_ref (the deployed policy); h and h_next in ONE
        # batched CBF forward (2B graphs): halves the GEMM call count and
        # doubles M for better CU fill
        raw = self.actor(g, e, msg_in=mi)
        prep = env.loss_prep(g, raw) if hasattr(env, "loss_prep") else None
        if prep is not None:
            # fused K16 kernel: u_ref + action clamp + euler + [cur; next]
            action, big_states = prep
            next_g = GraphBatch(states=big_states[B:], mask=g.mask,
                                n_agents=g.n_agents, n_rays=g.n_rays)
        else:
            action = 2 * raw + env.u_ref(g)
            next_g = env.forward_graph(g, action)
            big_states = torch.cat([g.states, next_g.states])
        big = GraphBatch(
            states=big_states,
            mask=torch.cat([g.mask, g.mask]),
            n_agents=g.n_agents, n_rays=g.n_rays,
        )
        e_big, mi_big = self._net_inputs(big)
        gated = (g.states.is_cuda and ops.hip_available() and self.gnn_layers == 1
                 and not os.environ.get("GCBF_NO_GATED_NG"))
        if gated:
            # The stop-gradient CBF evaluation has IDENTICAL VALUES to h_next
            # (same params — stop_gradient only cuts the param backward). So:
            # ONE batched forward over [g; next_g], where the next_g half's
            # parameter gradients are row-gated to labeled agents only
            # (ops.fused_linear row_gate). The actor gradient (through
            # next_g's states) flows for ALL rows, exactly as the reference's
            # two-evaluation scheme (gcbf_plus.py:398-408) — and one full
            # B-sized CBF forward+backward per minibatch disappears.
            N = self.n_agents
            labeled = (mb.safe | mb.unsafe)  # (B, N)
            wgate = torch.cat([torch.ones_like(labeled), labeled], dim=0)
            h_both = self.cbf(big, e_big, msg_in=mi_big,
                              row_gate=wgate).squeeze(-1)  # (2B, N)
            h = h_both[:B].reshape(-1)
            h_next = h_both[B:].reshape(-1)
            h_next_ng = h_next  # same tensor: the loss kernel's h_ng cotangent
            # (unlabeled rows) and h_next cotangent (labeled rows) are
            # row-disjoint and autograd sums them into one backward
        else:
            h_both = self.cbf(big, e_big, msg_in=mi_big).squeeze(-1)  # (2B, N)
            h = h_both[:B].reshape(-1)
            h_next = h_both[B:].reshape(-1)

            # stop-gradient branch: CBF params detached, actor path alive; the
            # VALUE equals h_next (same params) but gradients route differently
            det_params = {k: v.detach() for k, v in self.cbf.named_parameters()}
            e2 = None if e_big is None else e_big[B:]
            mi2 = None if mi_big is None else mi_big[B:]
            h_next_ng = functional_call(
                self.cbf, det_params, (next_g, e2), {"msg_in": mi2}
            ).squeeze(-1).reshape(-1)

        if g.states.is_cuda and ops.hip_available():
            # fused loss kernel (K10): identical math, 2 kernels vs ~160
            nu = action.shape[-1]
            total, parts = ops.gcbf_plus_loss(
                h, h_next, h_next_ng, action.reshape(-1, nu), mb.u_qp.reshape(-1, nu),
                safe_m, unsafe_m, env.dt, self.alpha, self.eps,
                self.loss_action_coef, self.loss_unsafe_coef, self.loss_safe_coef,
                self.loss_h_dot_coef,
            )
            info = {}
            if want_info:
                p = parts.tolist()
                info = {
                    "loss/action": p[0], "loss/unsafe": p[1], "loss/safe": p[2],
                    "loss/h_dot": p[3], "loss/total": float(total.detach()),
                    "acc/unsafe": p[4], "acc/safe": p[5], "acc/h_dot": p[6],
                    "acc/unsafe_data_ratio": p[7],
                }
            return total, info

        h_dot = (h_next - h) / env.dt
        h_dot_ng = (h_next_ng - h.detach()) / env.dt
        loss_unsafe, acc_unsafe = _hinge_unsafe(h, unsafe_m, self.eps)
        loss_safe, acc_safe = _hinge_safe(h, safe_m, self.eps)

        labeled = safe_m | unsafe_m
        val = torch.relu(-h_dot - self.alpha * h + self.eps)
        val_ng = torch.relu(-h_dot_ng - self.alpha * h + self.eps)
        loss_h_dot = torch.where(labeled, val, val_ng).mean()
        acc_h_dot = (h_dot + self.alpha * h > 0).float().mean()

        loss_action = (action - mb.u_qp).square().sum(-1).mean()

        total = (
            self.loss_action_coef * loss_action
            + self.loss_unsafe_coef * loss_unsafe
            + self.loss_safe_coef * loss_safe
            + self.loss_h_dot_coef * loss_h_dot
        )
        info = {}
        if want_info:
            with torch.no_grad():
                info = {
                    "loss/action": float(loss_action), "loss/unsafe": float(loss_unsafe),
                    "loss/safe": float(loss_safe), "loss/h_dot": float(loss_h_dot),
                    "loss/total": float(total), "acc/unsafe": float(acc_unsafe),
                    "acc/safe": float(acc_safe), "acc/h_dot": float(acc_h_dot),
                    "acc/unsafe_data_ratio": float(unsafe_m.float().mean()),
                }
        return total, info
