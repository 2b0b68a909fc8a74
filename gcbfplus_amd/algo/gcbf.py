"""GCBF (predecessor algorithm) — reference ``gcbfplus/algo/gcbf.py``.

Jointly trains a graph CBF h and an actor by hinge losses against
on-the-fly safe/unsafe labels, with the action loss pulling toward u_ref.
"""
from __future__ import annotations

import os
import pickle
from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor
from torch.func import functional_call

from .. import ops
from ..ops.optim import FusedAdamW
from ..parallel import dp
from ..trainer.buffer import FlatSampleBuffer, MaskedRolloutBuffer
from ..trainer.data import FlatBatch, Rollout
from ..utils.graph import GraphBatch
from .base import MultiAgentController
from .module.cbf import CBFNet
from .module.policy import DeterministicPolicyNet
from .utils import (clip_grads_, load_flax_pickle, net_from_flax_tree,
                    net_to_flax_tree, step_if_finite)


class GCBF(MultiAgentController):
    def __init__(
        self,
        env,
        node_dim: int,
        edge_dim: int,
        state_dim: int,
        action_dim: int,
        n_agents: int,
        gnn_layers: int = 1,
        batch_size: int = 256,
        buffer_size: int = 512,
        lr_actor: float = 3e-5,
        lr_cbf: float = 3e-5,
        alpha: float = 1.0,
        eps: float = 0.02,
        inner_epoch: int = 8,
        loss_action_coef: float = 0.001,
        loss_unsafe_coef: float = 1.0,
        loss_safe_coef: float = 1.0,
        loss_h_dot_coef: float = 0.2,
        max_grad_norm: float = 2.0,
        seed: int = 0,
        online_pol_refine: bool = False,
        **kwargs,
    ):
        super().__init__(env, node_dim, edge_dim, action_dim, n_agents)
        self.batch_size = batch_size
        self.lr_actor, self.lr_cbf = lr_actor, lr_cbf
        self.alpha, self.eps = alpha, eps
        self.inner_epoch = inner_epoch
        self.loss_action_coef = loss_action_coef
        self.loss_unsafe_coef = loss_unsafe_coef
        self.loss_safe_coef = loss_safe_coef
        self.loss_h_dot_coef = loss_h_dot_coef
        self.gnn_layers = gnn_layers
        self.max_grad_norm = max_grad_norm
        self.seed = seed
        self.online_pol_refine = online_pol_refine

        dev = env.device
        torch.manual_seed(seed)
        self.cbf = CBFNet(node_dim, edge_dim, gnn_layers).to(dev)
        self.actor = DeterministicPolicyNet(node_dim, edge_dim, action_dim, gnn_layers).to(dev)
        self._create_optimizers()

        self.buffer = MaskedRolloutBuffer(size=buffer_size)
        self.unsafe_buffer = FlatSampleBuffer(size=buffer_size // 2)
        self.rng = np.random.default_rng(seed=seed + 1 + 7919 * dp.rank())
        self._mb_graph = None
        dp.broadcast_modules([self.cbf, self.actor])
        self._refresh_optim_bf16()

    def _refresh_optim_bf16(self):
        """Re-sync FusedAdamW bf16 weight shadows after out-of-band param
        writes (checkpoint load / DP broadcast)."""
        for opt in (getattr(self, "cbf_optim", None), getattr(self, "actor_optim", None)):
            if opt is not None and hasattr(opt, "refresh_bf16"):
                opt.refresh_bf16()

    def _graphed_mb(self):
        if self._mb_graph is None:
            from ..trainer.graphing import GraphedMinibatchStep

            self._mb_graph = GraphedMinibatchStep(self)
        return self._mb_graph

    def _use_fused_optim(self) -> bool:
        return self._env.device.type == "cuda" and ops.hip_available()

    def _create_optimizers(self):
        """Adam (reference gcbf.py:101,118); GPU: fused flat-buffer kernel
        with the clip+finite-guard folded in (K13)."""
        if self._use_fused_optim():
            self._make_fused_optimizers(weight_decay=0.0)
        else:
            self.cbf_optim = torch.optim.Adam(self.cbf.parameters(), lr=self.lr_cbf)
            self.actor_optim = torch.optim.Adam(self.actor.parameters(), lr=self.lr_actor)

    def _make_fused_optimizers(self, weight_decay: float):
        """Both nets' grad buffers are slices of ONE flat tensor (dp_gbuf):
        DP then averages everything with a single all-reduce (no cat, no
        extra copies; the bucket is HIP-graph capturable)."""
        dev = self._env.device
        n_cbf = sum(p.numel() for p in self.cbf.parameters() if p.requires_grad)
        n_act = sum(p.numel() for p in self.actor.parameters() if p.requires_grad)
        self.dp_gbuf = torch.zeros(n_cbf + n_act, device=dev)
        self.cbf_optim = FusedAdamW(self.cbf, self.lr_cbf, weight_decay,
                                    self.max_grad_norm, gflat_buf=self.dp_gbuf[:n_cbf])
        self.actor_optim = FusedAdamW(self.actor, self.lr_actor, weight_decay,
                                      self.max_grad_norm, gflat_buf=self.dp_gbuf[n_cbf:])

    # ---- config / io -----------------------------------------------------
    @property
    def config(self) -> dict:
        return {
            "batch_size": self.batch_size,
            "lr_actor": self.lr_actor,
            "lr_cbf": self.lr_cbf,
            "alpha": self.alpha,
            "eps": self.eps,
            "inner_epoch": self.inner_epoch,
            "loss_action_coef": self.loss_action_coef,
            "loss_unsafe_coef": self.loss_unsafe_coef,
            "loss_safe_coef": self.loss_safe_coef,
            "loss_h_dot_coef": self.loss_h_dot_coef,
            "gnn_layers": self.gnn_layers,
            "seed": self.seed,
            "max_grad_norm": self.max_grad_norm,
        }

    def save(self, save_dir: str, step: int):
        """Pickle the params as the reference's flax tree layout into
        <save_dir>/<step>/{actor.pkl, cbf.pkl} (reference gcbf.py:344-349)."""
        model_dir = os.path.join(save_dir, str(step))
        os.makedirs(model_dir, exist_ok=True)
        with open(os.path.join(model_dir, "actor.pkl"), "wb") as f:
            pickle.dump(net_to_flax_tree(self.actor, "PolicyHead", "OutputDense"), f)
        with open(os.path.join(model_dir, "cbf.pkl"), "wb") as f:
            pickle.dump(net_to_flax_tree(self.cbf, "CBFHead", "Dense_0"), f)

    def load(self, load_dir: str, step: int):
        """Loads either this framework's checkpoints or the reference's own
        pretrained pickles (jax arrays unpickled jax-free via
        load_flax_pickle)."""
        path = os.path.join(load_dir, str(step))
        net_from_flax_tree(self.actor, load_flax_pickle(os.path.join(path, "actor.pkl")),
                           "PolicyHead", "OutputDense")
        net_from_flax_tree(self.cbf, load_flax_pickle(os.path.join(path, "cbf.pkl")),
                           "CBFHead", "Dense_0")
        self._refresh_optim_bf16()

    # ---- full training resume (NOT in the reference, which saves params
    # only — gcbf.py:344-357 / SURVEY §5.4) --------------------------------
    def save_full(self, path: str, step: int):
        state = {
            "step": step,
            "cbf": self.cbf.state_dict(),
            "actor": self.actor.state_dict(),
            "cbf_optim": self.cbf_optim.state_dict(),
            "actor_optim": self.actor_optim.state_dict(),
            "rng": self.rng.bit_generator.state,
        }
        if hasattr(self, "cbf_tgt"):
            state["cbf_tgt"] = self.cbf_tgt.state_dict()
        torch.save(state, path)

    def load_full(self, path: str) -> int:
        state = torch.load(path, map_location=self._env.device, weights_only=False)
        self.cbf.load_state_dict(state["cbf"])
        self.actor.load_state_dict(state["actor"])
        self.cbf_optim.load_state_dict(state["cbf_optim"])
        self.actor_optim.load_state_dict(state["actor_optim"])
        self.rng.bit_generator.state = state["rng"]
        if "cbf_tgt" in state and hasattr(self, "cbf_tgt"):
            self.cbf_tgt.load_state_dict(state["cbf_tgt"])
        return state["step"]

    # ---- acting ----------------------------------------------------------
    def _edge_feats(self, graph: GraphBatch, states: Optional[Tensor] = None) -> Tensor:
        return self._env.edge_feats(graph, states)

    def _net_inputs(self, graph: GraphBatch, states: Optional[Tensor] = None):
        """(edge_feats, msg_in): the fused layer-0 input when the config
        allows (single GNN layer + state-diff edge family), else edge feats."""
        if self.gnn_layers == 1 and self._env.fused_edge:
            return None, self._env.edge_msg_in(graph, states)
        return self._edge_feats(graph, states), None

    def act(self, graph: GraphBatch) -> Tensor:
        if self.online_pol_refine:
            return self.online_policy_refinement(graph)
        with torch.no_grad():
            e, mi = self._net_inputs(graph)
            return 2 * self.actor(graph, e, msg_in=mi) + self._env.u_ref(graph)

    @torch.no_grad()
    def step(self, graph: GraphBatch) -> Tuple[Tensor, Tensor]:
        e, mi = self._net_inputs(graph)
        action = self.actor(graph, e, msg_in=mi)
        log_pi = torch.zeros_like(action)
        return 2 * action + self._env.u_ref(graph), log_pi

    def get_cbf(self, graph: GraphBatch) -> Tensor:
        e, mi = self._net_inputs(graph)
        return self.cbf(graph, e, msg_in=mi)

    def online_policy_refinement(self, graph: GraphBatch) -> Tensor:
        """Test-time gradient repair of the action (reference gcbf.py:161-201)."""
        env = self._env
        e = self._edge_feats(graph)
        with torch.no_grad():
            h = self.cbf(graph, e)
            u_ref = env.u_ref(graph)
            ng_ref = env.forward_graph(graph, u_ref)
            h_next_ref = self.cbf(ng_ref, self._edge_feats(ng_ref))
            viol_ref = torch.relu(-(h_next_ref - h) / env.dt - self.alpha * h)
            nn_action = 2 * self.actor(graph, e) + u_ref
            action = torch.where(viol_ref > 0, nn_action, u_ref)
        action = action.detach().clone().requires_grad_(True)
        lr, max_iter = 0.1, 30
        for _ in range(max_iter):
            ng = env.forward_graph(graph, action)
            h_next = self.cbf(ng, self._edge_feats(ng))
            val = torch.relu(-(h_next - h) / env.dt - self.alpha * h).mean()
            if val.item() <= 0:
                break
            (grad,) = torch.autograd.grad(val, action)
            with torch.no_grad():
                action -= lr * grad
        return action.detach()

    # ---- training --------------------------------------------------------
    def _flat_from_rollout(self, rollout: Rollout, safe: Tensor, unsafe: Tensor) -> FlatBatch:
        b, T = rollout.rewards.shape[:2]
        return FlatBatch(
            states=rollout.states.reshape(b * T, *rollout.states.shape[2:]),
            masks=rollout.masks.reshape(b * T, *rollout.masks.shape[2:]),
            safe=safe.reshape(b * T, -1),
            unsafe=unsafe.reshape(b * T, -1),
        )

    def _collect_masks(self, rollout: Rollout) -> Tuple[Tensor, Tensor]:
        """(safe, unsafe) of shape (b, T, N). GCBF labels are the env's
        instantaneous masks (reference gcbf.py:269-283 computes them
        per-minibatch; values are identical computed once here)."""
        env = self._env
        g = rollout.graph_at(env)
        b, T = rollout.rewards.shape[:2]
        unsafe = env.unsafe_mask(g).reshape(b, T, self.n_agents)
        safe = env.safe_mask(g).reshape(b, T, self.n_agents)
        return safe, unsafe

    def update(self, rollout: Rollout, step: int) -> dict:
        safe, unsafe = self._collect_masks(rollout)
        if self.buffer.n_data > self.batch_size:
            mem_r, mem_s, mem_u = self.buffer.sample(rollout.length // 2, self.rng)
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
            batch = FlatBatch.cat([unsafe_flat, flat])
        else:
            self._append_buffers(rollout, safe, unsafe)
            batch = self._flat_from_rollout(rollout, safe, unsafe)

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
        return info

    def _append_buffers(self, rollout: Rollout, safe: Tensor, unsafe: Tensor):
        self.buffer.append(rollout, safe, unsafe)
        row_mask = unsafe.any(dim=-1)  # (b, T)
        flat = self._flat_from_rollout(rollout, safe, unsafe)
        sel = row_mask.reshape(-1)
        self.unsafe_buffer.append(flat[sel])

    def _loss(self, mb: FlatBatch, want_info: bool = True) -> Tuple[Tensor, dict]:
        """Reference gcbf.py:258-321 loss; action target is u_ref and the
        action fed to forward_graph is the RAW actor output."""
        env = self._env
        g = mb.graph(env)
        e, mi = self._net_inputs(g)
        h = self.cbf(g, e, msg_in=mi).squeeze(-1).reshape(-1)
        safe_m = mb.safe.reshape(-1)
        unsafe_m = mb.unsafe.reshape(-1)

        loss_unsafe, acc_unsafe = _hinge_unsafe(h, unsafe_m, self.eps)
        loss_safe, acc_safe = _hinge_safe(h, safe_m, self.eps)

        action = self.actor(g, e, msg_in=mi)
        next_g = env.forward_graph(g, action)
        e2, mi2 = self._net_inputs(next_g)
        h_next = self.cbf(next_g, e2, msg_in=mi2).squeeze(-1).reshape(-1)
        h_dot = (h_next - h) / env.dt
        val = torch.relu(-h_dot - self.alpha * h + self.eps)
        loss_h_dot = val.mean()
        acc_h_dot = (h_dot + self.alpha * h > 0).float().mean()

        u_ref = env.u_ref(g)
        loss_action = (action - u_ref).square().sum(-1).mean()

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

    def _update_minibatch(self, mb: FlatBatch, want_info: bool = True) -> dict:
        total, info = self._loss(mb, want_info)
        self.cbf_optim.zero_grad(set_to_none=False)
        self.actor_optim.zero_grad(set_to_none=False)
        total.backward()
        if isinstance(self.cbf_optim, FusedAdamW):
            dp.allreduce_mean_flat([self.dp_gbuf])
            cbf_norm = self.cbf_optim.step()
            actor_norm = self.actor_optim.step()
        else:
            cbf_params = [p for p in self.cbf.parameters()]
            actor_params = [p for p in self.actor.parameters()]
            dp.allreduce_mean_grads(cbf_params + actor_params)
            cbf_norm = clip_grads_(cbf_params, self.max_grad_norm)
            actor_norm = clip_grads_(actor_params, self.max_grad_norm)
            step_if_finite(self.cbf_optim, cbf_params, cbf_norm)
            step_if_finite(self.actor_optim, actor_params, actor_norm)
        if want_info:
            info["grad_norm/cbf"] = float(cbf_norm)
            info["grad_norm/actor"] = float(actor_norm)
        return info


def _hinge_unsafe(h: Tensor, unsafe_m: Tensor, eps: float) -> Tuple[Tensor, Tensor]:
    """sum(relu(h+eps) over unsafe)/count (reference gcbf_plus.py:374-380)."""
    h_unsafe = torch.where(unsafe_m, h, torch.full_like(h, -eps * 2))
    loss = torch.relu(h_unsafe + eps).sum() / (unsafe_m.float().sum() + 1e-6)
    acc_mask = torch.where(unsafe_m, h, torch.ones_like(h))
    acc = ((acc_mask < 0).float().sum() + 1e-6) / (unsafe_m.float().sum() + 1e-6)
    return loss, acc


def _hinge_safe(h: Tensor, safe_m: Tensor, eps: float) -> Tuple[Tensor, Tensor]:
    h_safe = torch.where(safe_m, h, torch.full_like(h, eps * 2))
    loss = torch.relu(-h_safe + eps).sum() / (safe_m.float().sum() + 1e-6)
    acc_mask = torch.where(safe_m, h, -torch.ones_like(h))
    acc = ((acc_mask > 0).float().sum() + 1e-6) / (safe_m.float().sum() + 1e-6)
    return loss, acc
