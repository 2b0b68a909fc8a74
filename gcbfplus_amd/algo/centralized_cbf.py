"""CentralizedCBF baseline: one QP over all agents with hand-derived pairwise
CBFs over the k=3 nearest neighbors (reference ``algo/centralized_cbf.py``).

  min ||u - u_ref||^2 + 10||r||^2 + 1e3 r
  s.t. -Lg_h u - r <= Lf_h + alpha h,  u in box,  r >= 0
with h: (N, k) rows; variables (N nu + N k). Batched over envs through the
K11 ProxQP op; jacobians are analytic (algo/pwise.py).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from ..ops.qp import proxqp_solve
from ..utils.graph import GraphBatch
from .base import MultiAgentController
from .pwise import pwise_cbf


class CentralizedCBF(MultiAgentController):
    def __init__(self, env, node_dim, edge_dim, state_dim, action_dim, n_agents,
                 alpha: float = 1.0, **kwargs):
        super().__init__(env, node_dim, edge_dim, action_dim, n_agents)
        self.alpha = alpha
        self.k = 3
        self.qp_iters = 150

    @property
    def config(self) -> dict:
        return {"alpha": self.alpha}

    def step(self, graph, **kw):
        raise NotImplementedError

    def update(self, rollout, step):
        raise NotImplementedError

    def save(self, save_dir, step):
        raise NotImplementedError

    def load(self, load_dir, step):
        raise NotImplementedError

    def get_cbf(self, graph: GraphBatch) -> Tensor:
        return pwise_cbf(self._env, graph, self.k)[0]

    @torch.no_grad()
    def act(self, graph: GraphBatch) -> Tensor:
        return self.get_qp_action(graph)[0]

    def get_qp_action(self, graph: GraphBatch, relax_penalty: float = 1e3
                      ) -> Tuple[Tensor, Tensor]:
        env = self._env
        B, N, k, nu = graph.batch_size, self.n_agents, self.k, self.action_dim
        h, h_x, _ = pwise_cbf(env, graph, k)  # (B,N,k), (B,N,k,N,S)
        f, gdyn = env.control_affine_dyn(graph.agent_states)
        Lf_h = torch.einsum("bikjs,bjs->bik", h_x, f).reshape(B, N * k)
        Lg_h = torch.einsum("bikjs,bjsu->bikju", h_x, gdyn).reshape(B, N * k, N * nu)
        h = h.reshape(B, N * k)

        u_lb, u_ub = env.action_lim()
        dev = graph.device
        u_lb, u_ub = u_lb.to(dev).repeat(N), u_ub.to(dev).repeat(N)
        u_ref = env.u_ref(graph).reshape(B, N * nu)

        nv = N * nu + N * k
        H = torch.eye(nv, device=dev).expand(B, nv, nv).clone()
        H[:, N * nu :, N * nu :] *= 10.0
        gvec = torch.cat([-u_ref, relax_penalty * torch.ones(B, N * k, device=dev)], 1)
        C = -torch.cat([Lg_h, torch.eye(N * k, device=dev).expand(B, N * k, N * k)], 2)
        bvec = Lf_h + self.alpha * h
        l_box = torch.cat([u_lb, torch.zeros(N * k, device=dev)]).expand(B, nv).contiguous()
        u_box = torch.cat([u_ub, torch.full((N * k,), float("inf"), device=dev)]) \
            .expand(B, nv).contiguous()
        x = proxqp_solve(H, gvec, C, bvec, l_box, u_box, iters=self.qp_iters)
        return x[:, : N * nu].reshape(B, N, nu), x[:, N * nu :]
