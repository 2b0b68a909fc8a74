"""DecShareCBF baseline: per-agent small QPs with shared responsibility
(reference ``algo/dec_share_cbf.py``): each agent solves over its own action
only, using its k=3 nearest pairwise CBFs with responsibility 1/2 for
agent-agent constraints (1 for obstacles). B*N tiny QPs in one K11 batch.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from ..ops.qp import proxqp_solve
from ..utils.graph import GraphBatch
from .base import MultiAgentController
from .pwise import pwise_cbf


class DecShareCBF(MultiAgentController):
    def __init__(self, env, node_dim, edge_dim, state_dim, action_dim, n_agents,
                 alpha: float = 1.0, **kwargs):
        super().__init__(env, node_dim, edge_dim, action_dim, n_agents)
        if hasattr(env, "enable_stop"):
            env.enable_stop = False  # reference dec_share_cbf.py:34
        self.cbf_alpha = alpha
        self.k = 3
        self.qp_iters = 150

    @property
    def config(self) -> dict:
        return {"alpha": self.cbf_alpha}

    def step(self, graph, **kw):
        raise NotImplementedError

    def update(self, rollout, step):
        raise NotImplementedError

    def save(self, save_dir, step):
        raise NotImplementedError

    def load(self, load_dir, step):
        raise NotImplementedError

    def get_cbf(self, graph: GraphBatch):
        h, _, isobs = pwise_cbf(self._env, graph, self.k)
        return h, isobs

    @torch.no_grad()
    def act(self, graph: GraphBatch) -> Tensor:
        return self.get_qp_action(graph)[0]

    def get_qp_action(self, graph: GraphBatch, relax_penalty: float = 1e3
                      ) -> Tuple[Tensor, Tensor]:
        env = self._env
        B, N, k, nu = graph.batch_size, self.n_agents, self.k, self.action_dim
        h, h_x, isobs = pwise_cbf(env, graph, k)
        f, gdyn = env.control_affine_dyn(graph.agent_states)
        Lf_h = torch.einsum("bikjs,bjs->bik", h_x, f)  # (B,N,k)
        # own-action block only (dec_share_cbf.py:104-107)
        iidx = torch.arange(N, device=graph.device)
        hx_self = h_x[:, iidx, :, iidx, :].permute(1, 0, 2, 3)  # (B,N,k,S)
        Lg_h_self = torch.einsum("biks,bisu->biku", hx_self, gdyn)  # (B,N,k,nu)

        u_ref = env.u_ref(graph)  # (B,N,nu)
        resp = torch.where(isobs, 1.0, 0.5)
        M = B * N
        nv = nu + k
        dev = graph.device
        H = torch.eye(nv, device=dev).expand(M, nv, nv).clone()
        # reference quirk (dec_share_cbf.py:124): the whole k x k block is
        # SET to 10.0 (not just the diagonal) — a rank-1 PSD block
        H[:, nu:, nu:] = 10.0
        gvec = torch.cat(
            [-u_ref.reshape(M, nu), relax_penalty * torch.ones(M, k, device=dev)], 1
        )
        C = -torch.cat(
            [Lg_h_self.reshape(M, k, nu), torch.eye(k, device=dev).expand(M, k, k)], 2
        )
        bvec = (resp * (Lf_h + self.cbf_alpha * h)).reshape(M, k)
        u_lb, u_ub = env.action_lim()
        l_box = torch.cat([u_lb.to(dev), torch.zeros(k, device=dev)]).expand(M, nv).contiguous()
        u_box = torch.cat([u_ub.to(dev), torch.full((k,), float("inf"), device=dev)]) \
            .expand(M, nv).contiguous()
        x = proxqp_solve(H, gvec, C, bvec, l_box, u_box, iters=self.qp_iters)
        return x[:, :nu].reshape(B, N, nu), x[:, nu:].reshape(B, N, k)
