"""Policies: deterministic GNN actor (the one the shipped algorithms use) and
the PPO TanhNormal head (reference ships it unused; kept for parity).

Mirrors ``/root/reference/gcbfplus/algo/module/policy.py``:
  Deterministic (63-73): GNN base + MLP(256,256) head + Dense(nu) + tanh
  TanhNormal (30-60): GNN base + scaled Dense(256) + mean/std heads
"""
from __future__ import annotations

import math
from typing import Tuple

import torch
from torch import Tensor, nn

from ...nn.gnn import GNN
from ...nn.mlp import MLP, Dense
from ...utils.graph import GraphBatch


class DeterministicPolicyNet(nn.Module):
    def __init__(self, node_dim: int, edge_dim: int, action_dim: int, gnn_layers: int = 1):
        super().__init__()
        self.gnn = GNN(node_dim=node_dim, edge_dim=edge_dim, msg_dim=128, out_dim=128,
                       n_layers=gnn_layers)
        self.head = MLP(128, (256, 256), act="relu", act_final=False)
        self.out = Dense(256, action_dim, act="tanh")

    def forward(self, graph: GraphBatch, edge_feats: Tensor, msg_in=None) -> Tensor:
        """-> raw policy output in [-1, 1]^nu, shape (B, N, nu)."""
        x = self.gnn(graph, edge_feats, msg_in0=msg_in)
        return self.out(self.head(x))


class TanhNormalPolicyNet(nn.Module):
    """Stochastic tanh-squashed Gaussian (reference TanhNormal, policy.py:30-60).

    Used by the PPO policy (policy.py:139-176) which no shipped algorithm
    trains; provided for capability parity.
    """

    def __init__(self, node_dim: int, edge_dim: int, action_dim: int, gnn_layers: int = 1,
                 scale_final: float = 0.01, std_dev_init: float = 0.5, std_dev_min: float = 1e-5):
        super().__init__()
        self.gnn = GNN(node_dim=node_dim, edge_dim=edge_dim, msg_dim=64, out_dim=64,
                       n_layers=gnn_layers,
                       hid_msg=(128, 128), hid_aggr=(128, 128), hid_update=(128, 128))
        self.scale_hid = Dense(64, 256, scale=scale_final)
        self.mean_head = Dense(256, action_dim)
        self.std_head = Dense(256, action_dim)
        self.std_dev_min = std_dev_min
        # inverse softplus of std_dev_init (policy.py:37-42)
        self.std_init_inv = math.log(math.exp(std_dev_init) - 1)
        self.action_dim = action_dim

    def dist_params(self, graph: GraphBatch, edge_feats: Tensor) -> Tuple[Tensor, Tensor]:
        x = self.scale_hid(self.gnn(graph, edge_feats))
        mean = self.mean_head(x)
        std = torch.nn.functional.softplus(self.std_head(x) + self.std_init_inv) + self.std_dev_min
        return mean, std

    def mode(self, graph: GraphBatch, edge_feats: Tensor) -> Tensor:
        mean, _ = self.dist_params(graph, edge_feats)
        return torch.tanh(mean)

    def sample(self, graph: GraphBatch, edge_feats: Tensor) -> Tuple[Tensor, Tensor]:
        mean, std = self.dist_params(graph, edge_feats)
        z = mean + std * torch.randn_like(mean)
        action = torch.tanh(z)
        log_pi = self.log_prob_z(mean, std, z).sum(-1)
        return action, log_pi

    def log_prob(self, graph: GraphBatch, edge_feats: Tensor, action: Tensor) -> Tensor:
        mean, std = self.dist_params(graph, edge_feats)
        a = torch.clamp(action, -0.999999, 0.999999)
        z = torch.atanh(a)
        return self.log_prob_z(mean, std, z).sum(-1)

    @staticmethod
    def log_prob_z(mean: Tensor, std: Tensor, z: Tensor) -> Tensor:
        normal = -0.5 * (((z - mean) / std) ** 2 + 2 * torch.log(std) + math.log(2 * math.pi))
        # tanh change of variables: log(1 - tanh(z)^2) = 2*(log2 - z - softplus(-2z))
        corr = 2.0 * (math.log(2.0) - z - torch.nn.functional.softplus(-2.0 * z))
        return normal - corr
