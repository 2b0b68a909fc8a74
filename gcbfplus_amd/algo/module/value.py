"""Centralized value function: GNN + attention pooling over agents -> scalar
(reference ``gcbfplus/algo/module/value.py:15-76``). Shipped for parity with
the reference's PPO surface; no shipped algorithm trains it.
"""
from __future__ import annotations

import torch
from torch import Tensor, nn

from ...nn.gnn import GNN
from ...nn.mlp import MLP, Dense
from ...utils.graph import GraphBatch


class ValueNet(nn.Module):
    def __init__(self, node_dim: int, edge_dim: int, gnn_layers: int = 1):
        super().__init__()
        self.gnn = GNN(node_dim=node_dim, edge_dim=edge_dim, msg_dim=64, out_dim=64,
                       n_layers=gnn_layers,
                       hid_msg=(128, 128), hid_aggr=(128, 128), hid_update=(128, 128))
        self.attn_mlp = MLP(64, (128, 128), act="relu", act_final=False)
        self.attn_out = Dense(128, 1)
        self.head = MLP(64, (128, 128), act="relu", act_final=False)
        self.out = Dense(128, 1)

    def forward(self, graph: GraphBatch, edge_feats: Tensor, msg_in=None) -> Tensor:
        """-> V: (B,) pooled over agents by attention (value.py:25-35)."""
        x = self.gnn(graph, edge_feats, msg_in0=msg_in)  # (B, N, 64)
        gate = self.attn_out(self.attn_mlp(x)).squeeze(-1)  # (B, N)
        attn = torch.softmax(gate, dim=-1)
        pooled = (attn[..., None] * x).sum(dim=1)  # (B, 64)
        return self.out(self.head(pooled)).squeeze(-1)
