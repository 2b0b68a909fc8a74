"""Graph CBF network h(x): GNN base + MLP head + tanh Dense(1).

Mirrors ``/root/reference/gcbfplus/algo/module/cbf.py:12-53`` (dims: msg 128,
msg MLP (256,256), attn MLP (128,128), update MLP (256,256), out 128; head
(256,256) + Dense(1) + tanh).
"""
from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor, nn

from ...nn.gnn import GNN
from ...nn.mlp import MLP, Dense
from ...utils.graph import GraphBatch


class CBFNet(nn.Module):
    def __init__(self, node_dim: int, edge_dim: int, gnn_layers: int = 1):
        super().__init__()
        self.gnn = GNN(node_dim=node_dim, edge_dim=edge_dim, msg_dim=128, out_dim=128,
                       n_layers=gnn_layers)
        self.head = MLP(128, (256, 256), act="relu", act_final=False)
        self.out = Dense(256, 1, act="tanh")

    def forward(self, graph: GraphBatch, edge_feats: Tensor, msg_in=None,
                row_gate=None) -> Tensor:
        """-> h: (B, N, 1) in [-1, 1]. row_gate (B, N) bool: per-agent
        parameter stop-gradient (see ops.fused_linear)."""
        x = self.gnn(graph, edge_feats, msg_in0=msg_in, row_gate=row_gate)
        ag = None if row_gate is None else row_gate.reshape(-1).contiguous()
        return self.out(self.head(x, ag), ag)
