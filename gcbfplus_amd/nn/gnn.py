"""Attention message-passing GNN over the dense GraphBatch layout.

Computes exactly the reference GNN (``/root/reference/gcbfplus/nn/gnn.py:44-104``):
  msg   = Dense_msg( MLP_{256,256}( [edge_feat, sender_feat, recv_feat] ) )
  gate  = Dense_1( MLP_{128,128}( msg ) )
  attn  = softmax over incoming edges of each receiver (masked)
  aggr  = sum attn * msg
  node' = Dense_out( MLP_{256,256}( [node_feat, aggr] ) )
but on the dense (B, N, D) slot layout where receivers are always agents and
the softmax is a masked row softmax (see utils/graph.py docstring) — no
segment scatter ops. The final layer updates only agent nodes (the reference
returns type_nodes(0) anyway, gnn.py:100-104).
"""
from __future__ import annotations

import os
from typing import Optional

import torch
from torch import Tensor, nn

from .. import ops
from ..utils.graph import GraphBatch
from .mlp import MLP, Dense


_SENDER_IDX_CACHE = {}
_ONEHOT_CACHE = {}


def sender_index(n_agents: int, n_rays: int, device) -> Tensor:
    """(N, D) node indices of the sender in each edge slot (computable gather
    pattern — slot d of receiver i): agents | own goal | own lidar hits.
    Cached per shape/device (graph-capture friendly, zero per-call kernels)."""
    key = (n_agents, n_rays, str(device))
    hit = _SENDER_IDX_CACHE.get(key)
    if hit is not None:
        return hit
    n, r = n_agents, n_rays
    d = n + 1 + r
    idx = torch.empty(n, d, dtype=torch.long, device=device)
    idx[:, :n] = torch.arange(n, device=device)[None, :]
    idx[:, n] = torch.arange(n, device=device) + n
    base = 2 * n + torch.arange(n, device=device)[:, None] * r
    idx[:, n + 1 :] = base + torch.arange(r, device=device)[None, :]
    _SENDER_IDX_CACHE[key] = idx
    return idx


class GNNLayer(nn.Module):
    def __init__(self, node_dim: int, edge_dim: int, msg_dim: int, out_dim: int,
                 hid_msg=(256, 256), hid_aggr=(128, 128), hid_update=(256, 256)):
        super().__init__()
        in_msg = edge_dim + 2 * node_dim
        pad = (in_msg + 31) // 32 * 32  # fused edge_msg_in emits this K
        self.msg_mlp = MLP(in_msg, hid_msg, act="relu", act_final=False,
                           pad_first_to=pad if pad != in_msg else None)
        self.msg_out = Dense(hid_msg[-1], msg_dim)
        self.attn_mlp = MLP(msg_dim, hid_aggr, act="relu", act_final=False)
        self.attn_out = Dense(hid_aggr[-1], 1)
        self.update_mlp = MLP(node_dim + msg_dim, hid_update, act="relu", act_final=False)
        self.update_out = Dense(hid_update[-1], out_dim)
        self.msg_dim = msg_dim
        self.out_dim = out_dim

    def forward(
        self,
        node_feats: Tensor,  # (B, V, F)
        edge_feats: Tensor,  # (B, N, D, E)
        mask: Tensor,  # (B, N, D)
        send_idx: Tensor,  # (N, D)
        agents_only: bool,
        msg_in: Tensor = None,  # optional fused (B, N, D, K[pad]) input
        onehot_nodes: bool = False,  # node_feats ARE the constant one-hots
        row_gate: Tensor = None,  # (B, N) bool param stop-grad mask
    ) -> Tensor:
        B, V, F = node_feats.shape
        N, D = mask.shape[1], mask.shape[2]
        eg = ag = None  # edge-level / agent-level flat gates
        if row_gate is not None:
            eg = row_gate[:, :, None].expand(B, N, D).reshape(-1).contiguous()
            ag = row_gate.reshape(-1).contiguous()
        if msg_in is None:
            flat_idx = send_idx.reshape(-1)  # (N*D,)
            sender = node_feats[:, flat_idx].reshape(B, N, D, F)
            recv = node_feats[:, :N, None, :].expand(B, N, D, F)
            msg_in = torch.cat([edge_feats, sender, recv], dim=-1)
        msg = self.msg_out(self.msg_mlp(msg_in, eg), eg)  # (B,N,D,msg_dim)
        gate = self.attn_out(self.attn_mlp(msg, eg), eg).squeeze(-1)  # (B,N,D)
        aggr = ops.masked_softmax_aggr(gate, msg, mask)  # (B,N,msg_dim)
        d0 = self.update_mlp.layers[0]
        if agents_only and onehot_nodes and d0.in_dim == 3 + self.msg_dim:
            # agent one-hot is [0,0,1]: cat([onehot, aggr]) @ W ==
            # aggr @ W[3:] + (b + W[2]) — drops the cat/cast AND shrinks the
            # GEMM K from 3+msg_dim (padded) to msg_dim (glds-aligned)
            if aggr.is_cuda and not os.environ.get("GCBF_NO_ONEHOT_FOLD"):
                # direct-grad variant: dW/db accumulate into the param grads
                # inside the dW reduction (no slice-backward kernels)
                h = ops.fused_linear_onehot(aggr, d0.kernel, d0.bias, d0.act,
                                            row_gate=ag)
            elif aggr.is_cuda:
                h = ops.fused_linear(aggr, d0.kernel[3:].contiguous(),
                                     d0.bias + d0.kernel[2], d0.act, ag)
            else:
                h = ops.fused_linear(aggr, d0.kernel[3:], d0.bias + d0.kernel[2],
                                     d0.act)
            for l in self.update_mlp.layers[1:]:
                h = l(h, ag)
            return self.update_out(h, ag)
        assert row_gate is None, \
            "row_gate supports only the fused agents-only single-layer path"
        if agents_only:
            upd_in = torch.cat([node_feats[:, :N], aggr.to(node_feats.dtype)], dim=-1)
        else:
            aggr_full = torch.zeros(B, V, self.msg_dim, dtype=node_feats.dtype,
                                    device=node_feats.device)
            aggr_full = torch.cat([aggr.to(node_feats.dtype), aggr_full[:, N:]], dim=1)
            upd_in = torch.cat([node_feats, aggr_full], dim=-1)
        return self.update_out(self.update_mlp(upd_in))


class GNN(nn.Module):
    """n_layers of GNNLayer; returns per-agent features (B, N, out_dim)."""

    def __init__(self, node_dim: int, edge_dim: int, msg_dim: int = 128, out_dim: int = 128,
                 n_layers: int = 1, hid_msg=(256, 256), hid_aggr=(128, 128),
                 hid_update=(256, 256)):
        super().__init__()
        layers = []
        d_node = node_dim
        for i in range(n_layers):
            od = out_dim if i == n_layers - 1 else msg_dim
            layers.append(GNNLayer(d_node, edge_dim, msg_dim, od, hid_msg, hid_aggr, hid_update))
            d_node = od
        self.layers = nn.ModuleList(layers)
        self.node_dim = node_dim
        self.out_dim = out_dim

    def forward(self, graph: GraphBatch, edge_feats: Optional[Tensor],
                node_feats: Optional[Tensor] = None, msg_in0: Optional[Tensor] = None,
                row_gate: Optional[Tensor] = None) -> Tensor:
        B = graph.batch_size
        N, R, V = graph.n_agents, graph.n_rays, graph.n_nodes
        n_layers = len(self.layers)
        if edge_feats is None:
            assert msg_in0 is not None and n_layers == 1, \
                "edge_feats required unless a fused msg_in0 covers the single layer"
            device = msg_in0.device
        else:
            device = edge_feats.device
        onehot = node_feats is None
        if node_feats is None:
            node_feats = one_hot_node_feats(B, N, R, device, torch.float32)
        send_idx = sender_index(N, R, device)
        x = node_feats
        if row_gate is not None:
            assert n_layers == 1, "row_gate requires gnn_layers == 1"
        for i, layer in enumerate(self.layers):
            last = i == n_layers - 1
            x = layer(x, edge_feats, graph.mask, send_idx, agents_only=last,
                      msg_in=msg_in0 if i == 0 else None,
                      onehot_nodes=onehot and i == 0, row_gate=row_gate)
        return x  # (B, N, out_dim)


def one_hot_node_feats(B: int, N: int, R: int, device, dtype=torch.float32) -> Tensor:
    """Constant node features: agent=001, goal=010, obstacle=100
    (reference env/double_integrator.py:288-295). Cached per shape/device."""
    key = (N, R, str(device), dtype)
    f = _ONEHOT_CACHE.get(key)
    if f is None:
        V = 2 * N + N * R
        f = torch.zeros(V, 3, device=device, dtype=dtype)
        f[:N, 2] = 1.0
        f[N : 2 * N, 1] = 1.0
        f[2 * N :, 0] = 1.0
        _ONEHOT_CACHE[key] = f
    return f[None].expand(B, f.shape[0], 3)
