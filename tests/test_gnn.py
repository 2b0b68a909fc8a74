"""GNN correctness: the dense-slot masked-softmax GNN must equal a scatter
(edge-list, jraph-semantics) evaluation with the same weights — this is the
oracle for the reference's segment_softmax/segment_sum formulation
(reference nn/gnn.py:44-104)."""
import numpy as np
import pytest
import torch

from gcbfplus_amd.env import make_env
from gcbfplus_amd.nn.gnn import one_hot_node_feats, sender_index
from gcbfplus_amd.algo.module.cbf import CBFNet


def scatter_reference_forward(net: CBFNet, env, graph):
    """Edge-list evaluation with jraph semantics (per-receiver segment
    softmax), using the same module weights."""
    layer = net.gnn.layers[0]
    B = graph.batch_size
    N, R, V = graph.n_agents, graph.n_rays, graph.n_nodes
    e_dense = env.edge_feats(graph)  # (B,N,D,E)
    send_idx = sender_index(N, R, torch.device("cpu"))
    nf = one_hot_node_feats(B, N, R, torch.device("cpu"))
    outs = []
    for b in range(B):
        act = graph.mask[b]  # (N,D)
        aggr = torch.zeros(V, layer.msg_dim)
        new_agents = []
        for i in range(N):
            slots = act[i].nonzero().flatten()
            feats, msgs, gates = [], [], []
            for d in slots.tolist():
                s = send_idx[i, d].item()
                inp = torch.cat([e_dense[b, i, d], nf[b, s], nf[b, i]])
                m = layer.msg_out(layer.msg_mlp(inp[None]))[0]
                gt = layer.attn_out(layer.attn_mlp(m[None]))[0]
                msgs.append(m)
                gates.append(gt)
            msgs = torch.stack(msgs)
            gates = torch.stack(gates).flatten()
            attn = torch.softmax(gates, dim=0)
            aggr_i = (attn[:, None] * msgs).sum(0)
            upd_in = torch.cat([nf[b, i], aggr_i])
            new_agents.append(layer.update_out(layer.update_mlp(upd_in[None]))[0])
        x = torch.stack(new_agents)  # (N, out)
        h = net.out(net.head(x[None]))[0]
        outs.append(h)
    return torch.stack(outs)


@pytest.fixture(scope="module")
def setup():
    torch.manual_seed(0)
    env = make_env("DoubleIntegrator", num_agents=4, area_size=2.0, max_step=4, device="cpu")
    g = env.reset(2, np.random.default_rng(0))
    net = CBFNet(env.node_dim, env.edge_dim, gnn_layers=1)
    return env, g, net


def test_dense_gnn_equals_scatter_reference(setup):
    env, g, net = setup
    with torch.no_grad():
        h_dense = net(g, env.edge_feats(g))
        h_ref = scatter_reference_forward(net, env, g)
    assert torch.allclose(h_dense, h_ref, atol=1e-5), (h_dense - h_ref).abs().max()


def test_masked_slots_have_zero_gradient(setup):
    env, g, net = setup
    e = env.edge_feats(g).detach().requires_grad_(True)
    h = net(g, e)
    h.sum().backward()
    ge = e.grad
    masked = ~g.mask
    assert ge[masked].abs().max().item() == 0.0
    active_norm = ge[g.mask].abs().sum()
    assert active_norm > 0


def test_isolated_agent_gets_goal_edge_only(setup):
    env, g, net = setup
    # move agent 0 far away: only its goal edge (always on) remains
    st = g.states.clone()
    st[:, 0, :2] += 100.0
    g2 = g.replace(states=st)
    mask2 = env.build_mask(st)
    g2 = g2.replace(mask=mask2)
    n = env.num_agents
    assert mask2[:, 0, :n].sum() == 0 and mask2[:, 0, n].all()
    with torch.no_grad():
        h = net(g2, env.edge_feats(g2))
    assert torch.isfinite(h).all()


def test_two_layer_gnn_runs(setup):
    env, g, _ = setup
    torch.manual_seed(1)
    net2 = CBFNet(env.node_dim, env.edge_dim, gnn_layers=2)
    with torch.no_grad():
        h = net2(g, env.edge_feats(g))
    assert h.shape == (2, 4, 1) and torch.isfinite(h).all()
