import numpy as np
import torch

from gcbfplus_amd.env import make_env
from gcbfplus_amd.utils.graph import GraphBatch


def make_graph(n=4, b=2, seed=0):
    env = make_env("DoubleIntegrator", num_agents=n, area_size=2.0, max_step=4, device="cpu")
    g = env.reset(b, np.random.default_rng(seed))
    return env, g


def test_shapes():
    env, g = make_graph()
    n, r = 4, 32
    assert g.states.shape == (2, 2 * n + n * r, 4)
    assert g.mask.shape == (2, n, n + 1 + r)
    assert g.agent_states.shape == (2, n, 4)
    assert g.goal_states.shape == (2, n, 4)
    assert g.hit_states.shape == (2, n, r, 4)


def test_self_edges_masked_goal_always_on():
    env, g = make_graph()
    n = 4
    diag = g.mask[:, torch.arange(n), torch.arange(n)]
    assert not diag.any(), "self agent-agent slots must be masked"
    assert g.mask[:, :, n].all(), "goal edge always active"


def test_agent_agent_mask_matches_distance():
    env, g = make_graph()
    pos = g.agent_states[..., :2]
    d = torch.cdist(pos, pos)
    comm = env.params["comm_radius"]
    expect = (d < comm) & ~torch.eye(4, dtype=torch.bool)[None]
    assert torch.equal(g.mask[:, :, :4], expect)


def test_indexing_and_cat():
    env, g = make_graph()
    g0 = g[0:1]
    assert g0.batch_size == 1
    g2 = GraphBatch.cat([g0, g[1:2]])
    assert torch.equal(g2.states, g.states)
    assert torch.equal(g2.mask, g.mask)
    assert torch.equal(g2.env_states.center, g.env_states.center)


def test_with_agent_states_differentiable():
    env, g = make_graph()
    a = g.agent_states.clone().requires_grad_(True)
    g2 = g.with_agent_states(a)
    loss = g2.states.square().sum()
    loss.backward()
    assert a.grad is not None and torch.isfinite(a.grad).all()


def test_chunk_vmap_equivalent():
    """chunk_vmap (reference utils.py:96-114): chunked apply == full apply."""
    import torch
    from gcbfplus_amd.utils.utils import chunk_vmap

    x = torch.randn(11, 3)
    f = chunk_vmap(lambda t: t.square().sum(-1), 4)
    assert torch.allclose(f(x), x.square().sum(-1))
