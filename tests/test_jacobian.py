"""CBF jacobian: the one-backward edge-grad fast path must match the full
autograd jacobian (reference computes it with jax.jacobian, gcbf_plus.py:317)."""
import numpy as np
import pytest
import torch

from gcbfplus_amd.env import make_env
from gcbfplus_amd.algo import make_algo


@pytest.fixture(scope="module")
def setup():
    torch.manual_seed(0)
    env = make_env("DoubleIntegrator", num_agents=4, area_size=2.0, max_step=4, device="cpu")
    g = env.reset(3, np.random.default_rng(0))
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=env.num_agents, gnn_layers=1, batch_size=8, buffer_size=16,
                     horizon=4, seed=0)
    return env, g, algo


def autograd_jacobian(env, cbf, graph):
    M, N, S = graph.batch_size, graph.n_agents, graph.state_dim
    st = graph.states.detach().clone().requires_grad_(True)
    e = env.edge_feats(graph, st)
    h = cbf(graph, e).squeeze(-1)
    J = torch.zeros(M, N, N, S)
    for i in range(N):
        (gs,) = torch.autograd.grad(h[:, i].sum(), st, retain_graph=i < N - 1)
        J[:, i] = gs[:, :N]
    return J


def test_fast_jacobian_matches_autograd(setup):
    env, g, algo = setup
    h, J_fast = algo.cbf_and_jacobian(g, algo.cbf_tgt)
    J_ref = autograd_jacobian(env, algo.cbf_tgt, g)
    assert torch.allclose(J_fast, J_ref, atol=2e-5), (J_fast - J_ref).abs().max()


def test_jacobian_with_clipped_edges(setup):
    """Push an agent so its goal edge exceeds comm radius -> the pos-clip
    jacobian branch is exercised."""
    env, g, algo = setup
    st = g.states.clone()
    st[:, 0, :2] += 0.7  # goal edge length > comm_radius = 0.5
    g2 = g.replace(states=st, mask=env.build_mask(st))
    h, J_fast = algo.cbf_and_jacobian(g2, algo.cbf_tgt)
    J_ref = autograd_jacobian(env, algo.cbf_tgt, g2)
    assert torch.allclose(J_fast, J_ref, atol=2e-5), (J_fast - J_ref).abs().max()


def test_general_path_matches_fast(setup):
    env, g, algo = setup
    h1, J1 = algo.cbf_and_jacobian(g, algo.cbf_tgt)
    algo.gnn_layers = 2  # force general path (still 1-layer net -> same result)
    try:
        h2, J2 = algo.cbf_and_jacobian(g, algo.cbf_tgt)
    finally:
        algo.gnn_layers = 1
    assert torch.allclose(h1, h2, atol=1e-6)
    assert torch.allclose(J1, J2, atol=2e-5)


def test_crazyflie_analytic_edge_jac_matches_general():
    """CF fast path (one backward + batched 12x12 transform jacobian) vs the
    general N-backward autograd path."""
    from gcbfplus_amd.algo import make_algo
    from gcbfplus_amd.env import make_env

    torch.manual_seed(9)
    env = make_env("CrazyFlie", num_agents=3, area_size=2.0, max_step=4)
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=3, gnn_layers=1, batch_size=4, buffer_size=8,
                     horizon=2, seed=0)
    g = env.reset(2, np.random.default_rng(6))
    # randomize attitudes so the rotation chain is non-trivial
    st = g.states.clone()
    st[:, :3, 3:6] = torch.randn(2, 3, 3) * 0.3  # phi/theta/psi
    st[:, :3, 6:] = torch.randn(2, 3, 6) * 0.2
    g = g.replace(states=st, mask=env.build_mask(st))
    h1, J1 = algo.cbf_and_jacobian(g, algo.cbf_tgt)
    algo.gnn_layers = 2  # force the general path (net is still 1 layer)
    try:
        h2, J2 = algo.cbf_and_jacobian(g, algo.cbf_tgt)
    finally:
        algo.gnn_layers = 1
    assert torch.allclose(h1, h2, atol=1e-6)
    assert torch.allclose(J1, J2, atol=3e-5), (J1 - J2).abs().max()


@pytest.mark.parametrize("env_id,n", [("SingleIntegrator", 4), ("LinearDrone", 3),
                                      ("DubinsCar", 4)])
def test_fast_jacobian_all_envs(env_id, n):
    """Analytic edge→state jacobian chain vs full autograd for every env
    family that advertises analytic_edge_jac."""
    torch.manual_seed(3)
    env = make_env(env_id, num_agents=n, area_size=2.0, max_step=4, device="cpu")
    if not getattr(env, "analytic_edge_jac", True):
        pytest.skip("general path env")
    g = env.reset(2, np.random.default_rng(2))
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=n, gnn_layers=1, batch_size=8, buffer_size=16,
                     horizon=4, seed=0)
    h, J_fast = algo.cbf_and_jacobian(g, algo.cbf_tgt)
    J_ref = autograd_jacobian(env, lambda gr, e: algo.cbf_tgt(gr, e), g)
    assert torch.allclose(J_fast, J_ref, atol=3e-5), (J_fast - J_ref).abs().max()
