"""GPU numerics tests: every HIP kernel vs the plain-PyTorch fp32 reference
of the same op (tolerances sized for bf16 inputs / f32 accumulate)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from gcbfplus_amd import ops


def _rand(*shape, scale=1.0):
    return (torch.randn(*shape) * scale).to("cuda")


def ref_linear(x, w, b, act):
    y = x.to(torch.float32) @ w.to(torch.float32) + b
    return ops._apply_act(y, act)


@pytest.mark.parametrize("M,K,N,act", [
    (128, 32, 64, 0),
    (1000, 128, 256, 1),
    (5248, 10, 256, 1),     # first-layer concat shape (K padded inside)
    (2048, 256, 128, 0),
    (517, 384, 256, 1),     # ragged M
    (2048, 256, 1, 2),      # CBF head out (gemv path)
    (2048, 256, 2, 2),      # actor out
    (4096, 128, 1, 0),      # attn gate
])
def test_gemm_bias_act_matches_fp32(M, K, N, act):
    torch.manual_seed(0)
    x = _rand(M, K, scale=0.5)
    w = _rand(K, N, scale=0.2)
    b = _rand(N, scale=0.1).float()
    y = ops.fused_linear(x, w, b, act)
    y_ref = ref_linear(x, w, b, act)
    err = (y.float() - y_ref).abs()
    scale = y_ref.abs().mean().clamp_min(1.0)
    assert (err.mean() / scale) < 5e-3, (err.max().item(), err.mean().item())
    assert (err.max() / scale) < 8e-2


def test_fused_linear_backward_matches_fp32():
    """Backward GEMMs vs an fp32 oracle that uses the SAME forward rounding
    (kernel's y for the relu mask, bf16-rounded dz) — isolates kernel math
    from the inherent bf16 relu-boundary mask flips; plus a loose check vs
    the pure-fp32 chain."""
    torch.manual_seed(1)
    M, K, N = 777, 128, 256
    x = _rand(M, K, scale=0.5).requires_grad_(True)
    w = _rand(K, N, scale=0.2).requires_grad_(True)
    b = _rand(N, scale=0.1).float().requires_grad_(True)
    y = ops.fused_linear(x, w, b, ops.ACT_RELU)
    g = torch.randn_like(y)
    y.backward(g)

    # same-rounding oracle in fp32 math
    x_bf = x.detach().to(torch.bfloat16).float()
    w_bf = w.detach().to(torch.bfloat16).float()
    dz = (g * (y.detach() > 0)).to(torch.bfloat16).float()
    dx_ref = (dz @ w_bf.t()).cpu()
    dw_ref = (x_bf.t() @ dz).cpu()
    db_ref = dz.sum(0).cpu()
    for got, ref in ((x.grad, dx_ref), (w.grad, dw_ref), (b.grad, db_ref)):
        err = (got.float().cpu() - ref).abs()
        denom = ref.abs().mean().clamp_min(1e-3)
        assert (err.mean() / denom) < 5e-3, (err.mean(), denom, err.max())

    # loose end-to-end vs pure fp32 chain (precision-policy check)
    xf = x.detach().float().cpu().requires_grad_(True)
    wf = w.detach().float().cpu().requires_grad_(True)
    bf = b.detach().float().cpu().requires_grad_(True)
    yf = torch.relu(xf @ wf + bf)
    yf.backward(g.float().cpu())
    for got, ref in ((x.grad, xf.grad), (w.grad, wf.grad), (b.grad, bf.grad)):
        got = got.float().cpu()
        denom = ref.abs().mean().clamp_min(1e-3)
        assert ((got - ref).abs().mean() / denom) < 8e-2


def test_gemm_tn_deterministic():
    torch.manual_seed(2)
    x = _rand(4096, 128).to(torch.bfloat16)
    dz = _rand(4096, 256).to(torch.bfloat16)
    from gcbfplus_amd import _C
    dw1, db1 = _C.gemm_tn(x, dz, dz, 0)
    dw2, db2 = _C.gemm_tn(x, dz, dz, 0)
    assert torch.equal(dw1, dw2) and torch.equal(db1, db2)


def test_softmax_aggr_matches_fp32():
    torch.manual_seed(3)
    B, N, D, C = 8, 8, 41, 128
    gate = _rand(B, N, D)
    msg = _rand(B, N, D, C, scale=0.5)
    mask = torch.rand(B, N, D, device="cuda") < 0.5
    mask[:, :, N] = True  # goal slot always on
    out = ops.masked_softmax_aggr(gate, msg.to(torch.bfloat16), mask)
    ref = ops.masked_softmax_aggr(gate.cpu(), msg.cpu(), mask.cpu())
    err = (out.float().cpu() - ref).abs()
    assert err.mean() < 5e-3 and err.max() < 5e-2, (err.mean(), err.max())


def test_softmax_aggr_backward_matches_fp32():
    torch.manual_seed(4)
    B, N, D, C = 4, 6, 20, 128
    gate = _rand(B, N, D).requires_grad_(True)
    msg = _rand(B, N, D, C, scale=0.5).requires_grad_(True)
    mask = torch.rand(B, N, D, device="cuda") < 0.6
    mask[:, :, 0] = True
    out = ops.masked_softmax_aggr(gate, msg, mask)
    g = torch.randn_like(out)
    out.backward(g)

    gate_c = gate.detach().cpu().requires_grad_(True)
    msg_c = msg.detach().cpu().requires_grad_(True)
    ref = ops.masked_softmax_aggr(gate_c, msg_c, mask.cpu())
    ref.backward(g.float().cpu())
    for got, refg in ((gate.grad, gate_c.grad), (msg.grad, msg_c.grad)):
        err = (got.float().cpu() - refg).abs()
        denom = refg.abs().mean().clamp_min(1e-4)
        assert (err.mean() / denom) < 3e-2, (err.mean(), err.max())


def test_softmax_aggr_all_masked_row_zero():
    B, N, D, C = 2, 3, 10, 128
    gate = _rand(B, N, D)
    msg = _rand(B, N, D, C)
    mask = torch.zeros(B, N, D, dtype=torch.bool, device="cuda")
    mask[:, 1:] = True
    out = ops.masked_softmax_aggr(gate, msg, mask)
    assert torch.isfinite(out).all()
    assert out[:, 0].abs().max().item() == 0.0


def test_raytrace_matches_cpu():
    from gcbfplus_amd.env import make_env

    env = make_env("DoubleIntegrator", num_agents=8, area_size=4.0, max_step=4,
                   device="cuda")
    rng = np.random.default_rng(0)
    obs = env.sample_obstacles(3, rng)
    obs_gpu = type(obs)(*[t.cuda() for t in obs])
    pos = torch.rand(3, 8, 2) * 4.0
    hits_gpu = env.get_lidar_hits(pos.cuda(), obs_gpu).cpu()
    from gcbfplus_amd.env.utils import get_lidar

    hits_cpu = get_lidar(pos, obs, env.n_rays, env.params["comm_radius"])
    # in-range hits must agree to fp tolerance; no-hit beams land at huge
    # coords (1e6 scale) where tiny angle diffs blow up absolute error
    close = hits_cpu.abs().amax(dim=-1, keepdim=True) < 100.0
    err = (hits_gpu - hits_cpu).abs() * close
    assert err.max() < 1e-3, err.max()
    assert torch.isfinite(hits_gpu).all()


def test_full_cbf_forward_gpu_vs_cpu():
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.algo.module.cbf import CBFNet

    torch.manual_seed(5)
    env_c = make_env("DoubleIntegrator", num_agents=8, area_size=4.0, max_step=4,
                     device="cpu")
    g = env_c.reset(4, np.random.default_rng(1))
    net = CBFNet(env_c.node_dim, env_c.edge_dim, 1)
    with torch.no_grad():
        h_cpu = net(g, env_c.edge_feats(g))
    g_gpu = g.to("cuda")
    net_gpu = net.to("cuda")
    with torch.no_grad():
        h_gpu = net_gpu(g_gpu, env_c.edge_feats(g_gpu))
    err = (h_gpu.cpu() - h_cpu).abs()
    assert err.max() < 0.05, err.max()  # tanh-bounded output, bf16 path


def test_gpu_update_step_finite():
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.algo import make_algo
    from gcbfplus_amd.trainer.utils import collect_rollout

    torch.manual_seed(6)
    env = make_env("DoubleIntegrator", num_agents=8, area_size=4.0, max_step=8,
                   device="cuda")
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=env.num_agents, gnn_layers=1, batch_size=16,
                     buffer_size=16, horizon=4, inner_epoch=1, seed=0)
    rng = np.random.default_rng(2)
    g = env.reset(2, rng)
    ro = collect_rollout(env, algo.step, g)
    info = algo.update(ro, 0)
    assert all(np.isfinite(v) for v in info.values()), info


def test_proxqp_kernel_matches_torch_oracle():
    """K11 HIP solver vs the torch path (same algorithm) on CBF-QP-shaped
    problems."""
    from gcbfplus_amd.ops.qp import proxqp_solve

    rng = np.random.default_rng(7)
    N, nu = 8, 2
    nv = N * nu + N
    M = 64
    H = np.tile(np.eye(nv, dtype=np.float32), (M, 1, 1))
    H[:, N * nu:, N * nu:] *= 10.0
    u_ref = rng.uniform(-1, 1, size=(M, N * nu)).astype(np.float32)
    g = np.concatenate([-u_ref, 1e3 * np.ones((M, N), np.float32)], axis=1)
    Lg = rng.normal(size=(M, N, N * nu)).astype(np.float32)
    C = -np.concatenate([Lg, np.tile(np.eye(N, dtype=np.float32), (M, 1, 1))], axis=2)
    b = (rng.normal(size=(M, N)) * 0.5).astype(np.float32)
    l = np.concatenate([-np.ones((M, N * nu), np.float32), np.zeros((M, N), np.float32)], 1)
    u = np.concatenate([np.ones((M, N * nu), np.float32),
                        np.full((M, N), np.inf, np.float32)], 1)
    ts = [torch.from_numpy(t) for t in (H, g, C, b, l, u)]
    x_cpu = proxqp_solve(*ts, iters=150)
    x_gpu = proxqp_solve(*[t.cuda() for t in ts], iters=150).cpu()
    # identical algorithm; compare objectives and feasibility rather than
    # iterates (different fp orders)
    def obj(x):
        return 0.5 * torch.einsum("mi,mij,mj->m", x, ts[0], x) + (ts[1] * x).sum(1)
    og, oc = obj(x_gpu), obj(x_cpu)
    assert (og - oc).abs().max() < 1e-2, (og - oc).abs().max()
    viol = torch.einsum("mkn,mn->mk", ts[2], x_gpu) - ts[3]
    assert viol.max() < 1e-3
    assert (x_gpu >= ts[4] - 1e-4).all()


def test_fused_adamw_matches_torch():
    from gcbfplus_amd.ops.optim import FusedAdamW

    torch.manual_seed(7)
    net1 = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Linear(32, 4)).cuda()
    net2 = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Linear(32, 4)).cuda()
    net2.load_state_dict(net1.state_dict())
    opt1 = FusedAdamW(net1, lr=1e-2, weight_decay=1e-3, max_grad_norm=2.0)
    opt2 = torch.optim.AdamW(net2.parameters(), lr=1e-2, weight_decay=1e-3)
    x = torch.randn(64, 16, device="cuda")
    for _ in range(5):
        opt1.zero_grad()
        net1(x).square().mean().backward()
        opt1.step()

        opt2.zero_grad()
        loss = net2(x).square().mean()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(net2.parameters(), 2.0)
        opt2.step()
    for p1, p2 in zip(net1.parameters(), net2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), (p1 - p2).abs().max()


def test_fused_adamw_skips_nonfinite():
    from gcbfplus_amd.ops.optim import FusedAdamW

    net = torch.nn.Linear(8, 8).cuda()
    opt = FusedAdamW(net, lr=1e-2, max_grad_norm=2.0)
    before = opt.pflat.clone()
    opt.zero_grad()
    opt.gflat.fill_(float("nan"))
    opt.step()
    assert torch.equal(opt.pflat, before)
    assert opt.t.item() == 0


def test_graphed_update_matches_eager():
    """HIP-graph minibatch path must produce the same parameters as the
    eager path (identical kernel sequence)."""
    import os
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.algo import make_algo
    from gcbfplus_amd.trainer.utils import collect_rollout

    def run(no_graph: bool):
        if no_graph:
            os.environ["GCBF_NO_HIPGRAPH"] = "1"
        else:
            os.environ.pop("GCBF_NO_HIPGRAPH", None)
        try:
            torch.manual_seed(11)
            env = make_env("DoubleIntegrator", num_agents=4, area_size=2.0,
                           max_step=8, device="cuda")
            algo = make_algo("gcbf+", env=env, node_dim=env.node_dim,
                             edge_dim=env.edge_dim, state_dim=env.state_dim,
                             action_dim=env.action_dim, n_agents=4, gnn_layers=1,
                             batch_size=16, buffer_size=16, horizon=4,
                             inner_epoch=2, seed=3)
            rng = np.random.default_rng(5)
            g = env.reset(2, rng)
            ro = collect_rollout(env, algo.step, g)
            algo.update(ro, 0)
            return algo.cbf_optim.pflat.clone(), algo.actor_optim.pflat.clone()
        finally:
            os.environ.pop("GCBF_NO_HIPGRAPH", None)

    c1, a1 = run(no_graph=True)
    c2, a2 = run(no_graph=False)
    assert torch.allclose(c1, c2, atol=1e-6), (c1 - c2).abs().max()
    assert torch.allclose(a1, a2, atol=1e-6), (a1 - a2).abs().max()


def test_graphed_rollout_matches_eager():
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.algo import make_algo
    from gcbfplus_amd.trainer.graphing import GraphedRolloutStep
    from gcbfplus_amd.trainer.utils import collect_rollout

    torch.manual_seed(12)
    env = make_env("DoubleIntegrator", num_agents=4, area_size=2.0, max_step=8,
                   device="cuda")
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim, n_agents=4,
                     gnn_layers=1, batch_size=16, buffer_size=16, horizon=4, seed=3)
    rng = np.random.default_rng(6)
    g = env.reset(2, rng)
    ro_eager = collect_rollout(env, algo.step, g)
    graphed = GraphedRolloutStep(env, algo.step)
    ro_graph = collect_rollout(env, algo.step, g, graphed)
    assert torch.allclose(ro_eager.states, ro_graph.states, atol=1e-5)
    assert torch.allclose(ro_eager.rewards, ro_graph.rewards, atol=1e-4)
    assert torch.equal(ro_eager.masks, ro_graph.masks)
    # second rollout with fresh worlds through the SAME captured graph
    g2 = env.reset(2, rng)
    ro_eager2 = collect_rollout(env, algo.step, g2)
    ro_graph2 = collect_rollout(env, algo.step, g2, graphed)
    assert torch.allclose(ro_eager2.states, ro_graph2.states, atol=1e-5)


def test_edge_msg_in_matches_cpu_compose():
    from gcbfplus_amd.env import make_env

    torch.manual_seed(13)
    env = make_env("DoubleIntegrator", num_agents=6, area_size=3.0, max_step=4,
                   device="cpu")
    g = env.reset(3, np.random.default_rng(7))
    mi_cpu = ops.edge_msg_in(g.states, 6, env.n_rays, 2, env.params["comm_radius"])
    mi_gpu = ops.edge_msg_in(g.states.cuda(), 6, env.n_rays, 2,
                             env.params["comm_radius"]).float().cpu()
    err = (mi_gpu - mi_cpu).abs()
    assert err.max() < 5e-3, err.max()  # bf16 storage of f32 values

    # backward: fused vjp vs autograd through the CPU compose
    st = g.states.cuda().requires_grad_(True)
    mi = ops.edge_msg_in(st, 6, env.n_rays, 2, env.params["comm_radius"])
    gout = torch.randn_like(mi.float())
    mi.backward(gout.to(torch.bfloat16))
    st_c = g.states.clone().requires_grad_(True)
    mi_c = ops.edge_msg_in(st_c, 6, env.n_rays, 2, env.params["comm_radius"])
    mi_c.backward(gout.to(torch.bfloat16).float().cpu())
    err = (st.grad.cpu() - st_c.grad).abs()
    denom = st_c.grad.abs().mean().clamp_min(1e-4)
    assert (err.mean() / denom) < 2e-2, (err.mean(), err.max(), denom)


def test_fused_loss_matches_eager_math():
    """K10 fused loss vs a CPU fp32 transcription of gcbf_plus.py:364-431,
    values AND gradients."""
    torch.manual_seed(21)
    B, N, nu = 64, 8, 2
    n = B * N
    dt, alpha, eps = 0.03, 1.0, 0.02
    coefs = (1e-4, 1.0, 1.0, 0.01)
    h = (torch.randn(n, device="cuda") * 0.1).requires_grad_(True)
    h_next = (h.detach() + torch.randn(n, device="cuda") * 0.01).requires_grad_(True)
    h_ng = h_next.detach().clone().requires_grad_(True)
    action = torch.randn(n, nu, device="cuda").requires_grad_(True)
    u_qp = torch.randn(n, nu, device="cuda")
    safe = torch.rand(n, device="cuda") < 0.5
    unsafe = ~safe & (torch.rand(n, device="cuda") < 0.3)

    total, parts = ops.gcbf_plus_loss(h, h_next, h_ng, action, u_qp, safe, unsafe,
                                      dt, alpha, eps, *coefs)
    total.backward()

    def ref(hc, hnc, hgc, ac):
        h_dot = (hnc - hc) / dt
        h_dot_ng = (hgc - hc.detach()) / dt
        h_unsafe = torch.where(unsafe.cpu(), hc, torch.full_like(hc, -2 * eps))
        loss_unsafe = torch.relu(h_unsafe + eps).sum() / (unsafe.float().sum().cpu() + 1e-6)
        h_safe = torch.where(safe.cpu(), hc, torch.full_like(hc, 2 * eps))
        loss_safe = torch.relu(-h_safe + eps).sum() / (safe.float().sum().cpu() + 1e-6)
        labeled = (safe | unsafe).cpu()
        val = torch.relu(-h_dot - alpha * hc + eps)
        val_ng = torch.relu(-h_dot_ng - alpha * hc + eps)
        loss_h_dot = torch.where(labeled, val, val_ng).mean()
        loss_action = (ac - u_qp.cpu()).square().sum(-1).mean()
        return (coefs[0] * loss_action + coefs[1] * loss_unsafe
                + coefs[2] * loss_safe + coefs[3] * loss_h_dot)

    hc = h.detach().cpu().requires_grad_(True)
    hnc = h_next.detach().cpu().requires_grad_(True)
    hgc = h_ng.detach().cpu().requires_grad_(True)
    ac = action.detach().cpu().requires_grad_(True)
    t_ref = ref(hc, hnc, hgc, ac)
    t_ref.backward()

    assert abs(total.item() - t_ref.item()) < 1e-4, (total.item(), t_ref.item())
    for got, want in ((h.grad, hc.grad), (h_next.grad, hnc.grad),
                      (h_ng.grad, hgc.grad), (action.grad, ac.grad)):
        assert torch.allclose(got.cpu(), want, atol=1e-6), (got.cpu() - want).abs().max()


def test_fused_env_step_matches_torch():
    """K5-K8 fused DI step vs the torch-composed path (same math)."""
    from gcbfplus_amd.env import make_env

    env = make_env("DoubleIntegrator", num_agents=8, area_size=4.0, max_step=8,
                   device="cuda")
    rng = np.random.default_rng(9)
    g = env.reset(4, rng)
    torch.manual_seed(9)
    a = torch.randn(4, 8, 2, device="cuda")
    res_fused = env._step_fused(g, a)
    # torch path: force by calling the generic pieces directly
    action = env.clip_action(a)
    next_agent = env.agent_step_euler(g.agent_states, action)
    reward = -((action - env.u_ref(g)).square().sum(-1)).mean(-1)
    cost = env.get_cost(g)
    ref_graph = env.get_graph(next_agent, g.goal_states, g.env_states)

    assert torch.allclose(res_fused.reward, reward, atol=1e-5)
    assert torch.allclose(res_fused.cost, cost, atol=1e-5)
    s_err = (res_fused.graph.states[:, :16] - ref_graph.states[:, :16]).abs().max()
    assert s_err < 1e-5, s_err  # agents + goals
    hits_f = res_fused.graph.states[:, 16:]
    hits_r = ref_graph.states[:, 16:]
    close = hits_r.abs().amax(-1, keepdim=True) < 100.0
    assert ((hits_f - hits_r).abs() * close).max() < 1e-3
    mism = (res_fused.graph.mask != ref_graph.mask).float().mean()
    assert mism < 1e-3, mism  # boundary ties only


def test_edge_msg_in_dubins_mode():
    """Transform-mode (DubinsCar) fused edge input vs CPU compose + autograd."""
    from gcbfplus_amd.env import make_env

    torch.manual_seed(15)
    env = make_env("DubinsCar", num_agents=4, area_size=2.0, max_step=4, device="cpu")
    g = env.reset(2, np.random.default_rng(15))
    st = g.states.clone()
    st[:, :4, 3] += 0.3  # nonzero speeds exercise the transform
    mi_cpu = ops.edge_msg_in(st, 4, env.n_rays, 2, env.params["comm_radius"], mode=1)
    mi_gpu = ops.edge_msg_in(st.cuda(), 4, env.n_rays, 2, env.params["comm_radius"],
                             mode=1).float().cpu()
    assert (mi_gpu - mi_cpu).abs().max() < 5e-3

    stg = st.cuda().requires_grad_(True)
    mi = ops.edge_msg_in(stg, 4, env.n_rays, 2, env.params["comm_radius"], mode=1)
    gout = torch.randn_like(mi.float()).to(torch.bfloat16)
    mi.backward(gout)
    stc = st.clone().requires_grad_(True)
    mi_c = ops.edge_msg_in(stc, 4, env.n_rays, 2, env.params["comm_radius"], mode=1)
    mi_c.backward(gout.float().cpu())
    # only agent rows feed autograd in the training loss
    ga = stg.grad[:, :4].cpu()
    gc = stc.grad[:, :4]
    denom = gc.abs().mean().clamp_min(1e-4)
    assert ((ga - gc).abs().mean() / denom) < 3e-2, (ga - gc).abs().max()


def test_di_loss_prep_matches_compose():
    """K16 fused loss prologue vs the composed-torch path, values + grads."""
    from gcbfplus_amd.env import make_env

    env = make_env("DoubleIntegrator", num_agents=8, area_size=4.0, max_step=8,
                   device="cuda")
    rng = np.random.default_rng(31)
    g = env.reset(6, rng)
    torch.manual_seed(31)
    raw = (torch.randn(6, 8, 2, device="cuda") * 0.6).requires_grad_(True)
    K = env._K.to("cuda")
    p = env._params
    act_f, big_f = ops.di_loss_prep(g.states, raw, K, 8, env._dt,
                                    1.0 / p["m"], p["comm_radius"], 0.5)
    da = torch.randn_like(act_f)
    db = torch.randn_like(big_f)
    (act_f * da).sum().backward(retain_graph=True)
    g1 = raw.grad.clone()
    raw.grad = None
    (big_f * db).sum().backward()
    g2 = raw.grad.clone()

    # CPU compose path (forced by moving to CPU)
    act_c, big_c = ops.di_loss_prep(g.states.cpu(), raw.detach().cpu(), K.cpu(), 8,
                                    env._dt, 1.0 / p["m"], p["comm_radius"], 0.5)
    assert torch.allclose(act_f.cpu(), act_c, atol=1e-5)
    assert torch.allclose(big_f.cpu(), big_c, atol=1e-5)
    # grads: compose backward on CPU against the HIP grads
    raw3 = raw.detach().cpu().requires_grad_(True)
    a3, b3 = ops.di_loss_prep(g.states.cpu(), raw3, K.cpu(), 8, env._dt,
                              1.0 / p["m"], p["comm_radius"], 0.5)
    (a3 * da.cpu()).sum().backward()
    assert torch.allclose(g1.cpu(), raw3.grad, atol=1e-5)
    raw4 = raw.detach().cpu().requires_grad_(True)
    a4, b4 = ops.di_loss_prep(g.states.cpu(), raw4, K.cpu(), 8, env._dt,
                              1.0 / p["m"], p["comm_radius"], 0.5)
    (b4 * db.cpu()).sum().backward()
    assert torch.allclose(g2.cpu(), raw4.grad, atol=1e-5)


def test_fused_linear_direct_grad_accumulate():
    """When params carry live .grad buffers (FusedAdamW flat views), backward
    writes += into them and returns None; results must match the standard
    AccumulateGrad path bitwise."""
    torch.manual_seed(40)
    M, K, N = 512, 64, 128
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(K, N, device="cuda", requires_grad=True)
    b = torch.randn(N, device="cuda", requires_grad=True)
    y = ops.fused_linear(x, w, b, ops.ACT_RELU)
    g = torch.randn_like(y)
    (y * g).sum().backward()
    dw_ref, db_ref = w.grad.clone(), b.grad.clone()

    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    w2.grad = torch.zeros(K, N, device="cuda")  # pre-existing buffer: += path
    b2.grad = torch.zeros(N, device="cuda")
    y2 = ops.fused_linear(x, w2, b2, ops.ACT_RELU)
    (y2 * g).sum().backward()
    assert torch.equal(w2.grad, dw_ref)
    assert torch.equal(b2.grad, db_ref)
    # accumulation onto non-zero content (fp add, not overwrite)
    y3 = ops.fused_linear(x, w2, b2, ops.ACT_RELU)
    (y3 * g).sum().backward()
    assert torch.allclose(w2.grad, 2 * dw_ref, rtol=1e-6, atol=1e-6)
    assert torch.allclose(b2.grad, 2 * db_ref, rtol=1e-6, atol=1e-5)


def test_bf16_weight_shadow_in_sync():
    """FusedAdamW's bf16 shadow (p._bf) must track the f32 master through
    optimizer steps (kernel-updated) and checkpoint loads (refresh)."""
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.algo import make_algo
    from gcbfplus_amd.trainer.utils import collect_rollout

    torch.manual_seed(7)
    env = make_env("DoubleIntegrator", num_agents=8, area_size=4.0, max_step=8,
                   device="cuda")
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=env.num_agents, gnn_layers=1, batch_size=16,
                     buffer_size=16, horizon=4, inner_epoch=1, seed=0)
    rng = np.random.default_rng(3)
    g = env.reset(2, rng)
    ro = collect_rollout(env, algo.step, g)
    algo.update(ro, 0)
    for net in (algo.cbf, algo.actor):
        for p in net.parameters():
            assert hasattr(p, "_bf")
            assert torch.equal(p._bf, p.detach().to(torch.bfloat16)), p.shape
    for p in algo.cbf_tgt.parameters():
        assert not hasattr(p, "_bf")


@pytest.mark.parametrize("env_name,n,area", [
    ("DubinsCar", 4, 2.0),
    ("LinearDrone", 4, 2.0),
    ("CrazyFlie", 4, 2.0),
    ("SingleIntegrator", 4, 2.0),
])
def test_gpu_update_step_all_envs(env_name, n, area):
    """Every dynamics family trains one finite GCBF+ update on the GPU path
    (edge kernels, GEMMs, QP labels, fused optimizer)."""
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.algo import make_algo
    from gcbfplus_amd.trainer.utils import collect_rollout

    torch.manual_seed(11)
    env = make_env(env_name, num_agents=n, area_size=area, max_step=8,
                   device="cuda")
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=env.num_agents, gnn_layers=1, batch_size=16,
                     buffer_size=16, horizon=4, inner_epoch=1, seed=0)
    rng = np.random.default_rng(13)
    g = env.reset(2, rng)
    ro = collect_rollout(env, algo.step, g)
    info = algo.update(ro, 0)
    assert all(np.isfinite(v) for v in info.values()), (env_name, info)


def test_mb_gather_matches_index_select():
    """K18 fused 5-tensor gather vs torch.index_select."""
    from gcbfplus_amd import _C

    torch.manual_seed(50)
    Nb, V, S, N, D, nu, mb = 37, 272, 4, 8, 41, 2, 16
    states = torch.randn(Nb, V, S, device="cuda")
    masks = torch.rand(Nb, N, D, device="cuda") < 0.5
    safe = torch.rand(Nb, N, device="cuda") < 0.5
    unsafe = torch.rand(Nb, N, device="cuda") < 0.3
    u_qp = torch.randn(Nb, N, nu, device="cuda")
    idx = torch.randperm(Nb, device="cuda")[:mb]
    outs = (torch.empty(mb, V, S, device="cuda"),
            torch.empty(mb, N, D, dtype=torch.bool, device="cuda"),
            torch.empty(mb, N, dtype=torch.bool, device="cuda"),
            torch.empty(mb, N, dtype=torch.bool, device="cuda"),
            torch.empty(mb, N, nu, device="cuda"))
    _C.mb_gather(states, masks, safe, unsafe, u_qp, idx, *outs)
    for out, src in zip(outs, (states, masks, safe, unsafe, u_qp)):
        assert torch.equal(out, src[idx])


def test_fused_linear_onehot_matches_composed():
    """One-hot fold with direct grad accumulation vs the composed slice path
    (values and parameter grads)."""
    torch.manual_seed(80)
    M, C, N = 512, 128, 256
    x = torch.randn(M, C, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(3 + C, N, device="cuda", requires_grad=True)
    b = torch.randn(N, device="cuda", requires_grad=True)
    w.grad = torch.zeros_like(w)
    b.grad = torch.zeros_like(b)
    y = ops.fused_linear_onehot(x, w, b, ops.ACT_RELU)
    g = torch.randn_like(y)
    (y * g).sum().backward()
    dw1, db1 = w.grad.clone(), b.grad.clone()

    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    y2 = ops.fused_linear(x, w2[3:], b2 + w2[2], ops.ACT_RELU)
    assert torch.equal(y, y2)
    (y2 * g).sum().backward()
    assert torch.allclose(db1, b2.grad, atol=1e-5), (db1 - b2.grad).abs().max()
    assert torch.allclose(dw1, w2.grad, atol=1e-5), (dw1 - w2.grad).abs().max()
    assert dw1[:2].abs().max().item() == 0.0


def test_gated_ng_matches_functional_call():
    """The row-gated single-forward h_ng path must produce the same params
    after one update as the two-forward functional_call path (same data,
    same seeds) — values are identical by construction; this checks the
    gradient routing (GCBF+ stop-grad split, reference gcbf_plus.py:398-408)."""
    import os
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.algo import make_algo
    from gcbfplus_amd.trainer.utils import collect_rollout

    def run(gated: bool):
        if not gated:
            os.environ["GCBF_NO_GATED_NG"] = "1"
        os.environ["GCBF_NO_HIPGRAPH"] = "1"  # compare the eager paths
        try:
            torch.manual_seed(100)
            env = make_env("DoubleIntegrator", num_agents=8, area_size=4.0,
                           max_step=8, device="cuda")
            algo = make_algo("gcbf+", env=env, node_dim=env.node_dim,
                             edge_dim=env.edge_dim, state_dim=env.state_dim,
                             action_dim=env.action_dim, n_agents=8, gnn_layers=1,
                             batch_size=16, buffer_size=16, horizon=4,
                             inner_epoch=1, seed=0)
            rng = np.random.default_rng(101)
            g = env.reset(2, rng)
            ro = collect_rollout(env, algo.step, g)
            algo.update(ro, 0)
            return algo.cbf_optim.pflat.clone(), algo.actor_optim.pflat.clone()
        finally:
            os.environ.pop("GCBF_NO_GATED_NG", None)
            os.environ.pop("GCBF_NO_HIPGRAPH", None)

    c1, a1 = run(gated=False)
    c2, a2 = run(gated=True)
    # The two-forward path computes h_ng in a SEPARATE B-sized GEMM pass
    # whose bf16 tiling differs from the batched 2B pass, so hinge
    # boundaries can flip on a few rows — param diffs up to ~lr-scale on a
    # small fraction of entries are the expected discrepancy (the gated
    # path is the more exact one: h_ng == h_next holds identically there).
    for got, ref in ((c2, c1), (a2, a1)):
        d = (got - ref).abs()
        assert d.max() < 5e-4, d.max()


def test_gemm_tn_rowgate_matches_masked():
    """rowgate semantics: dW/db over gated rows only; both kernel variants."""
    from gcbfplus_amd import _C

    torch.manual_seed(110)
    for M in (4096, 20480):  # kernel1 and kernel3 dispatch
        x = (torch.randn(M, 64, device="cuda") * 0.5).to(torch.bfloat16)
        dz = (torch.randn(M, 128, device="cuda") * 0.5).to(torch.bfloat16)
        gate = torch.rand(M, device="cuda") < 0.5
        dw, db = _C.gemm_tn(x, dz, dz, 0, gate)
        dzm = dz * gate[:, None]
        dw_ref, db_ref = _C.gemm_tn(x, dzm.to(torch.bfloat16), dzm, 0)
        assert torch.allclose(dw, dw_ref, atol=1e-3), (M, (dw - dw_ref).abs().max())
        assert torch.allclose(db, db_ref, atol=1e-3), (M, (db - db_ref).abs().max())
