import os

import numpy as np
import pytest
import torch

from gcbfplus_amd.algo import make_algo
from gcbfplus_amd.algo.utils import horizon_safe_mask
from gcbfplus_amd.env import make_env
from gcbfplus_amd.trainer.utils import collect_rollout, eval_rollout_metrics


def brute_force_safe(unsafe, horizon):
    """Literal transcription of reference gcbf_plus.py:160-174."""
    b, T, N = unsafe.shape
    safe = np.ones((b, T, N), dtype=bool)
    for bb in range(b):
        for n in range(N):
            m = np.ones(T, dtype=bool)
            for i in range(T):
                start = 0 if i < horizon else i - horizon
                if unsafe[bb, i, n]:
                    m[start : i + 1] = False
            m[0] = True
            safe[bb, :, n] = m
    return safe


def test_horizon_safe_mask_matches_reference():
    rng = np.random.default_rng(0)
    unsafe = rng.random((3, 20, 4)) < 0.2
    for horizon in (1, 4, 32):
        ours = horizon_safe_mask(torch.from_numpy(unsafe), horizon).numpy()
        ref = brute_force_safe(unsafe, horizon)
        assert (ours == ref).all(), horizon


@pytest.fixture(scope="module")
def small_setup():
    torch.manual_seed(0)
    env = make_env("DoubleIntegrator", num_agents=4, area_size=2.0, max_step=8, device="cpu")
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=env.num_agents, gnn_layers=1, batch_size=8, buffer_size=16,
                     horizon=4, inner_epoch=2, seed=0)
    return env, algo


def test_gcbf_plus_update_changes_params(small_setup):
    env, algo = small_setup
    rng = np.random.default_rng(0)
    before = [p.detach().clone() for p in algo.cbf.parameters()]
    g = env.reset(2, rng)
    ro = collect_rollout(env, algo.step, g)
    info = algo.update(ro, 0)
    assert np.isfinite(info["loss/total"])
    after = list(algo.cbf.parameters())
    assert any(not torch.equal(a, b) for a, b in zip(after, before))
    # target moved toward online (tau=0.5)
    for pt, po in zip(algo.cbf_tgt.parameters(), algo.cbf.parameters()):
        assert torch.isfinite(pt).all()


def test_checkpoint_roundtrip(tmp_path, small_setup):
    env, algo = small_setup
    rng = np.random.default_rng(1)
    g = env.reset(1, rng)
    a_before = algo.act(g)
    algo.save(str(tmp_path), 7)
    assert os.path.exists(tmp_path / "7" / "actor.pkl")
    assert os.path.exists(tmp_path / "7" / "cbf.pkl")

    torch.manual_seed(123)
    algo2 = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                      state_dim=env.state_dim, action_dim=env.action_dim,
                      n_agents=env.num_agents, gnn_layers=1, batch_size=8, buffer_size=16,
                      horizon=4, seed=99)
    a_mid = algo2.act(g)
    assert not torch.allclose(a_mid, a_before)
    algo2.load(str(tmp_path), 7)
    a_after = algo2.act(g)
    assert torch.allclose(a_after, a_before, atol=1e-6)


def test_act_is_2actor_plus_uref(small_setup):
    env, algo = small_setup
    rng = np.random.default_rng(2)
    g = env.reset(1, rng)
    e = env.edge_feats(g)
    with torch.no_grad():
        expected = 2 * algo.actor(g, e) + env.u_ref(g)
    assert torch.allclose(algo.act(g), expected, atol=1e-6)


def test_eval_metrics(small_setup):
    env, algo = small_setup
    rng = np.random.default_rng(3)
    g = env.reset(2, rng)
    ro = collect_rollout(env, algo.act, g)
    m = eval_rollout_metrics(env, ro)
    for k in ("eval/reward", "eval/cost", "eval/unsafe_frac", "eval/finish"):
        assert np.isfinite(m[k])
    assert 0.0 <= m["eval/finish"] <= 1.0


def test_gcbf_update_runs():
    torch.manual_seed(0)
    env = make_env("SingleIntegrator", num_agents=3, area_size=2.0, max_step=8, device="cpu")
    algo = make_algo("gcbf", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=env.num_agents, gnn_layers=1, batch_size=8, buffer_size=16,
                     inner_epoch=2, seed=0)
    rng = np.random.default_rng(0)
    for step in range(2):
        g = env.reset(2, rng)
        ro = collect_rollout(env, algo.step, g)
        info = algo.update(ro, step)
    assert np.isfinite(info["loss/total"])


def test_online_policy_refinement_runs(small_setup):
    env, algo = small_setup
    rng = np.random.default_rng(4)
    g = env.reset(1, rng)
    algo.online_pol_refine = True
    try:
        a = algo.act(g)
    finally:
        algo.online_pol_refine = False
    assert a.shape == (1, env.num_agents, env.action_dim)
    assert torch.isfinite(a).all()


def test_value_net_and_tanh_normal():
    from gcbfplus_amd.algo.module.value import ValueNet
    from gcbfplus_amd.algo.module.policy import TanhNormalPolicyNet
    from gcbfplus_amd.env import make_env

    torch.manual_seed(0)
    env = make_env("DoubleIntegrator", num_agents=3, area_size=2.0, max_step=4,
                   device="cpu")
    g = env.reset(2, np.random.default_rng(0))
    e = env.edge_feats(g)
    v = ValueNet(env.node_dim, env.edge_dim)(g, e)
    assert v.shape == (2,) and torch.isfinite(v).all()
    pol = TanhNormalPolicyNet(env.node_dim, env.edge_dim, env.action_dim)
    a, logp = pol.sample(g, e)
    assert a.shape == (2, 3, 2) and (a.abs() <= 1).all()
    lp = pol.log_prob(g, e, a)
    assert torch.isfinite(lp).all()
    m = pol.mode(g, e)
    assert m.shape == a.shape


def test_tanh_transformed_distribution():
    """TanhTransformedDistribution (reference distribution.py:10-60):
    log_prob matches the change-of-variables formula inside the threshold,
    returns tail log-mass at the boundary, samples stay in (-1, 1)."""
    import math

    from gcbfplus_amd.algo.module import TanhTransformedDistribution

    torch.manual_seed(3)
    loc = torch.randn(32) * 0.5
    scale = torch.rand(32) * 0.5 + 0.2
    d = TanhTransformedDistribution(loc, scale)
    s = d.sample()
    assert s.abs().max() < 1.0
    v = torch.tanh(torch.randn(32) * 0.5)
    lp = d.log_prob(v)
    base = torch.distributions.Normal(loc, scale)
    ref = base.log_prob(torch.atanh(v)) - torch.log1p(-v * v)
    assert torch.allclose(lp, ref, atol=1e-5)
    # boundary: tail mass, finite
    lp_edge = d.log_prob(torch.ones(32))
    ref_edge = (1 - base.cdf(torch.full((32,), math.atanh(0.999)))).clamp_min(1e-38).log()
    assert torch.allclose(lp_edge, ref_edge, atol=1e-4)
    assert torch.isfinite(d.entropy()).all()
    assert torch.allclose(d.mode(), torch.tanh(loc))


def test_nn_utils_helpers():
    """nn/utils parity helpers (reference nn/utils.py:19-51)."""
    from gcbfplus_amd.nn.utils import (default_nn_init, get_act_from_str, safe_get,
                                       scaled_init, signal_last_enumerate)

    w = torch.empty(64, 32)
    default_nn_init(w)
    lim = (6.0 / (64 + 32)) ** 0.5
    assert w.abs().max() <= lim
    w2 = torch.empty(64, 32)
    scaled_init(default_nn_init, 0.01)(w2)
    assert w2.abs().max() <= lim * 0.01
    assert get_act_from_str("relu") is torch.relu
    out = list(signal_last_enumerate("abc"))
    assert out == [(False, 0, "a"), (False, 1, "b"), (True, 2, "c")]
    arr = torch.arange(12.0).reshape(4, 3)
    got = safe_get(arr, torch.tensor([0, 3, 4, -1]))
    assert torch.allclose(got[:2], arr[[0, 3]])
    assert torch.isnan(got[2]).all() and torch.isnan(got[3]).all()


def test_trainer_parity_helpers():
    """has_any_nan / tree_copy / centered_norm / plot_cbf (reference
    trainer/utils.py:58-177)."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    import numpy as np

    from gcbfplus_amd.algo.module.cbf import CBFNet
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.trainer.utils import (centered_norm, has_any_nan, plot_cbf,
                                            tree_copy)

    assert not has_any_nan({"a": torch.ones(3)})
    assert has_any_nan([torch.tensor([1.0, float("nan")])])
    t = {"x": torch.randn(4)}
    c = tree_copy(t)
    assert torch.equal(c["x"], t["x"]) and c["x"].data_ptr() != t["x"].data_ptr()
    nrm = centered_norm([-1.0, -3.0], [2.0])
    assert nrm.halfrange == 3.0 and nrm.vcenter == 0

    env = make_env("DoubleIntegrator", num_agents=4, area_size=2.0, max_step=4)
    g = env.reset(1, np.random.default_rng(0))
    net = CBFNet(env.node_dim, env.edge_dim, 1)
    cbf = lambda gr: net(gr, env.edge_feats(gr))
    fig = plt.figure()
    plot_cbf(fig, cbf, env, g, agent_id=0, n_mesh=6)
    plt.close(fig)


def test_full_resume_roundtrip(tmp_path, small_setup):
    """save_full/load_full (framework extra over the reference's params-only
    checkpoints): optimizer state, target net, and RNG resume."""
    from gcbfplus_amd.trainer.utils import collect_rollout

    env, algo = small_setup
    rng = np.random.default_rng(5)
    g = env.reset(1, rng)
    ro = collect_rollout(env, algo.step, g)
    algo.update(ro, 0)
    path = str(tmp_path / "resume.pt")
    algo.save_full(path, step=3)

    torch.manual_seed(321)
    algo2 = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                      state_dim=env.state_dim, action_dim=env.action_dim,
                      n_agents=env.num_agents, gnn_layers=1, batch_size=8,
                      buffer_size=16, horizon=4, seed=77)
    step = algo2.load_full(path)
    assert step == 3
    for p1, p2 in zip(algo.cbf.parameters(), algo2.cbf.parameters()):
        assert torch.equal(p1, p2)
    for p1, p2 in zip(algo.cbf_tgt.parameters(), algo2.cbf_tgt.parameters()):
        assert torch.equal(p1, p2)
    # same rng stream continues
    assert algo.rng.integers(1 << 30) == algo2.rng.integers(1 << 30)
    # one more update from the restored state runs and changes params
    algo2.update(ro, 4)


def test_trainer_loop_cpu(tmp_path):
    """Trainer.train() end-to-end on CPU: 2 steps, eval + checkpoint + jsonl
    metrics (reference trainer/trainer.py:76-143)."""
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.trainer.trainer import Trainer

    torch.manual_seed(0)
    env = make_env("SingleIntegrator", num_agents=3, area_size=2.0, max_step=8)
    env_test = make_env("SingleIntegrator", num_agents=3, area_size=2.0, max_step=8)
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=3, gnn_layers=1, batch_size=8, buffer_size=16,
                     horizon=4, inner_epoch=1, seed=0)
    tr = Trainer(env, env_test, algo, n_env_train=2, n_env_test=2,
                 log_dir=str(tmp_path), seed=0,
                 params={"run_name": "t", "training_steps": 2, "eval_interval": 1,
                         "eval_epi": 1, "save_interval": 2})
    tr.train()
    import glob

    assert glob.glob(str(tmp_path / "**" / "*.jsonl"), recursive=True) or \
        glob.glob(str(tmp_path / "*.jsonl"))


def test_two_layer_gnn_update_smoke():
    """gnn_layers=2 trains end-to-end (general jacobian path + eager GNN
    layer-1 input build)."""
    from gcbfplus_amd.env import make_env
    from gcbfplus_amd.trainer.utils import collect_rollout

    torch.manual_seed(2)
    env = make_env("DoubleIntegrator", num_agents=3, area_size=2.0, max_step=4)
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=3, gnn_layers=2, batch_size=4, buffer_size=8,
                     horizon=2, inner_epoch=1, seed=0)
    rng = np.random.default_rng(8)
    g = env.reset(1, rng)
    ro = collect_rollout(env, algo.step, g)
    info = algo.update(ro, 0)
    assert all(np.isfinite(v) for v in info.values()), info


def test_bench_contract_json(tmp_path):
    """bench.py emits the driver-contract JSON line on CPU (tiny config)."""
    import json
    import subprocess
    import sys

    env_vars = dict(os.environ, GCBF_BENCH_ENVS="2", GCBF_BENCH_T="16")
    out = subprocess.run(
        [sys.executable, "bench.py", "--cpu", "--steps", "1", "--warmup", "0",
         "--num-agents", "2"],
        capture_output=True, text=True, timeout=900, env=env_vars,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    line = [l for l in out.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    for k in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
              "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
              "dtype", "data", "config"):
        assert k in d, k
    assert d["n_gpus"] == 1 and d["steps"] == 1 and d["scaling"] == "weak"
    assert d["data"] == "synthetic" and d["higher_is_better"] is True


def test_tanh_distribution_normalizes():
    """log_prob of the tanh-squashed Normal integrates to ~1 over [-1, 1]
    (interior density + the two tail point-masses at +-threshold)."""
    import torch
    from gcbfplus_amd.algo.module.distribution import TanhTransformedDistribution

    d = TanhTransformedDistribution(torch.tensor([0.3]), torch.tensor([0.7]))
    t = d.threshold
    xs = torch.linspace(-t + 1e-4, t - 1e-4, 20001)
    dens = d.log_prob(xs).exp()
    interior = torch.trapz(dens, xs)
    lo = d.log_prob(torch.tensor([-1.0])).exp()  # tail mass below -t
    hi = d.log_prob(torch.tensor([1.0])).exp()   # tail mass above +t
    total = float(interior + lo + hi)
    assert abs(total - 1.0) < 2e-3, total


def test_tanh_distribution_entropy_sane():
    import torch
    from gcbfplus_amd.algo.module.distribution import TanhTransformedDistribution

    torch.manual_seed(0)
    d = TanhTransformedDistribution(torch.zeros(4096), torch.full((4096,), 0.5))
    ent = d.entropy().mean()
    # MC estimate of -E[log p(x)] over samples
    s = d.sample()
    mc = -d.log_prob(s).mean()
    assert abs(float(ent - mc)) < 0.05, (float(ent), float(mc))


def test_gae_matches_loop_reference():
    import torch
    from gcbfplus_amd.algo.utils import gae

    torch.manual_seed(1)
    b, T = 3, 7
    r = torch.randn(b, T)
    v = torch.randn(b, T)
    nv = torch.randn(b, T)
    dn = torch.rand(b, T) < 0.2
    adv = gae(0.99, 0.95, r, v, nv, dn)
    # plain transcription of reference algo/utils.py:18-41
    ref = torch.zeros(b, T)
    for i in range(b):
        last = 0.0
        for t in reversed(range(T)):
            nonterm = 1.0 - float(dn[i, t])
            delta = float(r[i, t]) + 0.99 * float(nv[i, t]) * nonterm - float(v[i, t])
            last = delta + 0.99 * 0.95 * nonterm * last
            ref[i, t] = last
    assert torch.allclose(adv, ref, atol=1e-5)
