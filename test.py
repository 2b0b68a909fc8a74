"""Evaluation CLI — flag-compatible with the reference test.py:239-264.

Rolls out a trained policy (or a CBF-QP baseline, or plain u_ref) for --epi
episodes and reports safety / finish / success rates (definitions:
reference test.py:184-198), with optional CSV logging and video rendering.
"""
import argparse
import datetime
import os

import numpy as np
import torch
import yaml

from gcbfplus_amd.algo import make_algo
from gcbfplus_amd.env import make_env
from gcbfplus_amd.trainer.utils import collect_rollout


def _load_run_config(f) -> dict:
    """Load a run config.yaml: ours are plain mappings; the reference dumps an
    argparse.Namespace with a python-object tag (loaded there via
    yaml.UnsafeLoader, reference test.py:37-38). Support both without
    arbitrary-object unpickling: only the Namespace tag is whitelisted."""

    class _Loader(yaml.SafeLoader):
        pass

    def _ns(loader, node):
        return loader.construct_mapping(node, deep=True)

    _Loader.add_constructor(
        "tag:yaml.org,2002:python/object:argparse.Namespace", _ns)
    return yaml.load(f, Loader=_Loader)


def test(args):
    print(f"> Running test.py {args}")
    np.random.seed(args.seed)
    torch.manual_seed(args.seed)
    device = "cpu" if args.cpu or not torch.cuda.is_available() else "cuda"

    # load config from the run dir (reference test.py:36-49)
    if args.path is not None:
        with open(os.path.join(args.path, "config.yaml")) as f:
            config = _load_run_config(f)
        env_id = config.get("env") if args.env is None else args.env
        num_agents = config.get("num_agents") if args.num_agents is None else args.num_agents
        area_size = config.get("area_size") if args.area_size is None else args.area_size
        algo_name = config.get("algo", "gcbf+")
        gnn_layers = config.get("gnn_layers", 1)
    else:
        assert args.num_agents is not None, "-n required without --path"
        env_id = args.env or "DoubleIntegrator"
        num_agents = args.num_agents
        area_size = args.area_size
        algo_name = args.algo or "gcbf+"
        gnn_layers = 1

    env = make_env(env_id, num_agents, area_size=area_size, max_step=args.max_step,
                   num_obs=args.obs, n_rays=args.n_rays, max_travel=args.max_travel,
                   device=device)

    if args.u_ref:
        act_fn = env.u_ref
        algo = None
    else:
        algo = make_algo(
            algo=algo_name, env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
            state_dim=env.state_dim, action_dim=env.action_dim, n_agents=num_agents,
            gnn_layers=gnn_layers, alpha=args.alpha,
        )
        if args.path is not None:
            model_path = os.path.join(args.path, "models")
            step = args.step
            if step is None:  # max numeric step dir (reference test.py:55-58)
                steps = [int(d) for d in os.listdir(model_path) if d.isdigit()]
                step = max(steps)
            print(f"Loading model from {model_path}, step {step}")
            algo.load(model_path, step)
        act_fn = algo.act

    rng = np.random.default_rng(args.seed + args.offset)
    safe_rates, finish_rates, success_rates = [], [], []
    rollouts = []
    n_epi = args.epi
    for epi in range(n_epi):
        graph0 = env.reset(1, rng)
        if args.nojit_rollout:
            # streamed path for large swarms (reference env/base.py:191-259):
            # step-by-step metrics, no stored graphs/edges
            g = graph0
            ever_coll = env.collision_mask(g).float()
            ever_fin = env.finish_mask(g).float()
            with torch.no_grad():
                for _ in range(env.max_episode_steps):
                    a = act_fn(g)
                    g = env.step(g, a).graph
                    ever_coll = torch.maximum(ever_coll, env.collision_mask(g).float())
                    ever_fin = torch.maximum(ever_fin, env.finish_mask(g).float())
            a_safe, a_finish = 1.0 - ever_coll, ever_fin
            rollout = None
        else:
            rollout = collect_rollout(env, act_fn, graph0)
            # Tp1 semantics: include the terminal (post-step) state, matching the
            # reference test.py:184-186 and this file's streamed branch above.
            g = rollout.graph_Tp1(env)
            T = rollout.time_horizon + 1
            coll = env.collision_mask(g).reshape(1, T, -1)
            finish = env.finish_mask(g).reshape(1, T, -1)
            a_safe = 1.0 - coll.amax(dim=1).float()  # (1, N)
            a_finish = finish.amax(dim=1).float()
        a_success = a_safe * a_finish
        safe_rates.append(a_safe.mean().item())
        finish_rates.append(a_finish.mean().item())
        success_rates.append(a_success.mean().item())
        print(f"epi {epi}: safe {a_safe.mean():.3f} finish {a_finish.mean():.3f} "
              f"success {a_success.mean():.3f}")
        if rollout is not None:
            rollouts.append(rollout)

    print(
        f"safe rate: {100*np.mean(safe_rates):.3f}%, "
        f"finish rate: {100*np.mean(finish_rates):.3f}%, "
        f"success rate: {100*np.mean(success_rates):.3f}%"
    )

    if args.log:  # CSV append (reference test.py:209-215)
        os.makedirs(args.log, exist_ok=True)
        with open(os.path.join(args.log, "log.csv"), "a") as f:
            f.write(f"{env_id},{num_agents},{args.epi},{np.mean(safe_rates)},"
                    f"{np.mean(finish_rates)},{np.mean(success_rates)}\n")

    if not args.no_video:
        from gcbfplus_amd.env.plot import render_video

        videos_dir = os.path.join(args.path or ".", "videos")
        os.makedirs(videos_dir, exist_ok=True)
        stamp = datetime.datetime.now().strftime("%Y%m%d%H%M%S")
        cbf_fn = None
        if args.cbf is not None and algo is not None and hasattr(algo, "get_cbf"):
            cbf_fn = algo.get_cbf
        for epi, ro in enumerate(rollouts[: args.max_videos]):
            g = ro.graph_at(env)
            unsafe = env.collision_mask(g).reshape(ro.time_horizon, -1)
            out = render_video(ro, os.path.join(videos_dir, f"{stamp}_epi{epi}.gif"),
                               env, b=0, Ta_is_unsafe=unsafe, cbf_fn=cbf_fn,
                               cbf_agent=args.cbf, dpi=args.dpi)
            print("video:", out)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("-n", "--num-agents", type=int, default=None)
    parser.add_argument("--algo", type=str, default=None)
    parser.add_argument("--env", type=str, default=None)
    parser.add_argument("--path", type=str, default=None)
    parser.add_argument("--step", type=int, default=None)
    parser.add_argument("--epi", type=int, default=5)
    parser.add_argument("--seed", type=int, default=1234)
    parser.add_argument("--offset", type=int, default=0)
    parser.add_argument("--obs", type=int, default=None)
    parser.add_argument("--n-rays", type=int, default=32)
    parser.add_argument("--area-size", type=float, default=None)
    parser.add_argument("--max-step", type=int, default=None)
    parser.add_argument("--u-ref", action="store_true", default=False)
    parser.add_argument("--cpu", action="store_true", default=False)
    parser.add_argument("--log", type=str, default=None)
    parser.add_argument("--cbf", type=int, default=None)
    parser.add_argument("--alpha", type=float, default=1.0)
    parser.add_argument("--max-travel", type=float, default=None)
    parser.add_argument("--dpi", type=int, default=100)
    parser.add_argument("--nojit-rollout", action="store_true", default=False,
                        help="streamed eval without storing rollouts (512+ agents)")
    parser.add_argument("--no-video", action="store_true", default=False)
    parser.add_argument("--max-videos", type=int, default=1)
    parser.add_argument("--debug", action="store_true", default=False)
    args = parser.parse_args()
    test(args)


if __name__ == "__main__":
    main()
