"""Driver benchmark contract (BASELINE.json): GCBF+ training throughput on
DoubleIntegrator n=8, area 4, horizon 32, 16 envs/GPU (weak scaling), the
reference recipe hyper-parameters (settings.yaml DoubleIntegrator row +
train.py fixed constants).

One "step" = one full outer training step exactly as the reference defines it
(trainer/trainer.py:133-139): collect 16 rollouts x 256 env steps, compute QP
labels, 8 inner epochs of minibatch SGD on CBF+actor, polyak target update.

value = environment transitions processed per second across the WHOLE JOB
(n_env * 256 * steps / elapsed summed over ranks); config.train_steps_per_sec
carries the per-step rate for BASELINE.md's table.
"""
import argparse
import json
import os
import time

import numpy as np
import torch

from gcbfplus_amd.algo import make_algo
from gcbfplus_amd.env import make_env
from gcbfplus_amd.parallel import dp
from gcbfplus_amd.trainer.utils import collect_rollout

# benchmark constants (BASELINE.json config) — env overrides exist ONLY for
# fast CI smoke of the JSON contract, never for reported numbers
N_ENV_PER_GPU = int(os.environ.get("GCBF_BENCH_ENVS", 16))
T_HORIZON = int(os.environ.get("GCBF_BENCH_T", 256))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--num-agents", type=int, default=8)
    ap.add_argument("--cpu", action="store_true")
    ap.add_argument("--torch-profile", type=str, default=None, metavar="TRACE.json",
                    help="wrap the timed steps in torch.profiler and export a "
                         "chrome trace (SURVEY.md §5.1 observability)")
    args = ap.parse_args()

    local_rank = dp.setup_from_env()
    world = dp.world_size()
    use_cuda = torch.cuda.is_available() and not args.cpu
    device = f"cuda:{local_rank}" if use_cuda else "cpu"
    if use_cuda:
        torch.cuda.set_device(local_rank)

    n = args.num_agents
    env = make_env("DoubleIntegrator", num_agents=n, area_size=4.0, max_step=T_HORIZON,
                   device=device)
    algo = make_algo(
        "gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
        state_dim=env.state_dim, action_dim=env.action_dim, n_agents=n,
        gnn_layers=1, batch_size=256, buffer_size=512, horizon=32,
        lr_actor=1e-5, lr_cbf=1e-5, alpha=1.0, eps=0.02, inner_epoch=8,
        loss_action_coef=1e-4, loss_unsafe_coef=1.0, loss_safe_coef=1.0,
        loss_h_dot_coef=0.01, max_grad_norm=2.0, seed=0,
    )
    rng = np.random.default_rng(1234 + 7919 * dp.rank())
    graphed = None
    if use_cuda:
        from gcbfplus_amd.trainer.graphing import GraphedRolloutStep

        graphed = GraphedRolloutStep(env, algo.step)

    def train_step(step):
        graph0 = env.reset(N_ENV_PER_GPU, rng)
        rollout = collect_rollout(env, algo.step, graph0, graphed)
        return algo.update(rollout, step)

    for i in range(args.warmup):
        train_step(i)

    if dp.is_active():
        torch.distributed.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    if args.torch_profile:
        from torch.profiler import ProfilerActivity, profile, record_function

        acts = [ProfilerActivity.CPU] + ([ProfilerActivity.CUDA] if use_cuda else [])
        with profile(activities=acts) as prof:
            for i in range(args.steps):
                with record_function(f"train_step_{i}"):
                    train_step(args.warmup + i)
            if use_cuda:
                torch.cuda.synchronize()
        if dp.rank() == 0:
            prof.export_chrome_trace(args.torch_profile)
    else:
        for i in range(args.steps):
            train_step(args.warmup + i)
    if use_cuda:
        torch.cuda.synchronize()
    if dp.is_active():
        torch.distributed.barrier()
    elapsed = time.perf_counter() - t0

    # max elapsed over ranks
    if dp.is_active():
        t = torch.tensor([elapsed], device=device if use_cuda else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    if dp.rank() == 0:
        steps_per_sec = args.steps / elapsed
        env_steps_per_sec = steps_per_sec * N_ENV_PER_GPU * T_HORIZON * world
        out = {
            "metric": "env_steps_per_sec (GCBF+ full training, collect+QP+8 epochs)",
            "value": env_steps_per_sec,
            "unit": "env_steps/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": 1000.0 * elapsed / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": "GCBF+ GNN-CBF+actor (DoubleIntegrator)",
                "global_batch": 256,
                "n_agents": n,
                "n_env_per_gpu": N_ENV_PER_GPU,
                "rollout_T": T_HORIZON,
                "train_steps_per_sec": steps_per_sec,
                "parallelism": f"dp{world}",
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
