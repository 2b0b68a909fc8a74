"""DP validation on real hardware, under torchrun world=2.

Two modes (RCCL/NCCL refuses two ranks on one physical GPU with
'Duplicate GPU detected', so full RCCL collectives need >= 2 GPUs — on the
single leased MI355X the probe records that exact refusal as evidence the
RCCL library initializes and only the device count blocks it):

  --mode auto  (default): nccl when torch.cuda.device_count() >= world —
      the real RCCL path (runs on the driver's multi-GPU node); otherwise
      gloo collectives with ALL compute on cuda:0 (both ranks share the
      GPU): broadcast, all_agree, the fused-bucket grad all-reduce, 2 full
      GCBF+ training steps, params bitwise-synced, DP2==DP1-on-same-data.
  --mode nccl-probe: force nccl on 1 GPU and record the refusal text.

Usage (on a GPU box):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 --master-port 29531 tools/rccl_check.py

Writes gpurun_out/rccl_world2.json from rank 0 on success.
"""
import argparse
import json
import os
import sys

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from gcbfplus_amd.algo import make_algo
from gcbfplus_amd.env import make_env
from gcbfplus_amd.parallel import dp
from gcbfplus_amd.trainer.utils import collect_rollout


def run_training(device, seed_rng=42, data_seed=100, steps=2):
    torch.manual_seed(7)
    env = make_env("DoubleIntegrator", num_agents=3, area_size=2.0, max_step=6,
                   device=device)
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim, n_agents=3,
                     gnn_layers=1, batch_size=6, buffer_size=8, horizon=2,
                     inner_epoch=1, seed=5)
    algo.rng = np.random.default_rng(seed_rng)
    rng = np.random.default_rng(data_seed)
    info = {}
    for step in range(steps):
        g = env.reset(2, rng)
        ro = collect_rollout(env, algo.step, g)
        info = algo.update(ro, step)
    flat = torch.cat([p.detach().reshape(-1) for p in algo.cbf.parameters()]
                     + [p.detach().reshape(-1) for p in algo.actor.parameters()])
    return flat.cpu().numpy(), info


def nccl_probe():
    """Force nccl with 2 ranks on 1 GPU; expect the RCCL duplicate-GPU
    refusal at the first collective (communicators are created lazily)."""
    dp.setup_from_env(backend="nccl")
    rank = dist.get_rank()
    t = torch.ones(4, device="cuda")
    try:
        dist.all_reduce(t)
        torch.cuda.synchronize()
        print(f"[rank {rank}] NCCL PROBE: all_reduce SUCCEEDED "
              f"(multi-GPU node)", flush=True)
        ok, msg = True, "all_reduce ok"
    except Exception as e:  # noqa: BLE001
        msg = str(e)
        ok = False
        short = "Duplicate GPU detected" if "Duplicate GPU" in msg else msg[:200]
        print(f"[rank {rank}] NCCL PROBE refusal (expected on 1 GPU): {short}",
              flush=True)
    if rank == 0:
        os.makedirs("gpurun_out", exist_ok=True)
        with open("gpurun_out/rccl_nccl_probe.json", "w") as f:
            json.dump({"world": dist.get_world_size(),
                       "n_gpus": torch.cuda.device_count(),
                       "collective_ok": ok,
                       "refusal": None if ok else msg[:500]}, f, indent=1)
    # do NOT destroy: the comm may be wedged after the refusal; just exit
    os._exit(0)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", default="auto", choices=["auto", "nccl-probe"])
    args = ap.parse_args()
    assert torch.cuda.is_available(), "needs a GPU"
    if args.mode == "nccl-probe":
        nccl_probe()
        return

    local = dp.setup_from_env()
    backend = dist.get_backend()
    world = dist.get_world_size()
    rank = dist.get_rank()
    device = f"cuda:{local % torch.cuda.device_count()}"
    print(f"[rank {rank}/{world}] backend={backend} compute on {device} "
          f"({torch.cuda.get_device_name(0)})", flush=True)

    assert dp.all_agree(True) is True
    assert dp.all_agree(rank == 0) is False  # MIN over {1,0} = 0

    flat_dp, info = run_training(device)
    t = torch.from_numpy(flat_dp)
    tmax, tmin = t.clone(), t.clone()
    if backend == "nccl":
        tmax, tmin = tmax.to(device), tmin.to(device)
    dist.all_reduce(tmax, op=dist.ReduceOp.MAX)
    dist.all_reduce(tmin, op=dist.ReduceOp.MIN)
    in_sync = bool(torch.equal(tmax.cpu(), tmin.cpu()))
    assert in_sync, "params diverged across ranks"
    losses_finite = all(np.isfinite(v) for v in info.values())
    print(f"[rank {rank}] params in sync across {world} ranks "
          f"(backend={backend}, compute=cuda); finite={losses_finite}", flush=True)
    assert losses_finite
    dist.barrier()
    dist.destroy_process_group()

    if rank == 0:
        flat_dp1, _ = run_training(device)  # dp.is_active() now False
        max_abs = float(np.abs(flat_dp - flat_dp1).max())
        match = bool(np.array_equal(flat_dp, flat_dp1))
        print(f"DP{world} vs DP1 on same data: bitwise_equal={match} "
              f"max_abs_diff={max_abs:.3e}", flush=True)
        assert match, f"DP{world} != DP1 (max abs diff {max_abs})"
        os.makedirs("gpurun_out", exist_ok=True)
        with open("gpurun_out/rccl_world2.json", "w") as f:
            json.dump({
                "world_size": world, "backend": backend,
                "device": torch.cuda.get_device_name(0),
                "compute_device": "cuda",
                "params_bitwise_in_sync": in_sync,
                "dp2_equals_dp1_bitwise": match,
                "last_info_finite": losses_finite,
                "info_keys": sorted(info.keys()),
            }, f, indent=1)
        print("PASS: DP world=2 validated on device", flush=True)


if __name__ == "__main__":
    main()
