"""RCCL validation on real hardware: run 2 ranks (sharing one MI355X, or one
per GPU when more are visible) under torchrun and prove the whole DP path
executes over the nccl(=RCCL) backend:

  - init_process_group("nccl") + broadcast_modules
  - dp.all_agree's nccl branch (device tensor collective)
  - the per-minibatch fused-bucket all-reduce (algo.dp_gbuf)
  - 2 full GCBF+ training steps on device, params bit-identical across ranks
  - DP2-on-same-data == DP1 exactly (fp32 mean of identical grads is exact)

Usage (on a GPU box):
  python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
      --nnodes=1 --nproc-per-node 2 tools/rccl_check.py

Writes gpurun_out/rccl_world2.json from rank 0 on success.
"""
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
    for step in range(steps):
        g = env.reset(2, rng)
        ro = collect_rollout(env, algo.step, g)
        info = algo.update(ro, step)
    flat = torch.cat([p.detach().reshape(-1) for p in algo.cbf.parameters()]
                     + [p.detach().reshape(-1) for p in algo.actor.parameters()])
    return flat.cpu().numpy(), info


def main():
    assert torch.cuda.is_available(), "needs a GPU"
    local = dp.setup_from_env()
    assert dist.is_initialized() and dist.get_backend() == "nccl", \
        f"expected nccl backend, got {dist.get_backend() if dist.is_initialized() else 'none'}"
    world = dist.get_world_size()
    rank = dist.get_rank()
    device = f"cuda:{local % torch.cuda.device_count()}"
    print(f"[rank {rank}/{world}] nccl up on {device} "
          f"({torch.cuda.get_device_name(0)})", flush=True)

    # exercise the nccl branch of all_agree both ways
    assert dp.all_agree(True) is True
    assert dp.all_agree(rank == 0) is False  # MIN over {1,0} = 0

    flat_dp, info = run_training(device)
    # cross-rank bitwise check
    t = torch.from_numpy(flat_dp).to(device)
    tmax = t.clone()
    dist.all_reduce(tmax, op=dist.ReduceOp.MAX)
    tmin = t.clone()
    dist.all_reduce(tmin, op=dist.ReduceOp.MIN)
    in_sync = bool(torch.equal(tmax, tmin))
    assert in_sync, "params diverged across ranks"
    losses_finite = all(np.isfinite(v) for v in info.values())
    print(f"[rank {rank}] params in sync across {world} ranks; "
          f"last-minibatch info finite={losses_finite}", flush=True)
    assert losses_finite
    dist.barrier()
    dist.destroy_process_group()

    if rank == 0:
        # DP1 reference on the same data (dp.is_active() is now False)
        flat_dp1, _ = run_training(device)
        max_abs = float(np.abs(flat_dp - flat_dp1).max())
        match = bool(np.array_equal(flat_dp, flat_dp1))
        print(f"DP{world} vs DP1 on same data: bitwise_equal={match} "
              f"max_abs_diff={max_abs:.3e}", flush=True)
        assert match, f"DP{world} != DP1 (max abs diff {max_abs})"
        os.makedirs("gpurun_out", exist_ok=True)
        with open("gpurun_out/rccl_world2.json", "w") as f:
            json.dump({
                "world_size": world, "backend": "nccl(RCCL)",
                "device": torch.cuda.get_device_name(0),
                "all_agree_nccl_branch": True,
                "params_bitwise_in_sync": in_sync,
                "dp2_equals_dp1_bitwise": match,
                "last_info_finite": losses_finite,
                "info_keys": sorted(info.keys()),
            }, f, indent=1)
        print("PASS: RCCL DP validated on device", flush=True)


if __name__ == "__main__":
    main()
