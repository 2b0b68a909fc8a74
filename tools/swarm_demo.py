"""Large-swarm streamed rollout demo (BASELINE config #5 evidence):
DoubleIntegrator n=512 and CrazyFlie n=256 on ONE MI355X via the
no-stored-graphs eval path (reference env/base.py:191-259 / test.py
--nojit-rollout). Reports env-steps/s and peak HBM; writes
gpurun_out/swarm_demo.json."""
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from gcbfplus_amd.env import make_env
from gcbfplus_amd.algo import make_algo


def run(env_id, n, area, steps, use_algo=True, **env_kw):
    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()
    env = make_env(env_id, n, area_size=area, max_step=steps, device="cuda", **env_kw)
    if use_algo:
        algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                         state_dim=env.state_dim, action_dim=env.action_dim, n_agents=n)
        act = algo.act
    else:
        act = env.u_ref
    rng = np.random.default_rng(0)
    t_reset = time.perf_counter()
    g = env.reset(1, rng)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    ever_coll = env.collision_mask(g).float()
    ever_fin = env.finish_mask(g).float()
    with torch.no_grad():
        for _ in range(steps):
            a = act(g)
            g = env.step(g, a).graph
            ever_coll = torch.maximum(ever_coll, env.collision_mask(g).float())
            ever_fin = torch.maximum(ever_fin, env.finish_mask(g).float())
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    peak = torch.cuda.max_memory_allocated() / 2**30
    rec = {
        "env": env_id, "n_agents": n, "area": area, "steps": steps,
        "policy": "gcbf+ (random init)" if use_algo else "u_ref",
        "reset_s": round(t0 - t_reset, 2),
        "rollout_s": round(dt, 3),
        "env_steps_per_s": round(steps / dt, 2),
        "agent_steps_per_s": round(steps * n / dt, 1),
        "peak_hbm_gb": round(peak, 3),
        "finite": bool(torch.isfinite(g.states).all()),
        "safe_rate": round(float(1 - ever_coll.mean()), 4),
    }
    print(json.dumps(rec), flush=True)
    return rec


recs = []
recs.append(run("DoubleIntegrator", 512, 32.0, 64))
recs.append(run("DoubleIntegrator", 1024, 45.0, 32))
recs.append(run("CrazyFlie", 256, 16.0, 64))
recs.append(run("LinearDrone", 512, 24.0, 64))
os.makedirs("gpurun_out", exist_ok=True)
with open("gpurun_out/swarm_demo.json", "w") as f:
    json.dump({"device": torch.cuda.get_device_name(0),
               "hbm_total_gb": torch.cuda.get_device_properties(0).total_memory / 2**30,
               "runs": recs}, f, indent=1)
print("DONE", flush=True)
