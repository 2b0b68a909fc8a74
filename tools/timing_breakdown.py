"""Per-phase wall timing of one GCBF+ training step on GPU (dev tool)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import numpy as np
import torch

from gcbfplus_amd.algo import make_algo
from gcbfplus_amd.env import make_env
from gcbfplus_amd.trainer.utils import collect_rollout
from gcbfplus_amd.algo.utils import horizon_safe_mask, polyak_


def sync():
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def main():
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    env = make_env("DoubleIntegrator", num_agents=8, area_size=4.0, max_step=256, device=dev)
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim, n_agents=8,
                     gnn_layers=1, batch_size=256, buffer_size=512, horizon=32,
                     inner_epoch=8, lr_actor=1e-5, lr_cbf=1e-5,
                     loss_action_coef=1e-4, loss_h_dot_coef=0.01, seed=0)
    rng = np.random.default_rng(0)

    # warm step 1 (populates buffers, compiles nothing but warms pools)
    g0 = env.reset(16, rng)
    ro = collect_rollout(env, algo.step, g0)
    algo.update(ro, 0)
    sync()

    for rep in range(2):
        t = {}
        t0 = time.perf_counter()
        g0 = env.reset(16, rng)
        sync()
        t["reset(host sampling)"] = time.perf_counter() - t0

        t0 = time.perf_counter()
        ro = collect_rollout(env, algo.step, g0)
        sync()
        t["rollout 256 steps"] = time.perf_counter() - t0

        t0 = time.perf_counter()
        g = ro.graph_at(env)
        unsafe = env.unsafe_mask(g).reshape(16, 256, 8)
        safe = horizon_safe_mask(unsafe, 32)
        sync()
        t["masks"] = time.perf_counter() - t0

        t0 = time.perf_counter()
        batch = algo._sample_batch(ro, safe, unsafe)
        sync()
        t["buffer sample"] = time.perf_counter() - t0

        t0 = time.perf_counter()
        u_qp = algo._get_b_u_qp(batch, n_chunks=8)
        sync()
        t[f"qp labels ({batch.n})"] = time.perf_counter() - t0
        batch = batch._replace(u_qp=u_qp)

        t0 = time.perf_counter()
        n_epochs_timed = 2
        for ep in range(n_epochs_timed):
            perm = torch.from_numpy(algo.rng.permutation(batch.n)).to(batch.states.device)
            chunks = torch.chunk(perm, max(1, batch.n // algo.batch_size))
            for mb_idx in chunks:
                algo._update_minibatch(batch[mb_idx], want_info=False)
        sync()
        dt = time.perf_counter() - t0
        t[f"inner epochs (x{len(chunks)} mb, scaled to 8)"] = dt * 8 / n_epochs_timed

        t0 = time.perf_counter()
        polyak_(algo.cbf_tgt, algo.cbf, 0.5)
        sync()
        t["polyak"] = time.perf_counter() - t0

        # the real path (graphed minibatches) for comparison
        t0 = time.perf_counter()
        g0b = env.reset(16, rng)
        rob = collect_rollout(env, algo.step, g0b)
        sync()
        tb = time.perf_counter()
        algo.update(rob, 99)
        sync()
        t["FULL algo.update (graphed)"] = time.perf_counter() - tb

        total = sum(t.values())
        print(f"--- rep {rep}: total ~{total:.3f}s ---")
        for k, v in t.items():
            print(f"  {k:38s} {v*1000:9.1f} ms  ({100*v/total:4.1f}%)")


if __name__ == "__main__":
    main()
