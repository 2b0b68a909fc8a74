"""Bisect the n=64 GPU NaN: check finiteness of every stage from rollout to
QP inputs (tools aid; not part of the test suite)."""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from gcbfplus_amd.env import make_env
from gcbfplus_amd.algo import make_algo
from gcbfplus_amd.trainer.utils import collect_rollout


def chk(name, t):
    t = t if isinstance(t, torch.Tensor) else torch.as_tensor(t)
    ok = torch.isfinite(t.float()).all().item()
    mx = t.float().abs().max().item() if ok else float("nan")
    print(f"{name:24s} finite={ok} max|.|={mx:.4g} shape={tuple(t.shape)}", flush=True)
    return ok


def main():
    torch.manual_seed(64)
    env = make_env("DoubleIntegrator", num_agents=64, area_size=8.0, max_step=8,
                   device="cuda")
    algo = make_algo("gcbf+", env=env, node_dim=env.node_dim, edge_dim=env.edge_dim,
                     state_dim=env.state_dim, action_dim=env.action_dim,
                     n_agents=64, gnn_layers=1, batch_size=8, buffer_size=16,
                     horizon=4, inner_epoch=1, seed=0)
    rng = np.random.default_rng(65)
    g = env.reset(2, rng)
    chk("reset.states", g.states)
    ro = collect_rollout(env, algo.step, g)
    chk("rollout.states", ro.states)
    chk("rollout.actions", ro.actions)
    chk("rollout.next_states", ro.next_states)

    b, T = ro.rewards.shape[:2]
    gall = ro.graph_at(env)
    from gcbfplus_amd.trainer.data import FlatBatch
    flat = FlatBatch(
        states=ro.states.reshape(b * T, *ro.states.shape[2:]),
        masks=ro.masks.reshape(b * T, *ro.masks.shape[2:]),
        safe=torch.zeros(b * T, 64, dtype=torch.bool, device="cuda"),
        unsafe=torch.zeros(b * T, 64, dtype=torch.bool, device="cuda"),
    )
    mb = flat[torch.arange(2, device="cuda")]
    graph = mb.graph(env)
    h, h_x = algo.cbf_and_jacobian(graph, algo.cbf_tgt)
    chk("h", h)
    chk("h_x", h_x)
    agent = graph.agent_states
    f, gdyn = env.control_affine_dyn(agent)
    chk("f", f)
    chk("gdyn", gdyn)
    uref = env.u_ref(graph)
    chk("u_ref", uref)
    Lf_h = torch.einsum("mijs,mjs->mi", h_x, f)
    chk("Lf_h", Lf_h)

    # assemble the QP exactly as get_qp_action and dump the inputs so the
    # failure reproduces offline
    N, nu = 64, 2
    M = graph.batch_size
    Lg_h = torch.einsum("mijs,mjsu->miju", h_x, gdyn).reshape(M, N, N * nu)
    u_lb, u_ub = env.action_lim()
    dev = agent.device
    u_lb = u_lb.to(dev).repeat(N)
    u_ub = u_ub.to(dev).repeat(N)
    nv = N * nu + N
    H = torch.eye(nv, device=dev).expand(M, nv, nv).clone()
    H[:, N * nu:, N * nu:] *= 10.0
    gvec = torch.cat([-uref.reshape(M, N * nu), 1e3 * torch.ones(M, N, device=dev)], 1)
    eyeN = torch.eye(N, device=dev).expand(M, N, N)
    C = -torch.cat([Lg_h, eyeN], dim=2)
    bvec = Lf_h + algo.alpha * 0.1 * h
    l_box = torch.cat([u_lb, torch.zeros(N, device=dev)]).expand(M, nv)
    u_box = torch.cat([u_ub, torch.full((N,), float("inf"), device=dev)]).expand(M, nv)
    import os as _os
    _os.makedirs("gpurun_out", exist_ok=True)
    torch.save({k: v.cpu() for k, v in
                dict(H=H, g=gvec, C=C, b=bvec, l=l_box, u=u_box).items()},
               "gpurun_out/qp_inputs_n64.pt")
    print("saved QP inputs", flush=True)

    from gcbfplus_amd.ops.qp import proxqp_solve
    # instrumented solve on GPU: try and report
    try:
        x = proxqp_solve(H, gvec, C, bvec, l_box, u_box, iters=100)
        chk("x", x)
    except Exception as e:
        print("GPU proxqp FAILED:", type(e).__name__, e, flush=True)
    # same data on CPU
    xc = proxqp_solve(H.cpu(), gvec.cpu(), C.cpu(), bvec.cpu(), l_box.cpu(),
                      u_box.cpu(), iters=100)
    chk("x_cpu", xc)
    print("ALL STAGES DONE", flush=True)


if __name__ == "__main__":
    main()
