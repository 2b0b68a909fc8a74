"""Rollout collection + metric helpers (reference trainer/utils.py:25-55)."""
from __future__ import annotations

import json
import os
import time
from typing import Callable, Optional

import numpy as np
import torch
from torch import Tensor

from ..utils.graph import GraphBatch
from .data import Rollout


@torch.no_grad()
def collect_rollout(env, act_fn: Callable, graph0: GraphBatch, graphed=None) -> Rollout:
    """Roll all B worlds forward max_episode_steps with act_fn(graph)->action.

    The reference scans a jitted body (trainer/utils.py:43-53); here the body
    is a batch of device kernels per step, replayed from a captured HIP graph
    when ``graphed`` (a GraphedRolloutStep) is provided on GPU.
    """
    B, T = graph0.batch_size, env.max_episode_steps
    V, S = graph0.n_nodes, graph0.state_dim
    N, D = graph0.n_agents, graph0.n_edge_slots
    dev = graph0.device
    states = torch.empty(B, T, V, S, device=dev)
    masks = torch.empty(B, T, N, D, dtype=torch.bool, device=dev)
    actions = torch.empty(B, T, N, env.action_dim, device=dev)
    rewards = torch.empty(B, T, device=dev)
    costs = torch.empty(B, T, device=dev)
    dones = torch.zeros(B, T, dtype=torch.bool, device=dev)

    if graphed is not None and graph0.states.is_cuda:
        graphed.reset(graph0)
        if not graphed.try_capture():
            graphed = None  # capture-illegal env step: eager fallback

    if graphed is not None and graph0.states.is_cuda:
        for t in range(T):
            states[:, t] = graphed.static_in.states
            masks[:, t] = graphed.static_in.mask
            action, _, reward, cost, done = graphed.step()
            actions[:, t] = action
            rewards[:, t] = reward
            costs[:, t] = cost
            dones[:, t] = done
            graphed.advance()
        return Rollout(
            states=states, masks=masks, actions=actions, rewards=rewards, costs=costs,
            dones=dones, next_states=graphed.static_in.states.clone(),
            next_mask=graphed.static_in.mask.clone(), obstacles=graph0.env_states,
        )

    graph = graph0
    for t in range(T):
        action = act_fn(graph)
        if isinstance(action, tuple):
            action = action[0]
        states[:, t] = graph.states
        masks[:, t] = graph.mask
        actions[:, t] = action
        res = env.step(graph, action)
        rewards[:, t] = res.reward
        costs[:, t] = res.cost
        dones[:, t] = res.done
        graph = res.graph

    return Rollout(
        states=states, masks=masks, actions=actions, rewards=rewards, costs=costs,
        dones=dones, next_states=graph.states, next_mask=graph.mask,
        obstacles=graph0.env_states,
    )


@torch.no_grad()
def eval_rollout_metrics(env, rollout: Rollout) -> dict:
    """Eval metrics as the reference trainer computes them
    (trainer/trainer.py:105-129)."""
    total_reward = rollout.rewards.sum(dim=-1)
    g = rollout.graph_at(env)
    b, T = rollout.rewards.shape[:2]
    finish = env.finish_mask(g).reshape(b, T, -1).float()
    finish = finish.amax(dim=1).mean()
    cost = rollout.costs.sum(dim=-1).mean()
    unsafe_frac = (rollout.costs.amax(dim=-1) >= 1e-6).float().mean()
    return {
        "eval/reward": float(total_reward.mean()),
        "eval/reward_min": float(total_reward.min()),
        "eval/reward_max": float(total_reward.max()),
        "eval/reward_final": float(rollout.rewards[:, -1].mean()),
        "eval/cost": float(cost),
        "eval/unsafe_frac": float(unsafe_frac),
        "eval/finish": float(finish),
    }


class MetricsLogger:
    """stdout + JSONL metrics; wandb only if importable (no network here)."""

    def __init__(self, log_dir: Optional[str], run_name: str = "run", use_wandb: bool = True):
        self.log_dir = log_dir
        self._f = None
        if log_dir is not None:
            os.makedirs(log_dir, exist_ok=True)
            self._f = open(os.path.join(log_dir, "metrics.jsonl"), "a")
        self._wandb = None
        if use_wandb:
            try:  # pragma: no cover - wandb absent in this image
                import wandb

                self._wandb = wandb
                wandb.init(name=run_name, project="gcbf-amd", dir=log_dir, mode="offline")
            except Exception:
                self._wandb = None

    def log(self, metrics: dict, step: int):
        if self._f is not None:
            self._f.write(json.dumps({"step": step, **metrics}) + "\n")
            self._f.flush()
        if self._wandb is not None:
            self._wandb.log(metrics, step=step)

    def close(self):
        if self._f is not None:
            self._f.close()


# ---- misc reference-parity helpers (reference trainer/utils.py:58-177) ----

def has_any_nan(tree) -> bool:
    """True if any tensor leaf contains a non-finite value (reference :58-63)."""
    from ..utils.utils import tree_map

    found = []
    tree_map(lambda t: found.append(not torch.isfinite(t).all().item()), tree)
    return any(found)


def tree_copy(tree):
    """Deep clone of a tensor tree (reference :78-79)."""
    from ..utils.utils import tree_map

    return tree_map(lambda t: t.detach().clone(), tree)


def is_connected(host: str = "8.8.8.8", port: int = 53, timeout: float = 2.0) -> bool:
    """Network reachability probe gating online logging (reference :100-109)."""
    import socket

    try:
        socket.setdefaulttimeout(timeout)
        socket.socket(socket.AF_INET, socket.SOCK_STREAM).connect((host, port))
        return True
    except OSError:
        return False


def centered_norm(vmin, vmax):
    """Zero-centered colormap normalization (reference :171-177)."""
    from matplotlib.colors import CenteredNorm

    if isinstance(vmin, list):
        vmin = min(vmin)
    if isinstance(vmax, list):
        vmax = max(vmax)
    halfrange = max(abs(vmin), abs(vmax))
    return CenteredNorm(0, halfrange)


def plot_cbf(fig, cbf, env, graph, agent_id: int, x_dim: int = 0, y_dim: int = 1,
             n_mesh: int = 30):
    """Filled CBF contours over a 2D slice of one agent's state
    (reference :112-146): moves agent ``agent_id`` over an x/y mesh, rebuilds
    the graph per mesh point and evaluates h."""
    from ..utils.graph import GraphBatch

    b0 = 0
    states = graph.states[b0]
    mask = graph.mask[b0]
    side = env.area_size
    xs = torch.linspace(0, side, n_mesh)
    ys = torch.linspace(0, side, n_mesh)
    gx, gy = torch.meshgrid(xs, ys, indexing="xy")
    grid_states = states[None, None].repeat(n_mesh, n_mesh, 1, 1)
    grid_states[:, :, agent_id, x_dim] = gx.to(states.device)
    grid_states[:, :, agent_id, y_dim] = gy.to(states.device)
    flat = grid_states.reshape(n_mesh * n_mesh, *states.shape)
    g = GraphBatch(states=flat,
                   mask=mask[None].expand(n_mesh * n_mesh, *mask.shape).contiguous(),
                   n_agents=env.num_agents, n_rays=env.n_rays)
    with torch.no_grad():
        h = cbf(g)[:, agent_id, 0].reshape(n_mesh, n_mesh).cpu().numpy()
    ax = fig.gca()
    x, y = np.meshgrid(xs.numpy(), ys.numpy())
    cf = ax.contourf(x, y, h, levels=15, alpha=0.5, cmap="magma")
    fig.colorbar(cf, ax=ax)
    ax.contour(x, y, h, levels=[0.0], colors="blue")
    ax.axis("off")
    return fig
