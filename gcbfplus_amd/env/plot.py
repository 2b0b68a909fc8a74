"""Rollout rendering: 2D/3D agent/goal/obstacle animation + optional CBF
contour overlay (reference ``gcbfplus/env/plot.py:24-413``).

matplotlib only (lazily imported); with no ffmpeg in the image the writer
falls back to Pillow (*.gif) — pass a .gif path or let render_video rewrite
the suffix.
"""
from __future__ import annotations

import pathlib
from typing import Callable, Optional

import numpy as np
import torch


def _obstacle_patches(obstacles, b: int, ax):
    import matplotlib.patches as mpatches
    from .obstacle import Rectangle, Sphere

    patches = []
    if isinstance(obstacles, Rectangle):
        pts = obstacles.points[b].cpu().numpy()  # (K,4,2)
        for kk in range(pts.shape[0]):
            patches.append(mpatches.Polygon(pts[kk], closed=True, color="dimgray"))
    elif isinstance(obstacles, Sphere):
        c = obstacles.center[b].cpu().numpy()
        r = obstacles.radius[b].cpu().numpy()
        for kk in range(c.shape[0]):
            patches.append(mpatches.Circle(c[kk, :2], r[kk], color="dimgray"))
    for p in patches:
        ax.add_patch(p)
    return patches


def render_video(
    rollout,
    video_path,
    env,
    b: int = 0,
    Ta_is_unsafe: Optional[torch.Tensor] = None,
    cbf_fn: Optional[Callable] = None,
    cbf_agent: Optional[int] = None,
    dpi: int = 100,
    fps: int = 30,
    viz_opts: Optional[dict] = None,
    **kwargs,
):
    """Animate world ``b`` of a Rollout to video_path (gif via Pillow when no
    ffmpeg). cbf_fn(graph) -> h (B,N,1) adds a CBF heatmap for ``cbf_agent``
    (reference plot.py:315-337)."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    from matplotlib.animation import FuncAnimation, PillowWriter

    side = env.area_size
    n, r = env.num_agents, env.params.get("car_radius", env.params.get("drone_radius", 0.05))
    dim = env.pos_dim
    T = rollout.states.shape[1]
    states = rollout.states[b].cpu()  # (T, V, S)

    fig = plt.figure(figsize=(7, 7), dpi=dpi)
    if dim == 3:
        ax = fig.add_subplot(projection="3d")
        ax.set_xlim(0, side), ax.set_ylim(0, side), ax.set_zlim(0, side)
    else:
        ax = fig.add_subplot()
        ax.set_xlim(0, side), ax.set_ylim(0, side)
        ax.set_aspect("equal")
        _obstacle_patches(rollout.obstacles, b, ax)
    ax.set_title(type(env).__name__)

    goal = states[0, n : 2 * n, :dim].numpy()
    if dim == 3:
        goal_sc = ax.scatter(goal[:, 0], goal[:, 1], goal[:, 2], marker="*", c="green", s=90)
        agent_sc = ax.scatter([], [], [], c="tab:blue", s=40)
        hit_sc = ax.scatter([], [], [], c="red", s=3)
    else:
        ax.scatter(goal[:, 0], goal[:, 1], marker="*", c="green", s=120, zorder=4)
        circles = [
            plt.Circle((0, 0), r, color="tab:blue", zorder=5) for _ in range(n)
        ]
        for c in circles:
            ax.add_patch(c)
        hit_sc = ax.scatter([], [], c="red", s=3, zorder=3)
        contour_state = {"cs": None}

    def update(t):
        ag = states[t, :n, :dim].numpy()
        hits = states[t, 2 * n :, :dim].numpy()
        hits = hits[np.abs(hits).max(axis=1) < side * 3]
        unsafe = None
        if Ta_is_unsafe is not None and t < Ta_is_unsafe.shape[0]:
            unsafe = Ta_is_unsafe[t].cpu().numpy()
        if dim == 3:
            agent_sc._offsets3d = (ag[:, 0], ag[:, 1], ag[:, 2])
            hit_sc._offsets3d = (hits[:, 0], hits[:, 1], hits[:, 2])
            return [agent_sc, hit_sc]
        for i, c in enumerate(circles):
            c.center = ag[i]
            c.set_color("red" if unsafe is not None and unsafe[i] else "tab:blue")
        hit_sc.set_offsets(hits[:, :2] if len(hits) else np.zeros((0, 2)))
        if cbf_fn is not None and cbf_agent is not None:
            if contour_state["cs"] is not None:
                contour_state["cs"].remove()
            xs, ys, hh = get_bb_cbf(cbf_fn, env, rollout, b, t, cbf_agent)
            contour_state["cs"] = ax.contourf(
                xs, ys, hh, levels=15, alpha=0.4, cmap="RdBu"
            )
        return circles + [hit_sc]

    anim = FuncAnimation(fig, update, frames=T, interval=1000 // fps, blit=False)
    video_path = pathlib.Path(video_path)
    if video_path.suffix != ".gif":
        video_path = video_path.with_suffix(".gif")
    anim.save(str(video_path), writer=PillowWriter(fps=fps))
    plt.close(fig)
    return video_path


@torch.no_grad()
def get_bb_cbf(cbf_fn, env, rollout, b: int, t: int, agent_id: int, n_mesh: int = 20):
    """CBF heatmap over a position mesh for one agent (reference
    trainer/utils.py:149-168): sweep agent_id's (x, y), rebuild edges, eval h."""
    from ..utils.graph import GraphBatch

    side = env.area_size
    xs = torch.linspace(0, side, n_mesh)
    ys = torch.linspace(0, side, n_mesh)
    states = rollout.states[b, t]
    mask = rollout.masks[b, t]
    dev = states.device
    grid_states = states[None, None].repeat(n_mesh, n_mesh, 1, 1)
    gx, gy = torch.meshgrid(xs, ys, indexing="xy")
    grid_states[:, :, agent_id, 0] = gx
    grid_states[:, :, agent_id, 1] = gy
    flat = grid_states.reshape(n_mesh * n_mesh, *states.shape).to(dev)
    g = GraphBatch(states=flat, mask=mask[None].expand(n_mesh * n_mesh, *mask.shape).to(dev),
                   n_agents=env.num_agents, n_rays=env.n_rays)
    h = cbf_fn(g)[:, agent_id, 0].reshape(n_mesh, n_mesh)
    return gx.numpy(), gy.numpy(), h.cpu().numpy()
