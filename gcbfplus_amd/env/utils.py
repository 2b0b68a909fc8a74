"""Env math helpers: LiDAR scan, discrete LQR, start/goal sampling.

Mirrors ``/root/reference/gcbfplus/env/utils.py`` semantics:
  - get_lidar (49-79): 2D fan of n beams / 3D theta x phi grid + 2 poles
  - raytracing (110-131): per-beam min-alpha over obstacles; points inside an
    obstacle return alpha=0 (hit at the start point); the reference then takes
    the ``max_returns`` *closest* hits (argsort). When max_returns == n_beams
    the selection is a permutation only — attention aggregation is
    permutation-invariant, so we skip the sort in that case.
  - lqr (24-46): discrete-ARE gain via scipy
  - get_node_goal_rng (134-226): rejection sampling of non-colliding
    starts/goals — perf-irrelevant (once per rollout), done in numpy on host.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from .obstacle import Cuboid, Rectangle, Sphere


def lqr(A: np.ndarray, B: np.ndarray, Q: np.ndarray, R: np.ndarray) -> np.ndarray:
    """Discrete-time LQR gain K for x+ = Ax + Bu (reference env/utils.py:24-46)."""
    from scipy.linalg import inv, solve_discrete_are

    X = solve_discrete_are(A, B, Q, R)
    return inv(B.T @ X @ B + R) @ (B.T @ X @ A)


def beam_dirs_2d(n_beams: int, device=None) -> Tensor:
    """(R, 2) unit beam directions, thetas = linspace(-pi, pi - 2pi/R, R)."""
    thetas = torch.linspace(-math.pi, math.pi - 2 * math.pi / n_beams, n_beams, device=device)
    return torch.stack([torch.cos(thetas), torch.sin(thetas)], dim=-1)


def beam_dirs_3d(n_beams: int, device=None) -> Tensor:
    """(R2+2, 3) unit beam dirs: theta x phi grid + up/down poles
    (reference env/utils.py:56-75). R2 = (n_beams//2) * n_beams."""
    nt = n_beams // 2
    thetas = torch.linspace(
        -math.pi / 2 + 2 * math.pi / n_beams, math.pi / 2 - 2 * math.pi / n_beams, nt, device=device
    )
    phis = torch.linspace(-math.pi, math.pi - 2 * math.pi / n_beams, n_beams, device=device)
    th = thetas[:, None].expand(nt, n_beams).reshape(-1)
    ph = phis[None, :].expand(nt, n_beams).reshape(-1)
    dirs = torch.stack(
        [torch.cos(th) * torch.cos(ph), torch.cos(th) * torch.sin(ph), torch.sin(th)], dim=-1
    )
    poles = torch.tensor([[0.0, 0.0, 1.0], [0.0, 0.0, -1.0]], device=device)
    return torch.cat([dirs, poles], dim=0)


def get_lidar(
    pos: Tensor,
    obstacles,
    n_rays: int,
    sense_range: float,
    max_returns: Optional[int] = None,
) -> Tensor:
    """Batched LiDAR scan.

    pos: (B, N, dim) scan origins. Returns hit points (B, N, max_returns, dim).

    2D (Rectangle): fan of ``n_rays`` beams; 3D (Cuboid/Sphere): grid + poles.
    Beams with no hit within range land at alpha=1e6 (far away -> masked by
    comm radius downstream, reference behavior). Origins inside an obstacle
    return the origin itself (alpha=0), reference env/utils.py:110-124.
    """
    B, N, dim = pos.shape
    if isinstance(obstacles, Rectangle):
        dirs = beam_dirs_2d(n_rays, device=pos.device)  # (R,2)
    else:
        dirs = beam_dirs_3d(n_rays, device=pos.device)  # (R,3)
    R = dirs.shape[0]
    if max_returns is None:
        max_returns = R
    starts = pos[:, :, None, :].expand(B, N, R, dim).reshape(B, N * R, dim)
    ends = starts + sense_range * dirs[None, None].expand(B, N, R, dim).reshape(B, N * R, dim)
    alphas = obstacles.raytrace(starts, ends)  # (B, N*R)
    is_in = obstacles.inside(pos)  # (B, N)
    alphas = alphas * (~is_in).float().repeat_interleave(R, dim=1)
    hits = starts + (ends - starts) * alphas[..., None]
    hits = hits.reshape(B, N, R, dim)
    if max_returns < R:
        # stable: the reference's jnp.argsort keeps tie order (all the
        # alpha=1e6 no-hit beams), so the selected beam INDICES matter, not
        # just the sorted distances — torch default argsort is unstable
        idx = torch.argsort(alphas.reshape(B, N, R), dim=-1, stable=True)[:, :, :max_returns]
        hits = torch.gather(hits, 2, idx[..., None].expand(B, N, max_returns, dim))
    return hits


def sample_starts_goals(
    rng: np.random.Generator,
    batch: int,
    n_agents: int,
    dim: int,
    side_length: float,
    obstacles_np,
    min_dist: float,
    max_travel: Optional[float] = None,
    max_iter: int = 1024,
) -> Tuple[np.ndarray, np.ndarray]:
    """Host-side rejection sampling of valid starts/goals, one env at a time
    (reference env/utils.py:134-226, without the jit contortions).

    obstacles_np: callable (points (M,dim), r) -> (M,) bool inside-test, per env:
        obstacles_np(b, pts, r)
    Returns (batch, n, dim) starts and goals.
    """
    starts = np.zeros((batch, n_agents, dim), dtype=np.float32)
    goals = np.zeros((batch, n_agents, dim), dtype=np.float32)
    for b in range(batch):
        placed_s = np.full((n_agents, dim), 1e6, dtype=np.float32)
        placed_g = np.full((n_agents, dim), 1e6, dtype=np.float32)
        for i in range(n_agents):
            for _ in range(max_iter):
                cand = rng.uniform(0, side_length, size=dim).astype(np.float32)
                if np.linalg.norm(placed_s - cand, axis=1).min() <= min_dist:
                    continue
                if obstacles_np(b, cand[None], min_dist)[0]:
                    continue
                placed_s[i] = cand
                break
            else:
                placed_s[i] = rng.uniform(0, side_length, size=dim)
            for _ in range(max_iter):
                if max_travel is None:
                    cand = rng.uniform(0, side_length, size=dim).astype(np.float32)
                else:
                    cand = (placed_s[i] + rng.uniform(-max_travel, max_travel, size=dim)).astype(
                        np.float32
                    )
                if np.linalg.norm(placed_g - cand, axis=1).min() <= min_dist:
                    continue
                if obstacles_np(b, cand[None], min_dist)[0]:
                    continue
                if np.any(cand < 0) or np.any(cand > side_length):
                    continue
                if max_travel is not None and np.linalg.norm(cand - placed_s[i]) > max_travel:
                    continue
                placed_g[i] = cand
                break
            else:
                placed_g[i] = placed_s[i]
        starts[b] = placed_s
        goals[b] = placed_g
    return starts, goals
