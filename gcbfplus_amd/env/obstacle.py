"""Batched geometric obstacle primitives (torch, CPU/GPU).

Semantics mirror the reference's per-obstacle jax NamedTuples
(``/root/reference/gcbfplus/env/obstacle.py``): Rectangle (2D, 25-96),
Cuboid (3D, 99-222), Sphere (3D, 225-270) — re-expressed as *batched*
tensor ops with leading (B, K) dims (B envs, K obstacles each), which is the
shape the CDNA4 raytrace kernel consumes directly.
"""
from __future__ import annotations

import math
from typing import NamedTuple

import torch
from torch import Tensor


class Rectangle(NamedTuple):
    """B x K rotated rectangles. center (B,K,2), wh (B,K,2), theta (B,K),
    points (B,K,4,2) corner cache (ccw from +x+y corner, reference order)."""

    center: Tensor
    wh: Tensor
    theta: Tensor
    points: Tensor

    @staticmethod
    def create(center: Tensor, width: Tensor, height: Tensor, theta: Tensor) -> "Rectangle":
        # corners in body frame, order as reference obstacle.py:35-41
        hw, hh = width / 2, height / 2
        bx = torch.stack([hw, -hw, -hw, hw], dim=-1)  # (B,K,4)
        by = torch.stack([hh, hh, -hh, -hh], dim=-1)
        c, s = torch.cos(theta), torch.sin(theta)
        px = c[..., None] * bx - s[..., None] * by + center[..., 0:1]
        py = s[..., None] * bx + c[..., None] * by + center[..., 1:2]
        points = torch.stack([px, py], dim=-1)  # (B,K,4,2)
        wh = torch.stack([width, height], dim=-1)
        return Rectangle(center, wh, theta, points)

    @property
    def n_obs(self) -> int:
        return self.center.shape[1]

    def inside(self, point: Tensor, r: float = 0.0) -> Tensor:
        """point (B, M, 2) -> (B, M) bool: inside any rectangle, inflated by r.

        Mirrors reference obstacle.py:52-63 (rounded-corner inflation).
        """
        if self.n_obs == 0:
            return torch.zeros(point.shape[:-1], dtype=torch.bool, device=point.device)
        rel = point[:, :, None, :] - self.center[:, None, :, :]  # (B,M,K,2)
        c = torch.cos(self.theta)[:, None, :]
        s = torch.sin(self.theta)[:, None, :]
        rel_xx = (rel[..., 0] * c + rel[..., 1] * s).abs() - self.wh[:, None, :, 0] / 2
        rel_yy = (rel[..., 0] * s - rel[..., 1] * c).abs() - self.wh[:, None, :, 1] / 2
        in_down = (rel_xx < r) & (rel_yy < 0)
        in_up = (rel_xx < 0) & (rel_yy < r)
        out_corner = (rel_xx > 0) & (rel_yy > 0)
        in_circle = torch.sqrt(rel_xx**2 + rel_yy**2) < r
        is_in = in_down | in_up | (out_corner & in_circle)
        return is_in.any(dim=-1)

    def raytrace(self, starts: Tensor, ends: Tensor) -> Tensor:
        """starts/ends (B, M, 2) -> (B, M) min alpha in [0,1] over all K
        rectangles' 4 edges, 1e6 where no hit (reference obstacle.py:65-96)."""
        if self.n_obs == 0:
            return torch.full(starts.shape[:-1], 1e6, device=starts.device)
        x1 = starts[:, :, None, None, 0]
        y1 = starts[:, :, None, None, 1]
        x2 = ends[:, :, None, None, 0]
        y2 = ends[:, :, None, None, 1]
        x3 = self.points[:, None, :, :, 0]  # (B,1,K,4)
        y3 = self.points[:, None, :, :, 1]
        x4 = self.points[:, None, :, [3, 0, 1, 2], 0]
        y4 = self.points[:, None, :, [3, 0, 1, 2], 1]
        det = (x1 - x2) * (y4 - y3) - (y1 - y2) * (x4 - x3)
        det = torch.sign(det) * torch.clamp(det.abs(), 1e-7, 1e7)
        alphas = ((y4 - y3) * (x1 - x3) - (x4 - x3) * (y1 - y3)) / det
        betas = (-(y1 - y2) * (x1 - x3) + (x1 - x2) * (y1 - y3)) / det
        valid = (alphas >= 0) & (alphas <= 1) & (betas >= 0) & (betas <= 1)
        alphas = torch.where(valid, alphas, torch.full_like(alphas, 1e6))
        return alphas.flatten(2).min(dim=-1).values


class Sphere(NamedTuple):
    """B x K spheres: center (B,K,3), radius (B,K)."""

    center: Tensor
    radius: Tensor

    @staticmethod
    def create(center: Tensor, radius: Tensor) -> "Sphere":
        return Sphere(center, radius)

    @property
    def n_obs(self) -> int:
        return self.center.shape[1]

    def inside(self, point: Tensor, r: float = 0.0) -> Tensor:
        if self.n_obs == 0:
            return torch.zeros(point.shape[:-1], dtype=torch.bool, device=point.device)
        d = torch.linalg.vector_norm(point[:, :, None, :] - self.center[:, None, :, :], dim=-1)
        return (d < self.radius[:, None, :] + r).any(dim=-1)

    def raytrace(self, starts: Tensor, ends: Tensor) -> Tensor:
        """Quadratic ray-sphere intersection (reference obstacle.py:237-270)."""
        if self.n_obs == 0:
            return torch.full(starts.shape[:-1], 1e6, device=starts.device)
        o = starts[:, :, None, :] - self.center[:, None, :, :]  # (B,M,K,3)
        d = (ends - starts)[:, :, None, :]
        a = (d * d).sum(-1)
        b = 2 * (o * d).sum(-1)
        c = (o * o).sum(-1) - (self.radius[:, None, :]) ** 2
        disc = b * b - 4 * a * c
        ok = disc >= 0
        sq = torch.sqrt(torch.clamp(disc, min=0.0))
        a1 = (-b - sq) / (2 * a)
        a2 = (-b + sq) / (2 * a)
        # smallest non-negative root in [0,1]
        a1v = torch.where((a1 >= 0) & (a1 <= 1) & ok, a1, torch.full_like(a1, 1e6))
        a2v = torch.where((a2 >= 0) & (a2 <= 1) & ok, a2, torch.full_like(a2, 1e6))
        return torch.minimum(a1v, a2v).min(dim=-1).values


class Cuboid(NamedTuple):
    """B x K axis-angle cuboids: center (B,K,3), half (B,K,3) half-extents,
    rot (B,K,3,3) rotation matrices (body->world).

    The reference (obstacle.py:99-222) stores quaternion + 6 faces; we keep a
    rotation matrix and do slab-method ray intersection (equivalent geometry).
    """

    center: Tensor
    half: Tensor
    rot: Tensor

    @staticmethod
    def create_axis_aligned(center: Tensor, lengths: Tensor) -> "Cuboid":
        B, K = center.shape[:2]
        rot = torch.eye(3, device=center.device).expand(B, K, 3, 3).contiguous()
        return Cuboid(center, lengths / 2, rot)

    @property
    def n_obs(self) -> int:
        return self.center.shape[1]

    def _to_body(self, point: Tensor) -> Tensor:
        rel = point[:, :, None, :] - self.center[:, None, :, :]  # (B,M,K,3)
        return torch.einsum("bkij,bmki->bmkj", self.rot, rel)

    def inside(self, point: Tensor, r: float = 0.0) -> Tensor:
        if self.n_obs == 0:
            return torch.zeros(point.shape[:-1], dtype=torch.bool, device=point.device)
        q = self._to_body(point).abs() - self.half[:, None, :, :]
        # rounded-box SDF <= r
        outside = torch.linalg.vector_norm(torch.clamp(q, min=0.0), dim=-1)
        inside_d = torch.clamp(q.max(dim=-1).values, max=0.0)
        return (outside + inside_d < r) if r > 0 else ((outside + inside_d) < 0)
        # note: for r == 0 this is the plain box test

    def raytrace(self, starts: Tensor, ends: Tensor) -> Tensor:
        if self.n_obs == 0:
            return torch.full(starts.shape[:-1], 1e6, device=starts.device)
        o = self._to_body(starts)  # (B,M,K,3)
        e = self._to_body(ends)
        d = e - o
        d = torch.where(d.abs() < 1e-9, torch.full_like(d, 1e-9), d)
        h = self.half[:, None, :, :]
        t1 = (-h - o) / d
        t2 = (h - o) / d
        tmin = torch.minimum(t1, t2).max(dim=-1).values
        tmax = torch.maximum(t1, t2).min(dim=-1).values
        hit = (tmax >= tmin) & (tmax >= 0) & (tmin <= 1)
        alpha = torch.where(tmin >= 0, tmin, torch.zeros_like(tmin))
        alpha = torch.where(hit, alpha, torch.full_like(alpha, 1e6))
        return alpha.min(dim=-1).values
