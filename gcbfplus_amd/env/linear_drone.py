"""LinearDrone env: 3D damped double integrator with spherical obstacles
(reference ``gcbfplus/env/linear_drone.py``).

State (x,y,z,vx,vy,vz); action (ax,ay,az); xdot = A x + B u with damping
diag(-1.1,-1.1,-6) on velocities and B = 10 I (:55-67). LiDAR: 3D
theta x phi fan of params['n_rays'](=32) beams + poles, keeping the top-16
closest hits (:73, 290-299). Faithful quirk: the LQR gain mixes the
discretized A with the continuous B (:63-72).
"""
from __future__ import annotations

import os
import math
from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from .. import ops
from ..utils.graph import GraphBatch
from .double_integrator import DoubleIntegrator
from .obstacle import Sphere
from .utils import get_lidar, lqr, sample_starts_goals


class LinearDrone(DoubleIntegrator):
    PARAMS = {
        "drone_radius": 0.05,
        "comm_radius": 0.5,
        "n_rays": 32,
        "obs_len_range": [0.15, 0.3],
        "n_obs": 4,
    }

    N_HIT_RETURNS = 16  # top-k closest hits kept as graph nodes (:73)

    def __init__(self, num_agents, area_size, max_step=256, max_travel=None, dt=0.03,
                 params=None, device=None):
        super(DoubleIntegrator, self).__init__(num_agents, area_size, max_step, max_travel,
                                               dt, params, device)
        # radius key differs from the car envs; alias for the shared helpers
        self._params.setdefault("car_radius", self._params["drone_radius"])
        import scipy.linalg

        A = np.zeros((6, 6))
        A[0, 3] = A[1, 4] = 1.0
        A[2, 5] = 1.0
        A[3, 3] = A[4, 4] = -1.1
        A[5, 5] = -6.0
        self._A_np = A.astype(np.float32)
        A_discrete = scipy.linalg.expm(A * self._dt)
        B = np.zeros((6, 3))
        B[3, 0] = B[4, 1] = B[5, 2] = 10.0
        Q = np.diag([5e1, 5e1, 5e1, 1.0, 1.0, 1.0])
        self._K_np = lqr(A_discrete, B, Q, np.eye(3)).astype(np.float32)
        self._K = torch.from_numpy(self._K_np).to(self.device)
        self._B_np = B.astype(np.float32)
        self._A_t = torch.from_numpy(self._A_np).to(self.device)
        self._B_t = torch.from_numpy(self._B_np).to(self.device)

    # ---- dims -------------------------------------------------------------
    @property
    def state_dim(self) -> int:
        return 6

    @property
    def edge_dim(self) -> int:
        return 6

    @property
    def action_dim(self) -> int:
        return 3

    @property
    def pos_dim(self) -> int:
        return 3

    @property
    def n_rays(self) -> int:
        return self.N_HIT_RETURNS  # graph layout: 16 hit nodes per agent

    fused_edge = True  # state-diff + pos-clip family (pos_dim = 3)

    def state_lim(self, state=None) -> Tuple[Tensor, Tensor]:
        inf = math.inf
        return (torch.tensor([-inf, -inf, -inf, -0.5, -0.5, -0.5]),
                torch.tensor([inf, inf, inf, 0.5, 0.5, 0.5]))

    def action_lim(self) -> Tuple[Tensor, Tensor]:
        return -torch.ones(3), torch.ones(3)

    # ---- reset (reference :92-116) ------------------------------------------
    def sample_obstacles(self, batch: int, rng: np.random.Generator) -> Sphere:
        k = self._params["n_obs"]
        lo, hi = self._params["obs_len_range"]
        pos = rng.uniform(0, self.area_size, size=(batch, k, 3)).astype(np.float32)
        radius = rng.uniform(lo / 2, hi / 2, size=(batch, k)).astype(np.float32)
        return Sphere.create(torch.from_numpy(pos), torch.from_numpy(radius))

    def reset(self, batch: int, rng: np.random.Generator) -> GraphBatch:
        obs_cpu = self.sample_obstacles(batch, rng)

        def inside_np(b, pts, r):
            p = torch.from_numpy(np.asarray(pts, dtype=np.float32))[None]
            return Sphere(*[t[b : b + 1] for t in obs_cpu]).inside(p, r)[0].numpy()

        starts, goals = sample_starts_goals(
            rng, batch, self.num_agents, 3, self.area_size, inside_np,
            min_dist=4 * self._params["drone_radius"], max_travel=self.max_travel,
        )
        zeros = np.zeros((batch, self.num_agents, self.state_dim - 3), dtype=np.float32)
        agent = torch.from_numpy(np.concatenate([starts, zeros], -1)).to(self.device)
        goal = torch.from_numpy(np.concatenate([goals, zeros], -1)).to(self.device)
        obstacles = Sphere(*[t.to(self.device) for t in obs_cpu])
        return self.get_graph(agent, goal, obstacles)

    # ---- dynamics (reference :120-134, 238-243) -----------------------------
    def agent_xdot(self, agent_states: Tensor, action: Tensor) -> Tensor:
        return agent_states @ self._A_t.to(agent_states.device).t() + \
            action @ self._B_t.to(agent_states.device).t()

    def control_affine_dyn(self, state: Tensor) -> Tuple[Tensor, Tensor]:
        f = state @ self._A_t.to(state.device).t()
        g = self._B_t.to(state.device).expand(*state.shape[:-1], 6, 3)
        return f, g

    # ---- lidar: 3D fan with top-16 selection (reference :290-299) -----------
    def get_lidar_hits(self, agent_pos: Tensor, obstacles: Sphere) -> Tensor:
        if agent_pos.is_cuda and ops.hip_available() and obstacles.n_obs > 0:
            return ops.raytrace_sphere_topk(
                agent_pos, obstacles.center, obstacles.radius,
                self._params["n_rays"], self.N_HIT_RETURNS,
                self._params["comm_radius"])
        return get_lidar(agent_pos, obstacles, self._params["n_rays"],
                         self._params["comm_radius"], max_returns=self.N_HIT_RETURNS)

    def get_cost(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., :3]
        n = self.num_agents
        r = self._params["drone_radius"]
        dist = torch.cdist(pos, pos) + torch.eye(n, device=pos.device) * 1e6
        collision = (dist < 2 * r).any(dim=-1).float().mean(-1)
        inside = graph.env_states.inside(pos, r=r).float().mean(-1)
        return collision + inside

    # ---- fused GPU step: part A (drone3d_step kernel: u_ref/reward/euler/
    # cost/aa+goal mask) + part B (raytrace_sphere_topk in graph mode: hit
    # node rows + lidar mask), replacing ~40 eager launches per step --------
    def step(self, graph: GraphBatch, action: Tensor):
        if not os.environ.get("GCBF_NO_FUSED_ENV") and graph.states.is_cuda and type(self) is LinearDrone and ops.hip_available() \
                and graph.env_states.n_obs > 0:
            return self._step_fused(graph, action)
        return super().step(graph, action)

    def _step_fused(self, graph: GraphBatch, action: Tensor):
        from .base import StepResult

        ext = ops._require_ext()
        p = self._params
        obs = graph.env_states
        nxt, mask, reward, cost = ext.drone3d_step(
            graph.states.contiguous(), action.contiguous(),
            obs.center.contiguous(), obs.radius.contiguous(),
            self._K.to(graph.device).contiguous(),
            self._A_t.to(graph.device).contiguous(),
            self.num_agents, self.n_rays, self._dt, 10.0, p["comm_radius"],
            p["drone_radius"], 0.5,
        )
        # part B: lidar rows + lidar mask from the NEXT agent positions
        next_pos = nxt[:, : self.num_agents, :3].contiguous()
        ext.raytrace_sphere_graph(next_pos, obs.center.contiguous(),
                                  obs.radius.contiguous(), nxt, mask,
                                  p["n_rays"], self.N_HIT_RETURNS,
                                  p["comm_radius"])
        done = torch.zeros(graph.batch_size, dtype=torch.bool, device=graph.device)
        g = GraphBatch(states=nxt, mask=mask, n_agents=self.num_agents,
                       n_rays=self.n_rays, env_states=obs)
        return StepResult(g, reward, cost, done, {})

    # ---- masks (reference :346-404): velocity-free margins ------------------
    def safe_mask(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., :3]
        n = self.num_agents
        r = self._params["drone_radius"]
        dist = torch.cdist(pos, pos) + torch.eye(n, device=pos.device) * (2 * r + 1)
        safe_agent = (dist > 4 * r).all(dim=-1)
        safe_obs = ~graph.env_states.inside(pos, r=2 * r)
        return safe_agent & safe_obs

    def unsafe_mask(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., :3]
        n = self.num_agents
        r = self._params["drone_radius"]
        dist = torch.cdist(pos, pos) + torch.eye(n, device=pos.device) * (2 * r + 1)
        unsafe_agent = (dist < 2.5 * r).any(dim=-1)
        unsafe_obs = graph.env_states.inside(pos, r=1.5 * r)
        return unsafe_agent | unsafe_obs

    def collision_mask(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., :3]
        n = self.num_agents
        r = self._params["drone_radius"]
        dist = torch.cdist(pos, pos) + torch.eye(n, device=pos.device) * (2 * r + 1)
        unsafe_agent = (dist < 2 * r).any(dim=-1)
        unsafe_obs = graph.env_states.inside(pos, r=r)
        return unsafe_agent | unsafe_obs

    def finish_mask(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., :3]
        goal = graph.goal_states[..., :3]
        return torch.linalg.vector_norm(pos - goal, dim=-1) < 2 * self._params["drone_radius"]
