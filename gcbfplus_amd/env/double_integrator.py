"""DoubleIntegrator env — the benchmark environment (BASELINE config #2/#4).

Semantics mirror ``/root/reference/gcbfplus/env/double_integrator.py`` exactly
(cited per method); the implementation is batched torch with the hot paths
(LiDAR raytrace, graph build, step) routed to CDNA4 HIP kernels on GPU.

State: (x, y, vx, vy); action: (fx, fy) forces, mass m = 0.1; 2D rectangles
as obstacles, 32-ray LiDAR.
"""
from __future__ import annotations

import os
import math
from typing import NamedTuple, Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from ..utils.graph import GraphBatch
from .base import MultiAgentEnv, StepResult
from .obstacle import Rectangle
from .utils import get_lidar, lqr, sample_starts_goals


class DoubleIntegrator(MultiAgentEnv):
    AGENT, GOAL, OBS = 0, 1, 2

    PARAMS = {
        "car_radius": 0.05,
        "comm_radius": 0.5,
        "n_rays": 32,
        "obs_len_range": [0.1, 0.5],
        "n_obs": 8,
        "m": 0.1,
    }

    def __init__(self, num_agents, area_size, max_step=256, max_travel=None, dt=0.03,
                 params=None, device=None):
        super().__init__(num_agents, area_size, max_step, max_travel, dt, params, device)
        # discrete dynamics x+ = A x + B u (reference :54-64)
        A = np.eye(4, dtype=np.float64)
        A[0, 2] = A[1, 3] = self._dt
        Bm = np.zeros((4, 2), dtype=np.float64)
        Bm[2, 0] = Bm[3, 1] = self._dt / self._params["m"]
        Q = np.eye(4) * 5.0
        R = np.eye(2)
        self._K_np = lqr(A, Bm, Q, R).astype(np.float32)
        self._K = torch.from_numpy(self._K_np).to(self.device)

    # ---- dims ------------------------------------------------------------
    @property
    def state_dim(self) -> int:
        return 4

    @property
    def node_dim(self) -> int:
        return 3

    @property
    def edge_dim(self) -> int:
        return 4

    @property
    def action_dim(self) -> int:
        return 2

    # ---- limits (reference :322-330) -------------------------------------
    def state_lim(self, state=None) -> Tuple[Tensor, Tensor]:
        lo = torch.tensor([-math.inf, -math.inf, -0.5, -0.5])
        hi = torch.tensor([math.inf, math.inf, 0.5, 0.5])
        return lo, hi

    def action_lim(self) -> Tuple[Tensor, Tensor]:
        return -torch.ones(2), torch.ones(2)

    # ---- reset (reference :87-110) ----------------------------------------
    def sample_obstacles(self, batch: int, rng: np.random.Generator) -> Rectangle:
        k = self._params["n_obs"]
        lo, hi = self._params["obs_len_range"]
        pos = rng.uniform(0, self.area_size, size=(batch, k, 2)).astype(np.float32)
        length = rng.uniform(lo, hi, size=(batch, k, 2)).astype(np.float32)
        theta = rng.uniform(0, 2 * math.pi, size=(batch, k)).astype(np.float32)
        return Rectangle.create(
            torch.from_numpy(pos), torch.from_numpy(length[..., 0]),
            torch.from_numpy(length[..., 1]), torch.from_numpy(theta),
        )

    def reset(self, batch: int, rng: np.random.Generator) -> GraphBatch:
        obs_cpu = self.sample_obstacles(batch, rng)

        def inside_np(b, pts, r):
            p = torch.from_numpy(np.asarray(pts, dtype=np.float32))[None]
            one = Rectangle(*[t[b : b + 1] for t in obs_cpu])
            return one.inside(p, r)[0].numpy()

        starts, goals = sample_starts_goals(
            rng, batch, self.num_agents, 2, self.area_size, inside_np,
            min_dist=4 * self._params["car_radius"], max_travel=self.max_travel,
        )
        zeros = np.zeros_like(starts)
        agent = torch.from_numpy(np.concatenate([starts, zeros], axis=-1)).to(self.device)
        goal = torch.from_numpy(np.concatenate([goals, zeros], axis=-1)).to(self.device)
        obstacles = Rectangle(*[t.to(self.device) for t in obs_cpu])
        return self.get_graph(agent, goal, obstacles)

    # ---- dynamics (reference :112-148, 266-273) ---------------------------
    def agent_xdot(self, agent_states: Tensor, action: Tensor) -> Tensor:
        accel = action / self._params["m"]
        return torch.cat([agent_states[..., 2:], accel], dim=-1)

    def agent_step_euler(self, agent_states: Tensor, action: Tensor) -> Tensor:
        x_dot = self.agent_xdot(agent_states, action)
        return self.clip_state(agent_states + x_dot * self._dt)

    def control_affine_dyn(self, state: Tensor) -> Tuple[Tensor, Tensor]:
        f = torch.cat([state[..., 2:], torch.zeros_like(state[..., :2])], dim=-1)
        g = torch.zeros(*state.shape[:-1], 4, 2, device=state.device)
        g[..., 2, 0] = 1.0 / self._params["m"]
        g[..., 3, 1] = 1.0 / self._params["m"]
        return f, g

    # ---- graph build (reference :288-320, 223-264) -------------------------
    def get_lidar_hits(self, agent_pos: Tensor, obstacles: Rectangle) -> Tensor:
        """(B, N, 2) -> hit points (B, N, R, 2). GPU: K1 HIP kernel."""
        if agent_pos.is_cuda:
            from .. import ops

            return ops.raytrace_rect(
                agent_pos.contiguous(), obstacles.points, self.n_rays,
                self._params["comm_radius"],
            )
        return get_lidar(agent_pos, obstacles, self.n_rays, self._params["comm_radius"])

    def build_mask(self, states: Tensor) -> Tensor:
        """Edge-slot mask (B, N, D) from node states (topology at graph-build
        time; reference edge_blocks :223-264)."""
        B = states.shape[0]
        n, r = self.num_agents, self.n_rays
        pdim = self.pos_dim
        agent_pos = states[:, :n, :pdim]
        comm = self._params["comm_radius"]
        dist = torch.cdist(agent_pos, agent_pos)
        eye = torch.eye(n, dtype=torch.bool, device=states.device)
        aa = (dist < comm) & ~eye
        goal = torch.ones(B, n, 1, dtype=torch.bool, device=states.device)
        hit_pos = states[:, 2 * n :, :pdim].reshape(B, n, r, pdim)
        hit_dist = torch.linalg.vector_norm(agent_pos[:, :, None] - hit_pos, dim=-1)
        lidar = hit_dist < comm - 1e-1
        return torch.cat([aa, goal, lidar], dim=-1)

    def get_graph(self, agent: Tensor, goal: Tensor, obstacles: Rectangle) -> GraphBatch:
        B = agent.shape[0]
        n, r = self.num_agents, self.n_rays
        hits = self.get_lidar_hits(agent[..., : self.pos_dim], obstacles)  # (B,N,R,pdim)
        pad = self.state_dim - self.pos_dim  # zero velocity components (:304-306)
        hit_states = torch.cat([hits, hits.new_zeros(*hits.shape[:-1], pad)], dim=-1)
        states = torch.cat([agent, goal, hit_states.reshape(B, n * r, self.state_dim)], dim=1)
        mask = self.build_mask(states)
        return GraphBatch(states=states, mask=mask, n_agents=n, n_rays=r, env_states=obstacles)

    fused_edge = True  # state-diff + pos-clip family: ops.edge_msg_in applies

    def edge_msg_in(self, graph: GraphBatch, states: Optional[Tensor] = None) -> Tensor:
        """Fused layer-0 GNN input (see ops.edge_msg_in)."""
        from .. import ops

        if states is None:
            states = graph.states
        return ops.edge_msg_in(states, self.num_agents, self.n_rays, self.pos_dim,
                               self._params["comm_radius"])

    def edge_feats(self, graph: GraphBatch, states: Optional[Tensor] = None) -> Tensor:
        """Dense (B, N, D, edge_dim) differentiable edge features.

        e[i, d] = x_agent_i - x_sender(d), with the position components
        clipped to comm_radius by norm — the reference applies this clip to
        goal edges at build time (:236-247) and to ALL edges in
        add_edge_feats (:275-286); the two agree because agent-agent and
        lidar edges are only ever active within comm radius at build time.
        """
        if states is None:
            states = graph.states
        B = states.shape[0]
        n, r = self.num_agents, self.n_rays
        S, pdim = self.state_dim, self.pos_dim
        recv = states[:, :n, None, :]  # (B,N,1,S)
        senders = torch.cat(
            [
                states[:, None, :n].expand(B, n, n, S),  # agent senders
                states[:, n : 2 * n, None, :],  # own goal
                states[:, 2 * n :].reshape(B, n, r, S),  # own lidar hits
            ],
            dim=2,
        )
        e = recv - senders  # (B,N,D,S)
        pos = e[..., :pdim]
        norm = torch.sqrt(1e-6 + (pos * pos).sum(-1, keepdim=True))
        comm = self._params["comm_radius"]
        coef = torch.where(norm > comm, comm / torch.clamp(norm, min=comm), torch.ones_like(norm))
        return torch.cat([pos * coef, e[..., pdim:]], dim=-1)

    def edge_grad_to_state_jac(self, graph: GraphBatch, states: Tensor, ge: Tensor) -> Tensor:
        """Chain dh/d(edge_feats) -> dh_i/dx_j (M, N, N, S) analytically.

        Valid for 'state-diff + position-clip' edge features (SI/DI/drone
        family). ge: (M, N, D, S) gradient of h_i w.r.t. edge slot (i, d)
        (one backward pass; exact for gnn_layers == 1 where h_i touches only
        receiver-i edges). The clip jacobian for active clip (norm > comm):
        d(comm * p / n)/dp = comm * (I/n - p p^T / n^3), n = sqrt(1e-6+|p|^2).
        """
        M, N, D, S = ge.shape
        pdim = self.pos_dim
        n, r = self.num_agents, self.n_rays
        recv = states[:, :n, None, :]
        senders = torch.cat(
            [
                states[:, None, :n].expand(M, n, n, S),
                states[:, n : 2 * n, None, :],
                states[:, 2 * n :].reshape(M, n, r, S),
            ],
            dim=2,
        )
        v = recv - senders
        p = v[..., :pdim]
        nrm = torch.sqrt(1e-6 + (p * p).sum(-1, keepdim=True))
        comm = self._params["comm_radius"]
        active = nrm > comm
        gp = ge[..., :pdim]
        gp_clip = comm * (gp / nrm - p * (gp * p).sum(-1, keepdim=True) / nrm.pow(3))
        gp_out = torch.where(active, gp_clip, gp)
        c = torch.cat([gp_out, ge[..., pdim:]], dim=-1)  # (M,N,D,S) cotangent wrt v
        h_x = -c[:, :, :N, :].clone()  # agent-sender slots
        diag = c.sum(dim=2)  # (M,N,S) receiver side over all slots
        idx = torch.arange(N, device=states.device)
        h_x[:, idx, idx, :] += diag
        return h_x

    # ---- step / forward (reference :145-181, 340-354) ----------------------
    def step(self, graph: GraphBatch, action: Tensor) -> StepResult:
        if not os.environ.get("GCBF_NO_FUSED_ENV") and graph.states.is_cuda and type(self) is DoubleIntegrator:
            return self._step_fused(graph, action)
        agent = graph.agent_states
        goal = graph.goal_states
        obstacles = graph.env_states
        action = self.clip_action(action)
        next_agent = self.agent_step_euler(agent, action)

        reward = -((action - self.u_ref(graph)).square().sum(-1)).mean(-1)  # (B,)
        cost = self.get_cost(graph)
        done = torch.zeros(agent.shape[0], dtype=torch.bool, device=agent.device)
        next_graph = self.get_graph(next_agent, goal, obstacles)
        return StepResult(next_graph, reward, cost, done, {})

    def _step_fused(self, graph: GraphBatch, action: Tensor) -> StepResult:
        """One-kernel env step on GPU (K5-K8, ops/hip/env_step.hip)."""
        from .. import ops

        ext = ops._require_ext()
        p = self._params
        nxt, mask, reward, cost = ext.di_env_step(
            graph.states.contiguous(), action.contiguous(),
            graph.env_states.points.contiguous(),
            self._K.to(graph.device).contiguous(), self.num_agents, self.n_rays,
            self._dt, 1.0 / p["m"], p["comm_radius"], p["car_radius"], 0.5,
        )
        done = torch.zeros(graph.batch_size, dtype=torch.bool, device=graph.device)
        g = GraphBatch(states=nxt, mask=mask, n_agents=self.num_agents,
                       n_rays=self.n_rays, env_states=graph.env_states)
        return StepResult(g, reward, cost, done, {})

    def forward_graph(self, graph: GraphBatch, action: Tensor) -> GraphBatch:
        action = self.clip_action(action)
        next_agent = self.agent_step_euler(graph.agent_states, action)
        return graph.with_agent_states(next_agent)

    def loss_prep(self, graph: GraphBatch, raw: Tensor):
        """Fused GCBF+ loss prologue (K16): u_ref -> action = clamp(2*raw +
        u_ref) -> euler next state -> [states; next_states]. Only for the
        exact class (subclasses override dynamics/u_ref)."""
        if type(self) is not DoubleIntegrator or not graph.states.is_cuda:
            return None
        from .. import ops

        if not ops.hip_available():
            return None
        p = self._params
        return ops.di_loss_prep(
            graph.states, raw, self._K.to(graph.device), self.num_agents,
            self._dt, 1.0 / p["m"], p["comm_radius"], 0.5,
        )

    # ---- cost / reward (reference :183-198) --------------------------------
    def get_cost(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., : self.pos_dim]
        n = self.num_agents
        r = self._params["car_radius"]
        dist = torch.cdist(pos, pos)
        dist = dist + torch.eye(n, device=pos.device) * 1e6
        collision = (dist < 2 * r).any(dim=-1).float().mean(-1)
        inside = graph.env_states.inside(pos, r=r).float().mean(-1)
        return collision + inside

    # ---- u_ref (reference :332-338) ----------------------------------------
    def u_ref(self, graph: GraphBatch) -> Tensor:
        error = graph.goal_states - graph.agent_states  # (B,N,4)
        norm = torch.linalg.vector_norm(error, dim=-1, keepdim=True).clamp_min(1e-9)
        error_max = (error / norm * self._params["comm_radius"]).abs()
        error = torch.clamp(error, -error_max, error_max)
        K = self._K.to(error.device)
        return self.clip_action(error @ K.t())

    # ---- safety masks (reference :356-440) ---------------------------------
    def safe_mask(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., :2]
        n = self.num_agents
        r = self._params["car_radius"]
        dist = torch.cdist(pos, pos) + torch.eye(n, device=pos.device) * (2 * r + 1)
        safe_agent = (dist > 4 * r).all(dim=-1)
        safe_obs = ~graph.env_states.inside(pos, r=2 * r)
        return safe_agent & safe_obs

    def unsafe_mask(self, graph: GraphBatch) -> Tensor:
        """Collision OR heading into the 'unsafe direction' cone
        (reference :376-417). The reference checks each agent against all
        N*R lidar hits with a block-diagonal validity mask; since hits are
        stored agent-major, that reduces to each agent vs its OWN R hits."""
        st = graph.agent_states
        pos = st[..., :2]
        vel = st[..., 2:]
        B = pos.shape[0]
        n, R = self.num_agents, self.n_rays
        r = self._params["car_radius"]

        apd = pos[:, None, :, :] - pos[:, :, None, :]  # [b,i,j] = pos_j - pos_i
        adist = torch.linalg.vector_norm(apd, dim=-1) + torch.eye(n, device=pos.device) * (
            2 * r + 1
        )
        unsafe_agent = (adist < 2 * r).any(dim=-1)
        unsafe_obs = graph.env_states.inside(pos, r=r)
        collision = unsafe_agent | unsafe_obs

        hit_pos = graph.hit_states[..., :2]  # (B,N,R,2) own hits
        opd = hit_pos - pos[:, :, None, :]
        odist = torch.linalg.vector_norm(opd, dim=-1)  # (B,N,R)

        pos_diff = torch.cat([apd, opd], dim=2)  # (B,N,n+R,2)
        warn = torch.cat([adist < 3 * r, odist < 2 * r], dim=2)
        pvec = pos_diff / (torch.linalg.vector_norm(pos_diff, dim=-1, keepdim=True) + 1e-4)
        speed = torch.linalg.vector_norm(vel, dim=-1, keepdim=True)
        hvec = (vel / (speed + 1e-4))[:, :, None, :]
        inner = (pvec * hvec).sum(-1)  # (B,N,n+R)
        th_agent = torch.atan2(
            torch.full_like(adist, 2 * r), torch.sqrt(adist**2 - 4 * r**2)
        )
        th_obs = torch.atan2(torch.full_like(odist, r), torch.sqrt(odist**2 - r**2))
        th = torch.cat([th_agent, th_obs], dim=2)
        # NaN theta (dist inside radius) -> cos NaN -> comparison False, as in jax
        unsafe_dir = (warn & (inner > torch.cos(th))).any(dim=-1)
        return collision | unsafe_dir

    def collision_mask(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., :2]
        n = self.num_agents
        r = self._params["car_radius"]
        dist = torch.cdist(pos, pos) + torch.eye(n, device=pos.device) * (2 * r + 1)
        unsafe_agent = (dist < 2 * r).any(dim=-1)
        unsafe_obs = graph.env_states.inside(pos, r=r)
        return unsafe_agent | unsafe_obs

    def finish_mask(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., :2]
        goal = graph.goal_states[..., :2]
        return torch.linalg.vector_norm(pos - goal, dim=-1) < 2 * self._params["car_radius"]
