"""DubinsCar env: 2D nonholonomic car (reference ``gcbfplus/env/dubins_car.py``).

State (x, y, theta, v); action (omega, a) with omega scaled x20 in the
dynamics (:118) and x10 in the control-affine form (:251); edge features use
the velocity-vector transform [x, y, v cos(theta), v sin(theta)] (:263-275);
agents freeze ("stop") within 0.5 r of their goal (:483-487).
BASELINE config #3.
"""
from __future__ import annotations

import os
import math
from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from ..utils.graph import GraphBatch
from .base import StepResult
from .double_integrator import DoubleIntegrator
from .obstacle import Rectangle
from .utils import sample_starts_goals


class DubinsCar(DoubleIntegrator):
    PARAMS = {
        "car_radius": 0.05,
        "comm_radius": 0.5,
        "n_rays": 16,
        "obs_len_range": [0.1, 0.6],
        "n_obs": 8,
    }

    fused_edge = True  # via the mode-1 transform in ops.edge_msg_in

    def edge_msg_in(self, graph, states=None):
        from .. import ops

        if states is None:
            states = graph.states
        return ops.edge_msg_in(states, self.num_agents, self.n_rays, 2,
                               self._params["comm_radius"], mode=1)

    def __init__(self, num_agents, area_size, max_step=256, max_travel=None, dt=0.03,
                 params=None, device=None):
        super(DoubleIntegrator, self).__init__(num_agents, area_size, max_step, max_travel,
                                               dt, params, device)
        self.enable_stop = True

    @property
    def state_dim(self) -> int:
        return 4  # x, y, theta, v

    @property
    def edge_dim(self) -> int:
        return 4  # x_rel, y_rel, vx_rel, vy_rel

    @property
    def action_dim(self) -> int:
        return 2  # omega, acc

    def state_lim(self, state=None) -> Tuple[Tensor, Tensor]:
        inf = math.inf
        return (torch.tensor([-inf, -inf, -inf, -0.8]), torch.tensor([inf, inf, inf, 0.8]))

    def action_lim(self) -> Tuple[Tensor, Tensor]:
        return -3.0 * torch.ones(2), 3.0 * torch.ones(2)

    # ---- reset (reference :72-102): random heading, goal heading toward goal
    def reset(self, batch: int, rng: np.random.Generator) -> GraphBatch:
        obs_cpu = self.sample_obstacles(batch, rng)

        def inside_np(b, pts, r):
            p = torch.from_numpy(np.asarray(pts, dtype=np.float32))[None]
            return Rectangle(*[t[b : b + 1] for t in obs_cpu]).inside(p, r)[0].numpy()

        starts, goals = sample_starts_goals(
            rng, batch, self.num_agents, 2, self.area_size, inside_np,
            min_dist=4 * self._params["car_radius"], max_travel=self.max_travel,
        )
        theta0 = rng.uniform(-math.pi, math.pi, size=(batch, self.num_agents, 1)).astype(
            np.float32
        )
        goal_theta = np.arctan2(goals[..., 1] - starts[..., 1],
                                goals[..., 0] - starts[..., 0])[..., None].astype(np.float32)
        zeros = np.zeros((batch, self.num_agents, 1), dtype=np.float32)
        agent = torch.from_numpy(np.concatenate([starts, theta0, zeros], -1)).to(self.device)
        goal = torch.from_numpy(np.concatenate([goals, goal_theta, zeros], -1)).to(self.device)
        obstacles = Rectangle(*[t.to(self.device) for t in obs_cpu])
        return self.get_graph(agent, goal, obstacles)

    # ---- dynamics (reference :104-122) -------------------------------------
    def agent_xdot(self, agent_states: Tensor, action: Tensor) -> Tensor:
        th, v = agent_states[..., 2], agent_states[..., 3]
        return torch.stack(
            [torch.cos(th) * v, torch.sin(th) * v, action[..., 0] * 20.0, action[..., 1]],
            dim=-1,
        )

    def stop_mask(self, graph: GraphBatch) -> Tensor:
        pos = graph.agent_states[..., :2]
        goal = graph.goal_states[..., :2]
        return torch.linalg.vector_norm(pos - goal, dim=-1) < self._params["car_radius"] * 0.5

    def _step_states(self, graph: GraphBatch, action: Tensor) -> Tensor:
        action = self.clip_action(action)
        stop = self.stop_mask(graph).float() if self.enable_stop \
            else torch.zeros_like(graph.agent_states[..., 0])
        x_dot = self.agent_xdot(graph.agent_states, action) * (1 - stop)[..., None]
        return self.clip_state(graph.agent_states + x_dot * self._dt)

    def step(self, graph: GraphBatch, action: Tensor) -> StepResult:
        if not os.environ.get("GCBF_NO_FUSED_ENV") and graph.states.is_cuda and type(self) is DubinsCar and self.enable_stop:
            return self._step_fused(graph, action)
        next_agent = self._step_states(graph, action)
        reward = -((self.clip_action(action) - self.u_ref(graph)).square().sum(-1)).mean(-1)
        cost = self.get_cost(graph)
        done = torch.zeros(graph.batch_size, dtype=torch.bool, device=graph.device)
        next_graph = self.get_graph(next_agent, graph.goal_states, graph.env_states)
        return StepResult(next_graph, reward, cost, done, {})

    def _step_fused(self, graph: GraphBatch, action: Tensor) -> StepResult:
        """One-kernel env step on GPU (K5-K8, env_step2d_kernel<1>):
        PID u_ref + stop freeze + euler + LiDAR rescan + masks."""
        from .. import ops

        ext = ops._require_ext()
        p = self._params
        if not hasattr(self, "_K_dummy"):
            self._K_dummy = torch.zeros(2, 4, device=graph.device)
        nxt, mask, reward, cost = ext.di_env_step(
            graph.states.contiguous(), action.contiguous(),
            graph.env_states.points.contiguous(), self._K_dummy,
            self.num_agents, self.n_rays, self._dt, 0.0, p["comm_radius"],
            p["car_radius"], 0.8, 1,
        )
        done = torch.zeros(graph.batch_size, dtype=torch.bool, device=graph.device)
        g = GraphBatch(states=nxt, mask=mask, n_agents=self.num_agents,
                       n_rays=self.n_rays, env_states=graph.env_states)
        return StepResult(g, reward, cost, done, {})

    def forward_graph(self, graph: GraphBatch, action: Tensor) -> GraphBatch:
        return graph.with_agent_states(self._step_states(graph, action))

    def control_affine_dyn(self, state: Tensor) -> Tuple[Tensor, Tensor]:
        """(:249-259): f = [v cos, v sin, 0, 0]; g = [[0,0],[0,0],[10,0],[0,1]]."""
        th, v = state[..., 2], state[..., 3]
        zero = torch.zeros_like(th)
        f = torch.stack([torch.cos(th) * v, torch.sin(th) * v, zero, zero], dim=-1)
        g = torch.zeros(*state.shape[:-1], 4, 2, device=state.device)
        g[..., 2, 0] = 10.0
        g[..., 3, 1] = 1.0
        return f, g

    # ---- edge features (reference :203-245, 261-275) ------------------------
    def _edge_states(self, states: Tensor) -> Tensor:
        """[x, y, v cos(theta), v sin(theta)] — goals/lidar have v = 0 so
        this is the identity transform for them."""
        th, v = states[..., 2], states[..., 3]
        return torch.stack(
            [states[..., 0], states[..., 1], v * torch.cos(th), v * torch.sin(th)], dim=-1
        )

    def edge_feats(self, graph: GraphBatch, states: Optional[Tensor] = None) -> Tensor:
        if states is None:
            states = graph.states
        es = self._edge_states(states)
        B, n, r = es.shape[0], self.num_agents, self.n_rays
        S = 4
        recv = es[:, :n, None, :]
        senders = torch.cat(
            [
                es[:, None, :n].expand(B, n, n, S),
                es[:, n : 2 * n, None, :],
                es[:, 2 * n :].reshape(B, n, r, S),
            ],
            dim=2,
        )
        e = recv - senders
        pos = e[..., :2]
        norm = torch.sqrt(1e-6 + (pos * pos).sum(-1, keepdim=True))
        comm = self._params["comm_radius"]
        coef = torch.where(norm > comm, comm / torch.clamp(norm, min=comm), torch.ones_like(norm))
        return torch.cat([pos * coef, e[..., 2:]], dim=-1)

    def edge_grad_to_state_jac(self, graph: GraphBatch, states: Tensor, ge: Tensor) -> Tensor:
        """Chain through the edge-state transform: d(es)/d(state) =
        [[I2, 0], [0, J_v]] with J_v = d[v cos, v sin]/d(theta, v)."""
        es = self._edge_states(states)
        # first chain dh/d(edge feats) -> dh/d(es) using the DI machinery on es
        h_es = DoubleIntegrator.edge_grad_to_state_jac(self, graph, es, ge)  # (M,N,N,4)
        th, v = states[:, : self.num_agents, 2], states[:, : self.num_agents, 3]
        ct, st = torch.cos(th), torch.sin(th)
        # d es_j / d x_j: [[1,0,0,0],[0,1,0,0],[0,0,-v sin, cos],[0,0,v cos, sin]]
        out = torch.zeros_like(h_es)
        out[..., 0] = h_es[..., 0]
        out[..., 1] = h_es[..., 1]
        # theta and v columns pick up from the velocity-vector rows (j = sender/receiver agent)
        vj = v[:, None, :]
        ctj, stj = ct[:, None, :], st[:, None, :]
        out[..., 2] = h_es[..., 2] * (-vj * stj) + h_es[..., 3] * (vj * ctj)
        out[..., 3] = h_es[..., 2] * ctj + h_es[..., 3] * stj
        return out

    # ---- u_ref (reference :328-379): PID heading + speed controller --------
    def u_ref(self, graph: GraphBatch) -> Tensor:
        ag = graph.agent_states
        goal = graph.goal_states
        pos_diff = ag[..., :2] - goal[..., :2]
        k_omega, k_v, k_a = 1.0, 2.3, 2.5
        two_pi = 2 * math.pi
        dist = torch.linalg.vector_norm(pos_diff, dim=-1)
        theta_t = torch.atan2(-pos_diff[..., 1], -pos_diff[..., 0]) % two_pi
        theta = ag[..., 2] % two_pi
        theta_diff = theta_t - theta
        adir = torch.stack([torch.cos(theta), torch.sin(theta)], dim=-1)
        inner = (-pos_diff * adir).sum(-1) / (dist + 1e-4)
        theta_between = torch.acos(torch.clamp(inner, -1.0, 1.0))
        fwd = (theta_diff < math.pi) & (theta_diff >= 0)
        bwd = (theta_diff > -math.pi) & (theta_diff <= 0)
        le_pi = theta <= math.pi
        omega = torch.where(
            le_pi,
            torch.where(fwd, k_omega * theta_between, -k_omega * theta_between),
            torch.where(bwd, -k_omega * theta_between, k_omega * theta_between),
        )
        omega = torch.clamp(omega, -5.0, 5.0)
        norm = torch.sqrt(1e-6 + (pos_diff * pos_diff).sum(-1, keepdim=True))
        comm = self._params["comm_radius"]
        coef = torch.where(norm > comm, comm / norm, torch.ones_like(norm))
        pd = coef * pos_diff
        a = -k_a * ag[..., 3] + k_v * torch.linalg.vector_norm(pd, dim=-1)
        return torch.stack([omega, a], dim=-1)

    # ---- masks (reference :399-487) -----------------------------------------
    def unsafe_mask(self, graph: GraphBatch) -> Tensor:
        """Like DI but heading from theta and obstacle margin 1.5r (:421)."""
        st = graph.agent_states
        pos = st[..., :2]
        n, R = self.num_agents, self.n_rays
        r = self._params["car_radius"]
        apd = pos[:, None, :, :] - pos[:, :, None, :]
        adist = torch.linalg.vector_norm(apd, dim=-1) + torch.eye(n, device=pos.device) * (
            2 * r + 1
        )
        unsafe_agent = (adist < 2 * r).any(dim=-1)
        unsafe_obs = graph.env_states.inside(pos, r=1.5 * r)
        collision = unsafe_agent | unsafe_obs

        hit_pos = graph.hit_states[..., :2]
        opd = hit_pos - pos[:, :, None, :]
        odist = torch.linalg.vector_norm(opd, dim=-1)
        pos_diff = torch.cat([apd, opd], dim=2)
        warn = torch.cat([adist < 3 * r, odist < 2 * r], dim=2)
        pvec = pos_diff / (torch.linalg.vector_norm(pos_diff, dim=-1, keepdim=True) + 1e-4)
        hvec = torch.stack([torch.cos(st[..., 2]), torch.sin(st[..., 2])], dim=-1)[:, :, None, :]
        inner = (pvec * hvec).sum(-1)
        th_agent = torch.atan2(torch.full_like(adist, 2 * r),
                               torch.sqrt(adist**2 - 4 * r**2))
        th_obs = torch.atan2(torch.full_like(odist, r), torch.sqrt(odist**2 - r**2))
        th = torch.cat([th_agent, th_obs], dim=2)
        unsafe_dir = (warn & (inner > torch.cos(th))).any(dim=-1)
        return collision | unsafe_dir
