"""SingleIntegrator env: 2D point mass, xdot = u (reference
``gcbfplus/env/single_integrator.py``). State (x, y); action (vx, vy);
edge_dim 2. BASELINE config #1 (CPU plumbing)."""
from __future__ import annotations

import math
from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from ..utils.graph import GraphBatch
from .base import StepResult
from .double_integrator import DoubleIntegrator
from .utils import lqr


class SingleIntegrator(DoubleIntegrator):
    PARAMS = {
        "car_radius": 0.05,
        "comm_radius": 0.5,
        "n_rays": 32,
        "obs_len_range": [0.1, 0.5],
        "n_obs": 8,
    }

    def __init__(self, num_agents, area_size, max_step=256, max_travel=None, dt=0.03,
                 params=None, device=None):
        # skip DoubleIntegrator.__init__ LQR setup; build our own
        super(DoubleIntegrator, self).__init__(num_agents, area_size, max_step, max_travel,
                                               dt, params, device)
        # x+ = x + u dt ; LQR on (A=I, B=dt*I) (reference single_integrator.py:44-52)
        A = np.eye(2)
        B = np.eye(2) * self._dt
        self._K_np = lqr(A, B, np.eye(2) * 5.0, np.eye(2)).astype(np.float32)
        self._K = torch.from_numpy(self._K_np).to(self.device)

    @property
    def state_dim(self) -> int:
        return 2

    @property
    def edge_dim(self) -> int:
        return 2

    @property
    def action_dim(self) -> int:
        return 2

    def state_lim(self, state=None) -> Tuple[Tensor, Tensor]:
        inf = math.inf
        return torch.tensor([-inf, -inf]), torch.tensor([inf, inf])

    def action_lim(self) -> Tuple[Tensor, Tensor]:
        return -torch.ones(2), torch.ones(2)

    def reset(self, batch: int, rng: np.random.Generator) -> GraphBatch:
        from .obstacle import Rectangle
        obs_cpu = self.sample_obstacles(batch, rng)

        def inside_np(b, pts, r):
            p = torch.from_numpy(np.asarray(pts, dtype=np.float32))[None]
            return Rectangle(*[t[b : b + 1] for t in obs_cpu]).inside(p, r)[0].numpy()

        from .utils import sample_starts_goals
        starts, goals = sample_starts_goals(
            rng, batch, self.num_agents, 2, self.area_size, inside_np,
            min_dist=4 * self._params["car_radius"], max_travel=self.max_travel,
        )
        agent = torch.from_numpy(starts).to(self.device)
        goal = torch.from_numpy(goals).to(self.device)
        obstacles = Rectangle(*[t.to(self.device) for t in obs_cpu])
        return self.get_graph(agent, goal, obstacles)

    def agent_xdot(self, agent_states: Tensor, action: Tensor) -> Tensor:
        return action

    def control_affine_dyn(self, state: Tensor) -> Tuple[Tensor, Tensor]:
        f = torch.zeros_like(state)
        g = torch.eye(2, device=state.device).expand(*state.shape[:-1], 2, 2)
        return f, g

    def u_ref(self, graph: GraphBatch) -> Tensor:
        """LQR toward goal with error clipped by comm radius
        (reference single_integrator.py:298-304)."""
        error = graph.goal_states - graph.agent_states
        norm = torch.linalg.vector_norm(error, dim=-1, keepdim=True).clamp_min(1e-9)
        error_max = (error / norm * self._params["comm_radius"]).abs()
        error = torch.clamp(error, -error_max, error_max)
        return self.clip_action(error @ self._K.to(error.device).t())

    def safe_mask(self, graph: GraphBatch) -> Tensor:
        """dist > 2.5r to all AND not inside(1.5r) (reference
        single_integrator.py:323-341)."""
        pos = graph.agent_states[..., :2]
        n = self.num_agents
        r = self._params["car_radius"]
        dist = torch.cdist(pos, pos) + torch.eye(n, device=pos.device) * (2 * r + 1)
        safe_agent = (dist > 2.5 * r).all(dim=-1)
        safe_obs = ~graph.env_states.inside(pos, r=1.5 * r)
        return safe_agent & safe_obs

    def unsafe_mask(self, graph: GraphBatch) -> Tensor:
        """No velocity: unsafe == collision (reference
        single_integrator.py:343-360)."""
        pos = graph.agent_states[..., :2]
        n = self.num_agents
        r = self._params["car_radius"]
        dist = torch.cdist(pos, pos) + torch.eye(n, device=pos.device) * (2 * r + 1)
        unsafe_agent = (dist < 2 * r).any(dim=-1)
        unsafe_obs = graph.env_states.inside(pos, r=r)
        return unsafe_agent | unsafe_obs

    def collision_mask(self, graph: GraphBatch) -> Tensor:
        return self.unsafe_mask(graph)
