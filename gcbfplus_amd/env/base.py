"""Abstract multi-agent environment, batched over worlds.

The reference (``/root/reference/gcbfplus/env/base.py:34-269``) defines a
single-world env that is vmapped by the trainer. The MI355X build makes the
batch dimension explicit: every method takes/returns tensors with a leading
``B`` (n_env) dim so one kernel launch covers all worlds, and the whole
rollout step can be captured in a HIP graph.
"""
from __future__ import annotations

import abc
from typing import NamedTuple, Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from ..utils.graph import GraphBatch


class StepResult(NamedTuple):
    graph: GraphBatch
    reward: Tensor  # (B,)
    cost: Tensor  # (B,)
    done: Tensor  # (B,) bool
    info: dict


class MultiAgentEnv(abc.ABC):
    PARAMS: dict = {}
    fused_edge = False  # True when ops.edge_msg_in covers this env's edge features

    def __init__(
        self,
        num_agents: int,
        area_size: float,
        max_step: int = 256,
        max_travel: Optional[float] = None,
        dt: float = 0.03,
        params: Optional[dict] = None,
        device: Optional[torch.device] = None,
    ):
        self._num_agents = num_agents
        self._area_size = area_size
        self._max_step = max_step
        self._max_travel = max_travel
        self._dt = dt
        self._params = dict(self.PARAMS) if params is None else {**self.PARAMS, **params}
        if device is None:
            device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
        self.device = torch.device(device)

    # ---- properties ------------------------------------------------------
    @property
    def params(self) -> dict:
        return self._params

    @property
    def num_agents(self) -> int:
        return self._num_agents

    @property
    def area_size(self) -> float:
        return self._area_size

    @property
    def max_travel(self) -> Optional[float]:
        return self._max_travel

    @property
    def dt(self) -> float:
        return self._dt

    @property
    def max_episode_steps(self) -> int:
        return self._max_step

    @property
    def n_rays(self) -> int:
        return self._params.get("n_rays", 0)

    @property
    @abc.abstractmethod
    def state_dim(self) -> int: ...

    @property
    @abc.abstractmethod
    def node_dim(self) -> int: ...

    @property
    @abc.abstractmethod
    def edge_dim(self) -> int: ...

    @property
    @abc.abstractmethod
    def action_dim(self) -> int: ...

    @property
    def pos_dim(self) -> int:
        return 2

    # ---- limits ----------------------------------------------------------
    @abc.abstractmethod
    def state_lim(self, state: Optional[Tensor] = None) -> Tuple[Tensor, Tensor]: ...

    @abc.abstractmethod
    def action_lim(self) -> Tuple[Tensor, Tensor]: ...

    def _dev_lims(self, kind: str, device) -> Tuple[Tensor, Tensor]:
        """Cache limit tensors per device — keeps clip ops free of
        host->device copies (required for HIP-graph capture)."""
        cache = getattr(self, "_lim_cache", None)
        if cache is None:
            cache = self._lim_cache = {}
        key = (kind, str(device))
        if key not in cache:
            lo, hi = self.state_lim() if kind == "state" else self.action_lim()
            cache[key] = (lo.to(device), hi.to(device))
        return cache[key]

    def clip_state(self, state: Tensor) -> Tensor:
        lo, hi = self._dev_lims("state", state.device)
        return torch.clamp(state, lo, hi)

    def clip_action(self, action: Tensor) -> Tensor:
        lo, hi = self._dev_lims("action", action.device)
        return torch.clamp(action, lo, hi)

    # ---- core API --------------------------------------------------------
    @abc.abstractmethod
    def reset(self, batch: int, rng: np.random.Generator) -> GraphBatch:
        """Sample B fresh worlds (host-side rejection sampling) -> device graph."""

    @abc.abstractmethod
    def step(self, graph: GraphBatch, action: Tensor) -> StepResult:
        """Advance all worlds one dt. action: (B, N, nu)."""

    @abc.abstractmethod
    def forward_graph(self, graph: GraphBatch, action: Tensor) -> GraphBatch:
        """Differentiable one-step state prediction, topology (mask) unchanged
        (reference env/double_integrator.py:340-354)."""

    @abc.abstractmethod
    def edge_feats(self, graph: GraphBatch, states: Tensor) -> Tensor:
        """Differentiable dense edge features (B, N, D, edge_dim) from node
        states (B, V, S) under ``graph``'s fixed slot layout
        (reference env/double_integrator.py:306-320 + add_edge_feats)."""

    @abc.abstractmethod
    def u_ref(self, graph: GraphBatch) -> Tensor: ...

    @abc.abstractmethod
    def control_affine_dyn(self, state: Tensor) -> Tuple[Tensor, Tensor]:
        """f: (B, N, S), g: (B, N, S, nu) with x_dot = f + g @ u."""

    # ---- masks (B, N) bool ----------------------------------------------
    @abc.abstractmethod
    def safe_mask(self, graph: GraphBatch) -> Tensor: ...

    @abc.abstractmethod
    def unsafe_mask(self, graph: GraphBatch) -> Tensor: ...

    @abc.abstractmethod
    def collision_mask(self, graph: GraphBatch) -> Tensor: ...

    @abc.abstractmethod
    def finish_mask(self, graph: GraphBatch) -> Tensor: ...
