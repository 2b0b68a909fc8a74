"""Abstract multi-agent controller API (reference algo/base.py:10-68)."""
from __future__ import annotations

import abc
from typing import Optional, Tuple

import torch
from torch import Tensor

from ..trainer.data import Rollout
from ..utils.graph import GraphBatch


class MultiAgentController(abc.ABC):
    def __init__(self, env, node_dim: int, edge_dim: int, action_dim: int, n_agents: int):
        self._env = env
        self._node_dim = node_dim
        self._edge_dim = edge_dim
        self._action_dim = action_dim
        self._n_agents = n_agents

    node_dim = property(lambda self: self._node_dim)
    edge_dim = property(lambda self: self._edge_dim)
    action_dim = property(lambda self: self._action_dim)
    n_agents = property(lambda self: self._n_agents)

    @property
    @abc.abstractmethod
    def config(self) -> dict: ...

    @abc.abstractmethod
    def act(self, graph: GraphBatch) -> Tensor:
        """Deterministic evaluation action (B, N, nu)."""

    @abc.abstractmethod
    def step(self, graph: GraphBatch) -> Tuple[Tensor, Tensor]:
        """Rollout-collection action + log_pi."""

    @abc.abstractmethod
    def update(self, rollout: Rollout, step: int) -> dict: ...

    @abc.abstractmethod
    def save(self, save_dir: str, step: int): ...

    @abc.abstractmethod
    def load(self, load_dir: str, step: int): ...
