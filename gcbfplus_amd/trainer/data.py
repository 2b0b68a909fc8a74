"""Rollout record — device-resident, struct-of-tensors.

The reference Rollout (``/root/reference/gcbfplus/trainer/data.py:8-31``)
stores whole GraphsTuples per step; graphs here are reconstructible from
(states, mask, obstacles), so we store only those — ~3x less HBM traffic and
the replay buffer stays device-resident (SURVEY.md §2.4 note).
"""
from __future__ import annotations

from typing import NamedTuple, Optional

import torch
from torch import Tensor

from ..utils.graph import GraphBatch


class Rollout(NamedTuple):
    states: Tensor  # (b, T, V, S) graph states at t = 0..T-1
    masks: Tensor  # (b, T, N, D) edge masks at t
    actions: Tensor  # (b, T, N, nu)
    rewards: Tensor  # (b, T)
    costs: Tensor  # (b, T)
    dones: Tensor  # (b, T) bool
    next_states: Tensor  # (b, V, S) final graph states (t = T)
    next_mask: Tensor  # (b, N, D)
    obstacles: tuple  # per-env obstacle NamedTuple, fields (b, ...)

    @property
    def length(self) -> int:
        return self.rewards.shape[0]

    @property
    def time_horizon(self) -> int:
        return self.rewards.shape[1]

    @property
    def n_data(self) -> int:
        return self.length * self.time_horizon

    def graph_at(self, env, flatten: bool = False) -> GraphBatch:
        """All T graphs as one GraphBatch of b*T (obstacles repeated per t)."""
        b, T = self.rewards.shape[:2]
        obs = type(self.obstacles)(
            *[f.repeat_interleave(T, dim=0) for f in self.obstacles]
        )
        return GraphBatch(
            states=self.states.reshape(b * T, *self.states.shape[2:]),
            mask=self.masks.reshape(b * T, *self.masks.shape[2:]),
            n_agents=env.num_agents,
            n_rays=env.n_rays,
            env_states=obs,
        )

    def graph_Tp1(self, env) -> GraphBatch:
        """All T+1 graphs (pre-step states plus the terminal state) as one
        GraphBatch of b*(T+1) — the reference's ``Tp1_graph`` eval semantics
        (``/root/reference/test.py:184-186``, RolloutResult in env/base.py:25-31)."""
        b, T = self.rewards.shape[:2]
        obs = type(self.obstacles)(
            *[f.repeat_interleave(T + 1, dim=0) for f in self.obstacles]
        )
        states = torch.cat([self.states, self.next_states[:, None]], dim=1)
        masks = torch.cat([self.masks, self.next_mask[:, None]], dim=1)
        return GraphBatch(
            states=states.reshape(b * (T + 1), *states.shape[2:]),
            mask=masks.reshape(b * (T + 1), *masks.shape[2:]),
            n_agents=env.num_agents,
            n_rays=env.n_rays,
            env_states=obs,
        )


class FlatBatch(NamedTuple):
    """Flat per-timestep training samples (graphs only, obstacles not needed
    for the GCBF+ losses — masks are precomputed at collection)."""

    states: Tensor  # (M, V, S)
    masks: Tensor  # (M, N, D)
    safe: Tensor  # (M, N) bool
    unsafe: Tensor  # (M, N) bool
    u_qp: Optional[Tensor] = None  # (M, N, nu)

    def __getitem__(self, idx) -> "FlatBatch":
        return FlatBatch(
            self.states[idx], self.masks[idx], self.safe[idx], self.unsafe[idx],
            None if self.u_qp is None else self.u_qp[idx],
        )

    @property
    def n(self) -> int:
        return self.states.shape[0]

    @staticmethod
    def cat(xs: list["FlatBatch"]) -> "FlatBatch":
        has_qp = all(x.u_qp is not None for x in xs)
        return FlatBatch(
            torch.cat([x.states for x in xs]),
            torch.cat([x.masks for x in xs]),
            torch.cat([x.safe for x in xs]),
            torch.cat([x.unsafe for x in xs]),
            torch.cat([x.u_qp for x in xs]) if has_qp else None,
        )

    def graph(self, env) -> GraphBatch:
        return GraphBatch(states=self.states, mask=self.masks, n_agents=env.num_agents,
                          n_rays=env.n_rays, env_states=None)
