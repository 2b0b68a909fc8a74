"""Tanh-squashed Gaussian action distribution.

Torch equivalent of the reference's tfp-based
``algo/module/distribution.py:10-60`` (``TanhTransformedDistribution``): a
Normal pushed through tanh with the tail-mass handling of the original —
``log_prob`` at/beyond the ``threshold`` returns the log of the probability
mass of the whole tail (the reference inherits this from
``tfd.TransformedDistribution`` + its explicit threshold clipping), and
``entropy`` uses the one-sample estimator
``H(base) + E[log |d tanh / dx|]`` (reference :37-44).
Used by the PPO TanhNormal policy head (shipped unused in the reference;
kept for capability parity).
"""
from __future__ import annotations

import math

import torch
from torch import Tensor
from torch.distributions import Normal


class TanhTransformedDistribution:
    def __init__(self, loc: Tensor, scale: Tensor, threshold: float = 0.999):
        self.base = Normal(loc, scale)
        self.threshold = threshold
        self._inv_t = math.atanh(threshold)

    def sample(self) -> Tensor:
        with torch.no_grad():
            return torch.tanh(self.base.sample())

    def rsample(self) -> Tensor:
        return torch.tanh(self.base.rsample())

    def log_prob(self, value: Tensor) -> Tensor:
        t = self.threshold
        v = value.clamp(-t, t)
        x = torch.atanh(v)
        inside = self.base.log_prob(x) - torch.log1p(-v * v)
        # tail mass: P[X <= -atanh(t)] / P[X >= atanh(t)] (reference :25-35)
        left = self.base.cdf(torch.full_like(value, -self._inv_t)).clamp_min(1e-38).log()
        right = (1.0 - self.base.cdf(torch.full_like(value, self._inv_t))).clamp_min(1e-38).log()
        out = torch.where(value <= -t, left, inside)
        return torch.where(value >= t, right, out)

    def entropy(self) -> Tensor:
        x = self.base.rsample()
        return self.base.entropy() + torch.log1p(-torch.tanh(x) ** 2)

    def mode(self) -> Tensor:
        return torch.tanh(self.base.mean)
