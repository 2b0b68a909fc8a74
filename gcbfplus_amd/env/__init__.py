"""Env registry + factory (reference gcbfplus/env/__init__.py:23-46, minus its
class-level-PARAMS mutation bug: overrides are passed per-instance)."""
from __future__ import annotations

from typing import Optional

from .base import MultiAgentEnv
from .double_integrator import DoubleIntegrator
from .single_integrator import SingleIntegrator

ENV = {
    "SingleIntegrator": SingleIntegrator,
    "DoubleIntegrator": DoubleIntegrator,
}

try:  # registered as they land
    from .dubins_car import DubinsCar

    ENV["DubinsCar"] = DubinsCar
except ImportError:
    pass
try:
    from .linear_drone import LinearDrone

    ENV["LinearDrone"] = LinearDrone
except ImportError:
    pass
try:
    from .crazyflie import CrazyFlie

    ENV["CrazyFlie"] = CrazyFlie
except ImportError:
    pass

DEFAULT_MAX_STEP = 256


def make_env(
    env_id: str,
    num_agents: int,
    area_size: Optional[float] = None,
    max_step: Optional[int] = None,
    max_travel: Optional[float] = None,
    num_obs: Optional[int] = None,
    n_rays: Optional[int] = None,
    device=None,
) -> MultiAgentEnv:
    assert env_id in ENV, f"Environment {env_id} not implemented."
    cls = ENV[env_id]
    params = dict(cls.PARAMS)
    if num_obs is not None:
        params["n_obs"] = num_obs
    if n_rays is not None:
        params["n_rays"] = n_rays
    return cls(
        num_agents=num_agents,
        area_size=area_size,
        max_step=DEFAULT_MAX_STEP if max_step is None else max_step,
        max_travel=max_travel,
        dt=0.03,
        params=params,
        device=device,
    )
