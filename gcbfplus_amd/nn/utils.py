"""NN helper utilities (reference ``gcbfplus/nn/utils.py:19-51``).

The reference needs these to parameterise flax modules; in the torch build
init/activation live inside ``Dense``/``MLP`` (mlp.py), so these are thin
equivalents kept under the same names for API parity.
"""
from __future__ import annotations

import math
from typing import Callable, Generator, Iterable, Tuple, TypeVar

import torch
from torch import Tensor

_Elem = TypeVar("_Elem")


def default_nn_init(tensor: Tensor) -> Tensor:
    """Xavier-uniform, the reference's flax default (reference :19)."""
    fan_in, fan_out = tensor.shape[0], tensor.shape[-1]
    limit = math.sqrt(6.0 / (fan_in + fan_out))
    with torch.no_grad():
        return tensor.uniform_(-limit, limit)


def scaled_init(initializer: Callable[[Tensor], Tensor], scale: float) -> Callable[[Tensor], Tensor]:
    """Wrap an initializer to scale its output (reference :22-26)."""

    def inner(tensor: Tensor) -> Tensor:
        initializer(tensor)
        with torch.no_grad():
            tensor.mul_(scale)
        return tensor

    return inner


_ACT_STR = {
    "relu": torch.relu,
    "tanh": torch.tanh,
    "elu": torch.nn.functional.elu,
    "gelu": torch.nn.functional.gelu,
    "silu": torch.nn.functional.silu,
    "sigmoid": torch.sigmoid,
}


def get_act_from_str(act_str: str):
    """Activation registry (reference :32-36)."""
    return _ACT_STR[act_str]


def signal_last_enumerate(it: Iterable[_Elem]) -> Generator[Tuple[bool, int, _Elem], None, None]:
    """Yield (is_last, idx, elem) (reference :39-47)."""
    items = list(it)
    n = len(items)
    for i, elem in enumerate(items):
        yield i == n - 1, i, elem


def safe_get(arr: Tensor, idx: Tensor) -> Tensor:
    """Gather with NaN fill for out-of-range indices (reference :50-51).

    The dense edge-slot layout never produces out-of-range indices, so this
    exists only for users of the reference API.
    """
    ok = (idx >= 0) & (idx < arr.shape[0])
    safe_idx = idx.clamp(0, arr.shape[0] - 1)
    out = arr[safe_idx]
    return torch.where(ok.reshape(ok.shape + (1,) * (out.dim() - ok.dim())), out,
                       torch.full_like(out, float("nan")))
