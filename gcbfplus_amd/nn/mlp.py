"""Dense / MLP primitives routed through the fused CDNA4 GEMM op.

Mirrors the reference flax modules (``/root/reference/gcbfplus/nn/mlp.py:6-30``,
``nn/utils.py:19-26``): xavier-uniform kernels, zero bias, optional
``act_final``/``scale_final``. Weights are stored as fp32 master copies in the
(K, N) "kernel" layout (input-dim first, flax convention) so the MFMA GEMM
consumes them without transpose.
"""
from __future__ import annotations

import math
from typing import Optional, Sequence

import torch
from torch import Tensor, nn

from .. import ops

_ACT = {"relu": ops.ACT_RELU, "tanh": ops.ACT_TANH, "none": ops.ACT_NONE}


class Dense(nn.Module):
    """y = act(x @ kernel + bias); kernel (in_dim, out_dim) fp32.

    ``pad_to``: allocate the kernel with ``pad_to`` rows, zeroing rows beyond
    ``in_dim``. The MFMA GEMM needs K % 32 == 0, and the fused edge-input op
    emits K already padded — padding the PARAMETER once at init (instead of
    cat-padding the weight every call) keeps the hot path allocation-free.
    Zero rows stay zero under AdamW (grad 0, decay of 0); checkpoints
    export/import the logical ``[:in_dim]`` slice (algo/utils.py)."""

    def __init__(self, in_dim: int, out_dim: int, act: str = "none", scale: Optional[float] = None,
                 pad_to: Optional[int] = None):
        super().__init__()
        self.in_dim, self.out_dim = in_dim, out_dim
        self.act = _ACT[act]
        rows = pad_to if pad_to is not None else in_dim
        assert rows >= in_dim
        w = torch.zeros(rows, out_dim)
        # xavier uniform on (out, in) fan convention == flax default_nn_init
        limit = math.sqrt(6.0 / (in_dim + out_dim))
        nn.init.uniform_(w[:in_dim], -limit, limit)
        if scale is not None:
            w *= scale
        self.kernel = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(out_dim))

    def forward(self, x: Tensor, row_gate: Optional[Tensor] = None) -> Tensor:
        return ops.fused_linear(x, self.kernel, self.bias, self.act, row_gate)

    def extra_repr(self) -> str:
        return f"in={self.in_dim}, out={self.out_dim}, act={self.act}"


class MLP(nn.Module):
    """Stack of Dense layers. ``act_final=False`` leaves the LAST layer
    linear (reference MLP semantics: activation after every layer except,
    optionally, the last)."""

    def __init__(
        self,
        in_dim: int,
        hid_sizes: Sequence[int],
        act: str = "relu",
        act_final: bool = True,
        scale_final: Optional[float] = None,
        pad_first_to: Optional[int] = None,
    ):
        super().__init__()
        layers = []
        d = in_dim
        n = len(hid_sizes)
        for i, h in enumerate(hid_sizes):
            last = i == n - 1
            layer_act = "none" if (last and not act_final) else act
            layers.append(Dense(d, h, act=layer_act, scale=scale_final if last else None,
                                pad_to=pad_first_to if i == 0 else None))
            d = h
        self.layers = nn.ModuleList(layers)
        self.out_dim = d

    def forward(self, x: Tensor, row_gate: Optional[Tensor] = None) -> Tensor:
        for l in self.layers:
            x = l(x, row_gate)
        return x
