"""Flat-buffer fused AdamW/Adam (K13) — GPU path for the training loop.

All of a module's fp32 parameters live as views into ONE contiguous buffer;
grads accumulate into a matching flat buffer. One optimizer step is then:
  - one reduction pair for the global grad norm
  - one fused kernel: clip-by-global-norm + AdamW + finite guard
  - one flat all-reduce for DP (the bucket already exists)
No host synchronization anywhere — HIP-graph capturable.

Semantics == reference stack: optax adamw/adam + apply_if_finite +
compute_norm_and_clip (gcbf.py:101-119, gcbf_plus.py:109-128,
trainer/utils.py:66-75).
"""
from __future__ import annotations

from typing import List, Optional

import torch
from torch import Tensor, nn

from . import _require_ext


class FusedAdamW:
    def __init__(self, module: nn.Module, lr: float, weight_decay: float = 0.0,
                 max_grad_norm: float = float("inf"), betas=(0.9, 0.999), eps: float = 1e-8,
                 gflat_buf: Optional[Tensor] = None):
        params = [p for p in module.parameters() if p.requires_grad]
        assert all(p.dtype == torch.float32 for p in params)
        device = params[0].device
        n = sum(p.numel() for p in params)
        self.pflat = torch.empty(n, device=device)
        if gflat_buf is not None:
            # caller-provided slice of a shared DP comm bucket: several
            # optimizers' grads become ONE contiguous tensor so data-parallel
            # needs exactly one all-reduce per minibatch (SURVEY §5.8:
            # latency-bound tiny buckets — fewer calls beat overlap).
            assert gflat_buf.numel() == n and gflat_buf.is_contiguous()
            self.gflat = gflat_buf
            self.gflat.zero_()
        else:
            self.gflat = torch.zeros(n, device=device)
        # bf16 shadow of pflat, updated in lock-step by the adamw kernel:
        # fused_linear reads p._bf instead of casting w per call (~28 cast
        # kernels per minibatch saved). Cold-path param mutations (loads,
        # DP broadcast, polyak into an optimized net) must call
        # refresh_bf16().
        self.bflat = torch.empty(n, device=device, dtype=torch.bfloat16)
        off = 0
        self.params = params
        self._spans = []
        with torch.no_grad():
            for p in params:
                k = p.numel()
                self.pflat[off : off + k].copy_(p.reshape(-1))
                p.data = self.pflat[off : off + k].view_as(p)
                p.grad = self.gflat[off : off + k].view_as(p)
                p._bf = self.bflat[off : off + k].view_as(p)
                self._spans.append((off, k))
                off += k
        self.refresh_bf16()
        self.m = torch.zeros(n, device=device)
        self.v = torch.zeros(n, device=device)
        self.t = torch.zeros(1, dtype=torch.int32, device=device)
        self.lr, self.wd = lr, weight_decay
        self.b1, self.b2 = betas
        self.eps = eps
        self.max_grad_norm = max_grad_norm
        self.last_norm: Optional[Tensor] = None

    def zero_grad(self, set_to_none: bool = False):
        self.gflat.zero_()
        # re-pin views in case autograd replaced any .grad tensor
        for p, (off, k) in zip(self.params, self._spans):
            if p.grad is None or p.grad.data_ptr() != self.gflat[off : off + k].data_ptr():
                p.grad = self.gflat[off : off + k].view_as(p)

    def refresh_bf16(self):
        """Re-sync the bf16 weight shadow after any out-of-band param write
        (checkpoint load, DP broadcast, polyak copy into this module)."""
        with torch.no_grad():
            self.bflat.copy_(self.pflat)

    def step(self) -> Tensor:
        ext = _require_ext()
        self.last_norm = ext.fused_adamw_step(
            self.pflat, self.gflat, self.m, self.v, self.bflat, self.t,
            self.lr, self.b1, self.b2, self.eps, self.wd, self.max_grad_norm,
        )
        return self.last_norm

    def state_dict(self) -> dict:
        return {"m": self.m, "v": self.v, "t": self.t, "pflat": self.pflat}

    def load_state_dict(self, sd: dict):
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.t.copy_(sd["t"])
        self.pflat.copy_(sd["pflat"])
        self.refresh_bf16()
