"""Shared algorithm utilities: grad clip + finite guard, polyak update,
horizon safe-labels, flax-layout checkpoint trees, GAE (parity, unused by
shipped algos — reference algo/utils.py:18-41).
"""
from __future__ import annotations

from typing import Dict, Iterable, List

import numpy as np
import torch
from torch import Tensor, nn


def global_grad_norm(params: Iterable[Tensor]) -> Tensor:
    sq = None
    for p in params:
        if p.grad is not None:
            s = p.grad.float().square().sum()
            sq = s if sq is None else sq + s
    if sq is None:
        return torch.tensor(0.0)
    return torch.sqrt(sq)


def clip_grads_(params: List[Tensor], max_norm: float) -> Tensor:
    """Reference semantics (trainer/utils.py:66-75): scale by
    max_norm / max(max_norm, ||g||)."""
    norm = global_grad_norm(params)
    scale = max_norm / torch.clamp(norm, min=max_norm)
    for p in params:
        if p.grad is not None:
            p.grad.mul_(scale)
    return norm


def step_if_finite(optim: torch.optim.Optimizer, params: List[Tensor], norm: Tensor) -> bool:
    """optax.apply_if_finite analogue: skip the update if the grad norm is
    non-finite (gcbf.py:102,119)."""
    if torch.isfinite(norm):
        optim.step()
        return True
    return False


@torch.no_grad()
def polyak_(target: nn.Module, online: nn.Module, tau: float):
    """target <- tau * online + (1 - tau) * target (optax.incremental_update,
    gcbf_plus.py:188-191). Fused multi-tensor (two kernel launches)."""
    pts = list(target.parameters())
    pos = list(online.parameters())
    torch._foreach_mul_(pts, 1 - tau)
    torch._foreach_add_(pts, pos, alpha=tau)
    for bt, bo in zip(target.buffers(), online.buffers()):
        bt.copy_(bo)


def horizon_safe_mask(unsafe: Tensor, horizon: int) -> Tensor:
    """(b, T, N) bool unsafe flags -> safe labels: state t is safe iff no
    unsafe flag occurs at any step in [t, t+horizon] (within the rollout);
    t=0 is always safe (reference gcbf_plus.py:160-174)."""
    b, T, N = unsafe.shape
    x = unsafe.float().permute(0, 2, 1).reshape(b * N, 1, T)
    x = torch.nn.functional.pad(x, (0, horizon))
    win = torch.nn.functional.max_pool1d(x, kernel_size=horizon + 1, stride=1)
    safe = win.reshape(b, N, T).permute(0, 2, 1) < 0.5
    safe[:, 0, :] = True
    return safe


# ---- flax-layout checkpoint trees ----------------------------------------

def _dense_tree(d) -> Dict[str, np.ndarray]:
    return {
        # export the logical rows only (Dense may pad its kernel to K%32==0)
        "kernel": d.kernel.detach()[: d.in_dim].cpu().float().numpy(),
        "bias": d.bias.detach().cpu().float().numpy(),
    }


def _mlp_tree(mlp) -> Dict[str, dict]:
    return {f"Dense_{i}": _dense_tree(l) for i, l in enumerate(mlp.layers)}


def _gnn_tree(gnn) -> Dict[str, dict]:
    out = {}
    for i, layer in enumerate(gnn.layers):
        out[f"GNNLayer_{i}"] = {
            "msg": _mlp_tree(layer.msg_mlp),
            "attn": _mlp_tree(layer.attn_mlp),
            "update": _mlp_tree(layer.update_mlp),
            "Dense_0": _dense_tree(layer.msg_out),
            "Dense_1": _dense_tree(layer.attn_out),
            "Dense_2": _dense_tree(layer.update_out),
        }
    return out


def net_to_flax_tree(net, head_name: str, out_name: str = "Dense_0") -> dict:
    """Export CBFNet / DeterministicPolicyNet params as the reference's flax
    pickle tree layout (verified from reference pickles)."""
    return {
        "params": {
            "GNN_0": _gnn_tree(net.gnn),
            head_name: _mlp_tree(net.head),
            out_name: _dense_tree(net.out),
        }
    }


def _load_dense(d, tree):
    with torch.no_grad():
        src = torch.from_numpy(np.asarray(tree["kernel"], dtype=np.float32))
        if src.shape[0] < d.kernel.shape[0]:  # padded-at-init kernel
            d.kernel.zero_()
            d.kernel[: src.shape[0]].copy_(src)
        else:
            d.kernel.copy_(src)
        d.bias.copy_(torch.from_numpy(np.asarray(tree["bias"], dtype=np.float32)))


def _load_mlp(mlp, tree):
    for i, l in enumerate(mlp.layers):
        _load_dense(l, tree[f"Dense_{i}"])


def load_flax_pickle(path: str) -> dict:
    """Unpickle a reference checkpoint (flax param tree pickled with jax
    arrays, reference gcbf.py:344-357) WITHOUT jax installed: the only jax
    global the pickles use is ``jax._src.array._reconstruct_array`` (verified
    by opcode scan of /root/reference/pretrained/*/models/1000/*.pkl), whose
    args carry a plain numpy reconstruction — map it to a numpy-returning
    shim and load everything else normally."""
    import pickle

    def _reconstruct_np(fun, args, arr_state, aval_state):
        arr = fun(*args)
        arr.__setstate__(arr_state)
        return np.asarray(arr)

    # strict allowlist: checkpoint trees contain only dicts + numpy arrays
    # (+ the jax array wrapper); anything else is refused rather than
    # executed (tighter than the reference's yaml.UnsafeLoader habits)
    from numpy._core import multiarray as _np_ma  # numpy 2.x home; the
    # pickles reference the legacy "numpy.core.multiarray" module name

    _ALLOWED = {
        ("numpy.core.multiarray", "_reconstruct"): _np_ma._reconstruct,
        ("numpy._core.multiarray", "_reconstruct"): _np_ma._reconstruct,
        ("numpy", "ndarray"): np.ndarray,
        ("numpy", "dtype"): np.dtype,
    }

    class _U(pickle.Unpickler):
        def find_class(self, module, name):
            if module.startswith("jax") and name == "_reconstruct_array":
                return _reconstruct_np
            got = _ALLOWED.get((module, name))
            if got is None:
                raise pickle.UnpicklingError(
                    f"refusing to unpickle global {module}.{name} "
                    "(checkpoint allowlist: numpy arrays only)")
            return got

    with open(path, "rb") as f:
        return _U(f).load()


def net_from_flax_tree(net, tree: dict, head_name: str, out_name: str = "Dense_0"):
    p = tree["params"]
    for i, layer in enumerate(net.gnn.layers):
        lt = p["GNN_0"][f"GNNLayer_{i}"]
        _load_mlp(layer.msg_mlp, lt["msg"])
        _load_mlp(layer.attn_mlp, lt["attn"])
        _load_mlp(layer.update_mlp, lt["update"])
        _load_dense(layer.msg_out, lt["Dense_0"])
        _load_dense(layer.attn_out, lt["Dense_1"])
        _load_dense(layer.update_out, lt["Dense_2"])
    _load_mlp(net.head, p[head_name])
    _load_dense(net.out, p[out_name])


def gae(gamma: float, lam: float, rewards: Tensor, values: Tensor, next_values: Tensor,
        dones: Tensor) -> Tensor:
    """Generalized advantage estimation (reference algo/utils.py:18-41;
    unused by the shipped algorithms, kept for the PPO surface)."""
    T = rewards.shape[1]
    adv = torch.zeros_like(rewards)
    last = torch.zeros_like(rewards[:, 0])
    for t in reversed(range(T)):
        nonterm = 1.0 - dones[:, t].float()
        delta = rewards[:, t] + gamma * next_values[:, t] * nonterm - values[:, t]
        last = delta + gamma * lam * nonterm * last
        adv[:, t] = last
    return adv
