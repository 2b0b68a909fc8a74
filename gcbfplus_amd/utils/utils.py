"""Small tensor/tree helpers (reference ``gcbfplus/utils/utils.py:22-171``).

Most of the reference's helpers exist to make pytrees jit/vmap-friendly; in
torch the same operations are one-liners, kept here under the same names so
reference users find them.
"""
from __future__ import annotations

from typing import List, Sequence

import numpy as np
import torch
from torch import Tensor


def merge01(x: Tensor) -> Tensor:
    """(a, b, ...) -> (a*b, ...) (reference :22-23)."""
    return x.reshape(x.shape[0] * x.shape[1], *x.shape[2:])


def mask2index(mask: Tensor, n_true: int) -> Tensor:
    """Indices of the n_true True entries (reference :74-76 via top_k)."""
    return torch.nonzero(mask, as_tuple=False).flatten()[:n_true]


def tree_map(fn, tree):
    if tree is None:
        return None
    if isinstance(tree, Tensor):
        return fn(tree)
    if isinstance(tree, tuple) and hasattr(tree, "_fields"):
        return type(tree)(*[tree_map(fn, v) for v in tree])
    if isinstance(tree, (list, tuple)):
        return type(tree)(tree_map(fn, v) for v in tree)
    if isinstance(tree, dict):
        return {k: tree_map(fn, v) for k, v in tree.items()}
    return tree


def tree_index(tree, idx):
    return tree_map(lambda t: t[idx], tree)


def tree_stack(trees: Sequence):
    """Stack a list of congruent trees along a new leading dim (:164-171)."""
    first = trees[0]
    if isinstance(first, Tensor):
        return torch.stack(list(trees), dim=0)
    if isinstance(first, tuple) and hasattr(first, "_fields"):
        return type(first)(*[tree_stack([t[i] for t in trees]) for i in range(len(first))])
    raise TypeError(type(first))


def tree_merge(trees: Sequence):
    """Concatenate trees along dim 0 (:153-161)."""
    first = trees[0]
    if isinstance(first, Tensor):
        return torch.cat(list(trees), dim=0)
    if isinstance(first, tuple) and hasattr(first, "_fields"):
        return type(first)(*[tree_merge([t[i] for t in trees]) for i in range(len(first))])
    raise TypeError(type(first))


def tree_concat_at_front(t1, t2):
    return tree_merge([t1, t2])


def torch2np(x):
    """jax2np analogue (:66-71): detach to host numpy."""
    return tree_map(lambda t: t.detach().cpu().numpy(), x)


def np2torch(x, device="cpu"):
    """np2jax analogue (:66-71): numpy (trees) -> device tensors."""
    if isinstance(x, np.ndarray):
        return torch.from_numpy(x).to(device)
    if isinstance(x, tuple) and hasattr(x, "_fields"):
        return type(x)(*[np2torch(v, device) for v in x])
    return x


def chunk_vmap(fn, chunks: int):
    """Apply a batched fn in ``chunks`` pieces to bound peak memory
    (reference :96-114, used for huge eval batches). The torch build's fns
    are natively batched, so this is just chunked apply + tree_merge."""

    def wrapper(*args):
        batch = None
        for t in (args[0],) if isinstance(args[0], Tensor) else args[0]:
            batch = t.shape[0] if isinstance(t, Tensor) else len(t)
            break
        idxs = np.array_split(np.arange(batch), chunks)
        out = [fn(*[tree_index(a, torch.as_tensor(ix)) for a in args]) for ix in idxs]
        return tree_merge(out)

    return wrapper


class MutablePatchCollection:
    """matplotlib PatchCollection whose patches can be moved between frames
    (reference :116-124). Defined lazily so headless installs without
    matplotlib can still import this module."""

    def __new__(cls, patches, *args, **kwargs):
        import matplotlib.collections as mcollections

        class _Mutable(mcollections.PatchCollection):
            def __init__(self, patches_, *a, **kw):
                self._paths = None
                self.patches = patches_
                mcollections.PatchCollection.__init__(self, patches_, *a, **kw)

            def get_paths(self):
                self.set_paths(self.patches)
                return self._paths

        return _Mutable(patches, *args, **kwargs)


def save_anim(ani, path):
    """Save a matplotlib animation with frame progress (reference :141-149)."""
    total = getattr(ani, "_save_count", None) or 0

    def cb(curr_frame: int, total_frames: int):
        if total and curr_frame % max(1, total // 10) == 0:
            print(f"  animating frame {curr_frame}/{total}", flush=True)

    ani.save(str(path), progress_callback=cb)
