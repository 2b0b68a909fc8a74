"""Data parallelism over RCCL/xGMI (C1 in SURVEY.md §2.7).

The reference is strictly single-device; DP here shards the
n_env_train worlds across ranks (one process per GPU, torch.distributed with
backend "nccl" == RCCL on ROCm). Each rank runs rollout + QP labels + loss
fwd/bwd on its own shard; gradients are averaged with ONE fused-bucket
all-reduce per minibatch (the two nets' grads are ~2.9 MB fp32 total, so
latency dominates on 7x153 GB/s xGMI — a single flat bucket beats many small
calls; overlap-with-backward is pointless at this size).

RNG note (expected behavior, not a defect): each rank offsets its sampling
stream (``seed + 1 + 7919*rank`` in ``algo/gcbf.py``) so DP ranks draw
different replay minibatches and reset worlds. DP=1 matches the reference's
single-device stream; DP>1 is not bit-reproducible against a single-device
run by construction (the global batch differs — that is the point of DP).
"""
from __future__ import annotations

import os
from typing import Iterable, List

import torch
import torch.distributed as dist


def is_active() -> bool:
    return dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1


def rank() -> int:
    return dist.get_rank() if is_active() else 0


def world_size() -> int:
    return dist.get_world_size() if is_active() else 1


def setup_from_env(backend: str = None) -> int:
    """Init process group from torchrun env vars; returns local rank.

    Backend: nccl(=RCCL) when every rank can own its own GPU; gloo otherwise
    (CPU runs, or over-subscribed validation runs where ranks share one GPU —
    RCCL refuses two ranks on the same device: 'Duplicate GPU detected')."""
    if "WORLD_SIZE" not in os.environ or int(os.environ["WORLD_SIZE"]) <= 1:
        return 0
    world = int(os.environ["WORLD_SIZE"])
    if not dist.is_initialized():
        if backend is None:
            backend = ("nccl" if torch.cuda.is_available()
                       and torch.cuda.device_count() >= min(
                           world, int(os.environ.get("LOCAL_WORLD_SIZE", world)))
                       else "gloo")
        dist.init_process_group(backend=backend)
    local = int(os.environ.get("LOCAL_RANK", 0))
    if torch.cuda.is_available():
        # modulo so an over-subscribed validation run (2 ranks sharing the
        # one leased GPU) still maps to a real device
        torch.cuda.set_device(local % torch.cuda.device_count())
    return local


def _gloo_cuda(t: torch.Tensor) -> bool:
    """True when the collective must stage through host memory (gloo backend
    with device-resident tensors)."""
    return t.is_cuda and dist.get_backend() == "gloo"


def broadcast_modules(modules: Iterable[torch.nn.Module]):
    if not is_active():
        return
    for m in modules:
        for p in m.parameters():
            if _gloo_cuda(p.data):
                host = p.data.cpu()
                dist.broadcast(host, src=0)
                p.data.copy_(host)
            else:
                dist.broadcast(p.data, src=0)


@torch.no_grad()
def allreduce_mean_grads(params: List[torch.nn.Parameter]):
    """One fused flat-bucket mean all-reduce over all grads."""
    if not is_active():
        return
    grads = [p.grad for p in params if p.grad is not None]
    if not grads:
        return
    flat = torch.cat([g.reshape(-1) for g in grads])
    if _gloo_cuda(flat):
        host = flat.cpu()
        dist.all_reduce(host, op=dist.ReduceOp.SUM)
        flat = host.to(flat.device)
    else:
        dist.all_reduce(flat, op=dist.ReduceOp.SUM)
    flat /= world_size()
    off = 0
    for g in grads:
        n = g.numel()
        g.copy_(flat[off : off + n].reshape(g.shape))
        off += n


def all_agree(flag: bool) -> bool:
    """Collective AND — used for control-flow that must match across ranks
    (e.g. 'is the unsafe buffer non-empty everywhere')."""
    if not is_active():
        return flag
    t = torch.tensor([1 if flag else 0], dtype=torch.int64)
    if dist.get_backend() == "nccl":
        t = t.to("cuda")
    dist.all_reduce(t, op=dist.ReduceOp.MIN)
    return bool(t.item())


@torch.no_grad()
def allreduce_mean_flat(flats: List[torch.Tensor]):
    """Mean all-reduce of pre-flattened grad buckets (one call per bucket;
    callers fuse everything into a single bucket — see algo/gcbf.py
    dp_gbuf). Gloo + CUDA tensors stage through pinned host copies."""
    if not is_active():
        return
    w = world_size()
    for f in flats:
        if _gloo_cuda(f):
            host = f.cpu()
            dist.all_reduce(host, op=dist.ReduceOp.SUM)
            f.copy_(host)
        else:
            dist.all_reduce(f, op=dist.ReduceOp.SUM)
        f /= w


