"""HIP-graph capture of the hot loops (launch-bound on MI355X otherwise).

The reference gets this from XLA tracing (one fused program per jit region);
the MI355X build captures the eager kernel sequence once into a hipGraph and
replays it — SURVEY.md §2.6 'one HIP graph per step'.

GraphedMinibatchStep: one GCBF/GCBF+ minibatch = zero-grads + loss forward +
backward, captured once for the fixed minibatch size; the DP all-reduce and
the fused optimizer steps stay outside (tiny, and RCCL inside graphs is not
worth the risk). Measured: the eager minibatch loop is ~93% of a training
step (384 x ~14 ms of launch overhead).

GraphedRolloutStep: one env step (policy forward + dynamics + LiDAR rescan +
graph rebuild), replayed T times per rollout with ping-pong state copies.
"""
from __future__ import annotations

from typing import Optional

import torch

from ..utils.graph import GraphBatch
from .data import FlatBatch


def _capture(body, final=None) -> torch.cuda.CUDAGraph:
    # Warmup on the DEFAULT stream, not a side stream: running the fused
    # optimizer step on a freshly-created side stream produced intermittent
    # GPU memory faults on this ROCm build (tools/debug_graphed3.py trials:
    # full=False side-stream body OK, full=True faults in the FIRST body
    # even with per-body host syncs; the identical sequence on the default
    # stream — the eager path and the production bench — never faults; the
    # boxes also warn about missing iommu=pt). torch.cuda.graph uses its
    # own capture stream internally, so a side-stream warmup is not needed.
    # Capture happens once per run; the hard syncs are free.
    torch.cuda.synchronize()
    for _ in range(3):  # warmup (optimizer/grad lazily-built state)
        body()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        (final or body)()
    torch.cuda.synchronize()
    return g


class GraphedMinibatchStep:
    def __init__(self, algo):
        self.algo = algo
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.fb: Optional[FlatBatch] = None
        self.full = False
        self.warming = False

    def _alloc(self, batch: FlatBatch):
        mb = self.algo.batch_size
        dev = batch.states.device
        st = torch.empty(mb, *batch.states.shape[1:], device=dev)
        mk = torch.empty(mb, *batch.masks.shape[1:], dtype=torch.bool, device=dev)
        sf = torch.empty(mb, *batch.safe.shape[1:], dtype=torch.bool, device=dev)
        us = torch.empty(mb, *batch.unsafe.shape[1:], dtype=torch.bool, device=dev)
        uq = None
        if batch.u_qp is not None:
            uq = torch.empty(mb, *batch.u_qp.shape[1:], device=dev)
        self.fb = FlatBatch(st, mk, sf, us, uq)

    def _final_body(self):
        self.warming = False
        self._body()

    def _body(self):
        algo = self.algo
        algo.dp_gbuf.zero_()  # both optimizers' gflats are slices of this
        total, _ = algo._loss(self.fb, want_info=False)
        total.backward()
        # single-GPU: optimizer inside the CAPTURE, but not in the warmup
        # bodies — repeated loss+backward+fused-optimizer sequences outside
        # the replayed graph intermittently page-fault on this ROCm pool
        # (tools/debug_optstream.py; the replayed graph itself is stable
        # across thousands of steps). FusedAdamW has no lazy state, so
        # capturing it unwarmed is safe (its temps come from the capture
        # pool).
        if self.full and not self.warming:
            algo.cbf_optim.step()
            algo.actor_optim.step()

    def run(self, batch: FlatBatch, idx: torch.Tensor) -> bool:
        """Returns True if the graphed path handled this minibatch."""
        import os

        algo = self.algo
        if (
            os.environ.get("GCBF_NO_HIPGRAPH")
            or not batch.states.is_cuda
            or idx.numel() != algo.batch_size
            or not hasattr(algo.cbf_optim, "gflat")
        ):
            return False
        if self.fb is None:
            self._alloc(batch)
        fb = self.fb
        from .. import ops

        ext = ops._load_ext()
        if (fb.u_qp is not None and ext is not None and hasattr(ext, "mb_gather")
                and (batch.states.shape[-2] * batch.states.shape[-1]) % 4 == 0):
            # K18: all five gathers in one kernel
            ext.mb_gather(batch.states, batch.masks, batch.safe, batch.unsafe,
                          batch.u_qp, idx, fb.states, fb.masks, fb.safe,
                          fb.unsafe, fb.u_qp)
        else:
            torch.index_select(batch.states, 0, idx, out=fb.states)
            torch.index_select(batch.masks, 0, idx, out=fb.masks)
            torch.index_select(batch.safe, 0, idx, out=fb.safe)
            torch.index_select(batch.unsafe, 0, idx, out=fb.unsafe)
            if fb.u_qp is not None:
                torch.index_select(batch.u_qp, 0, idx, out=fb.u_qp)
        from ..parallel import dp

        if self.graph is None:
            self.full = not dp.is_active()
            # capture warmup EXECUTES the body (incl. optimizer steps on the
            # full path): snapshot the optimizer/param state and restore it
            # so capture is side-effect free; the replay below applies the
            # one real update.
            snap = []
            for opt in (algo.cbf_optim, algo.actor_optim):
                snap.append({k: v.clone() for k, v in opt.state_dict().items()})
            self.warming = True
            self.graph = _capture(self._body, final=self._final_body)
            self.warming = False
            for opt, sd in zip((algo.cbf_optim, algo.actor_optim), snap):
                opt.load_state_dict(sd)
        # stream capture records without executing -> always replay
        self.graph.replay()
        if not self.full:  # DP: ONE fused all-reduce between backward and the step
            dp.allreduce_mean_flat([algo.dp_gbuf])
            algo.cbf_optim.step()
            algo.actor_optim.step()
        return True


class GraphedRolloutStep:
    """Captures act_fn(graph) + env.step(graph, action) once; per-step replay
    copies the previous outputs into the static inputs. If capture fails
    (an op in some env's step is capture-illegal), ``broken`` is set and
    callers run the eager path instead."""

    def __init__(self, env, act_fn):
        self.env = env
        self.act_fn = act_fn
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.static_in: Optional[GraphBatch] = None
        self.outs = None
        self.broken = False

    def _body(self):
        with torch.no_grad():
            action = self.act_fn(self.static_in)
            if isinstance(action, tuple):
                action = action[0]
            res = self.env.step(self.static_in, action)
        self.outs = (action, res.graph, res.reward, res.cost, res.done)

    def reset(self, graph0: GraphBatch):
        """Prepare for a fresh batch of worlds (new obstacle tensors)."""
        if self.static_in is None:
            self.static_in = GraphBatch(
                states=graph0.states.clone(),
                mask=graph0.mask.clone(),
                n_agents=graph0.n_agents,
                n_rays=graph0.n_rays,
                env_states=type(graph0.env_states)(
                    *[f.clone() for f in graph0.env_states]
                ),
            )
        else:
            self.static_in.states.copy_(graph0.states)
            self.static_in.mask.copy_(graph0.mask)
            for dst, src in zip(self.static_in.env_states, graph0.env_states):
                dst.copy_(src)

    def step(self):
        """One captured env step; returns (action, graph_out, reward, cost,
        done) living in stable graph-pool storage (copy before next step)."""
        if self.graph is None:
            self.graph = _capture(self._body)
        self.graph.replay()
        return self.outs

    def try_capture(self) -> bool:
        """Capture once (warmup included); False + broken on failure."""
        if self.broken:
            return False
        if self.graph is not None:
            return True
        try:
            self.graph = _capture(self._body)
            return True
        except Exception:
            self.broken = True
            return False

    def advance(self):
        """Copy step outputs back into the static inputs for the next step."""
        _, g, _, _, _ = self.outs
        self.static_in.states.copy_(g.states)
        self.static_in.mask.copy_(g.mask)
