"""Micro-bench of the fused edge_msg_in kernels (dev tool)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

from gcbfplus_amd import ops


def bench(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    N, R, S = 8, 32, 4
    V = 2 * N + N * R
    for B in (512, 256, 16):
        states = torch.randn(B, V, S, device="cuda")
        x = ops.edge_msg_in(states, N, R, 2, 0.5)
        ref = ops.edge_msg_in(states.cpu(), N, R, 2, 0.5)
        err = (x.float().cpu() - ref).abs().max().item()
        us = bench(lambda: ops.edge_msg_in(states, N, R, 2, 0.5))
        # backward
        st2 = states.clone().requires_grad_(True)
        g = torch.randn_like(x)

        def fb():
            out = ops.edge_msg_in(st2, N, R, 2, 0.5)
            (out * g).sum().backward()

        us_fb = bench(fb, 50)
        print(f"B={B:4d} fwd {us:7.1f} us  maxerr {err:.4f}  fwd+bwd {us_fb:7.1f} us")


if __name__ == "__main__":
    main()
