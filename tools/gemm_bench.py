import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
"""Micro-bench the GEMM paths at the real training shapes."""
import time
import torch
from gcbfplus_amd import _C

def bench(M, K, N, iters=50):
    x = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    w = torch.randn(K, N, device="cuda").to(torch.bfloat16)
    b = torch.randn(N, device="cuda")
    for _ in range(5):
        y = _C.gemm_bias_act(x, w, b, 1)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        y = _C.gemm_bias_act(x, w, b, 1)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    tf = 2 * M * K * N / dt / 1e12
    # correctness spot check
    ref = (x[:256].float() @ w.float() + b).relu()
    err = (y[:256].float() - ref).abs().max().item()
    print(f"M={M:6d} K={K:3d} N={N:3d}: {dt*1e6:7.1f} us  {tf:6.1f} TF  maxerr {err:.3f}")

def bench_tn(M, K, N, iters=50):
    x = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    dz = torch.randn(M, N, device="cuda").to(torch.bfloat16)
    for _ in range(5):
        dw, db = _C.gemm_tn(x, dz, dz, 0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        dw, db = _C.gemm_tn(x, dz, dz, 0)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    tf = 2 * M * K * N / dt / 1e12
    ref = x[:, :].float().t() @ dz.float()
    err = (dw - ref).abs().max().item() / max(1.0, ref.abs().max().item())
    print(f"TN M={M:6d} K={K:3d} N={N:3d}: {dt*1e6:7.1f} us  {tf:6.1f} TF  relerr {err:.3f}")

def bench_gemv(M, K, N, iters=50):
    x = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    w = torch.randn(K, N, device="cuda").to(torch.bfloat16)
    b = torch.randn(N, device="cuda")
    for _ in range(5):
        y = _C.gemm_bias_act(x, w, b, 0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        y = _C.gemm_bias_act(x, w, b, 0)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    gb = (2.0 * M * K) / dt / 1e9
    ref = x[:256].float() @ w.float() + b
    err = (y[:256].float() - ref).abs().max().item()
    print(f"GEMV M={M:6d} K={K:3d} N={N:2d}: {dt*1e6:7.1f} us  {gb:6.0f} GB/s  maxerr {err:.3f}")

if __name__ == "__main__":
    for shape in [(167936, 256, 256), (167936, 32, 256), (167936, 256, 128),
                  (167936, 128, 128), (4096, 256, 256), (4096, 128, 256)]:
        bench(*shape)
    for shape in [(167936, 256, 256), (167936, 128, 128), (167936, 32, 256), (4096, 256, 256)]:
        bench_tn(*shape)
    for shape in [(167936, 128, 1), (335872, 128, 1), (4096, 256, 2), (4096, 256, 1)]:
        bench_gemv(*shape)
