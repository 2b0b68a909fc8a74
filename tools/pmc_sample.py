"""Tiny fixed-shape kernel sample for rocprofv3 --pmc runs (dev tool):
a handful of launches of the flagship kernels at training shapes."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from gcbfplus_amd import _C, ops

M, K, N = 167936, 256, 256
x = torch.randn(M, K, device="cuda").to(torch.bfloat16)
w = torch.randn(K, N, device="cuda").to(torch.bfloat16)
b = torch.randn(N, device="cuda")
dz = torch.randn(M, N, device="cuda").to(torch.bfloat16)
for _ in range(3):
    y = _C.gemm_bias_act(x, w, b, 1)           # glds fwd GEMM
    dw, db = _C.gemm_tn(x, dz, dz, 0)          # dW split-M
    dx = _C.gemm_bt(dz, w.t().contiguous(), y, 1)  # dX with fused relu-bwd
st = torch.randn(16, 272, 4, device="cuda")
for _ in range(3):
    e = ops.edge_msg_in(st, 8, 32, 2, 0.5)     # fused edge input
torch.cuda.synchronize()
print("pmc sample done")
