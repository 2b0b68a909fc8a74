"""RCCL (nccl backend on ROCm) data-parallel validation on a real GPU.

Launches tools/rccl_check.py under torchrun with world=2 — both ranks share
the single leased MI355X — proving nccl init, broadcast, all_agree,
the fused-bucket grad all-reduce, cross-rank param sync, and
DP2==DP1-on-same-data all execute over RCCL (the CPU suite
covers the same logic on gloo in test_dp.py).
"""
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_rccl_world2_on_device():
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = "29531"
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "2",
           "--master-addr", "127.0.0.1", "--master-port", "29531",
           os.path.join(ROOT, "tools", "rccl_check.py")]
    r = subprocess.run(cmd, cwd=ROOT, env=env, capture_output=True, text=True,
                       timeout=540)
    sys.stdout.write(r.stdout[-3000:])
    sys.stderr.write(r.stderr[-3000:])
    assert r.returncode == 0, "rccl_check failed"
    assert "PASS: RCCL DP validated on device" in r.stdout
