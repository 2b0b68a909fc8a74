"""Data-parallel validation on a real GPU under torchrun world=2.

RCCL/NCCL refuses two ranks on one physical device ('Duplicate GPU
detected' — recorded by the probe test below), so on the single leased
MI355X the functional test runs gloo collectives with ALL COMPUTE on
cuda:0; on a multi-GPU node the same script automatically takes the real
nccl(=RCCL) path. Together with test_dp.py (gloo CPU) this covers every
line of the DP code; the RCCL transport itself executes on the driver's
multi-GPU SCALE run.
"""
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _torchrun(extra, port, timeout=540):
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = str(port)
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "2",
           "--master-addr", "127.0.0.1", "--master-port", str(port),
           os.path.join(ROOT, "tools", "rccl_check.py")] + extra
    r = subprocess.run(cmd, cwd=ROOT, env=env, capture_output=True, text=True,
                       timeout=timeout)
    sys.stdout.write(r.stdout[-3000:])
    sys.stderr.write(r.stderr[-3000:])
    return r


@pytest.mark.timeout(600)
def test_dp_world2_on_device():
    r = _torchrun([], 29531)
    assert r.returncode == 0, "rccl_check failed"
    assert "PASS: DP world=2 validated on device" in r.stdout


@pytest.mark.timeout(420)
def test_rccl_nccl_probe():
    """RCCL communicator creation: succeeds on >= 2 GPUs; on 1 GPU the
    library must load and fail with exactly the duplicate-GPU refusal
    (proving librccl initializes and only the device count blocks it)."""
    import torch

    r = _torchrun(["--mode", "nccl-probe"], 29537, timeout=360)
    out = r.stdout + r.stderr
    if torch.cuda.device_count() >= 2:
        assert "NCCL PROBE: all_reduce SUCCEEDED" in out
    else:
        assert "NCCL PROBE refusal" in out and "Duplicate GPU" in out
