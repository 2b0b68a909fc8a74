"""gcbfplus_amd — MI355X-native neural graph-CBF multi-agent control framework.

A from-scratch build with the capabilities of MIT-REALM/gcbfplus (T-RO 2025):
GCBF+ / GCBF training, hand-derived CBF-QP baselines, five dynamics envs with
LiDAR obstacle sensing — on PyTorch-ROCm with hand-written CDNA4 (gfx950) HIP
kernels for the hot paths and RCCL-over-xGMI data parallelism.
"""

__version__ = "0.1.0"
