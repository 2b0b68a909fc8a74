"""Build the in-tree gfx950 HIP extension: python setup.py build_ext --inplace."""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

setup(
    name="gcbfplus_amd",
    version="0.1.0",
    packages=["gcbfplus_amd"],
    ext_modules=[
        CUDAExtension(
            name="gcbfplus_amd._C",
            sources=[
                "gcbfplus_amd/ops/hip/gemm.hip",
                "gcbfplus_amd/ops/hip/softmax_aggr.hip",
                "gcbfplus_amd/ops/hip/raytrace.hip",
                "gcbfplus_amd/ops/hip/proxqp.hip",
                "gcbfplus_amd/ops/hip/edge_msg.hip",
                "gcbfplus_amd/ops/hip/loss.hip",
                "gcbfplus_amd/ops/hip/env_step.hip",
                "gcbfplus_amd/ops/hip/loss_prep.hip",
                "gcbfplus_amd/ops/hip/gather.hip",
                "gcbfplus_amd/ops/hip/optimizer.hip",
                "gcbfplus_amd/ops/hip/bindings.hip",
            ],
            extra_compile_args={"cxx": ["-O3"], "nvcc": ["-O3"]},
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
