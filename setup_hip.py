"""Build the gfx950 HIP extension `evotorch_amd._C` in-tree.

Usage:  python setup_hip.py build_ext --inplace
(hipcc cross-compiles for gfx950 without a GPU present; the built .so
travels to the GPU box with the repo snapshot.)
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", "8")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join("evotorch_amd", "ops", "hip")

setup(
    name="evotorch_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="evotorch_amd._C",
            sources=[
                os.path.join(SRC, "bindings.cpp"),
                os.path.join(SRC, "es_kernels.hip"),
                os.path.join(SRC, "rollout.hip"),
                os.path.join(SRC, "rollout_v7.hip"),
                os.path.join(SRC, "pareto.hip"),
                os.path.join(SRC, "cma.hip"),
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
