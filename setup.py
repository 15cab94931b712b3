"""In-tree build of the deepof_amd HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands in deepof_amd/ops/hip/ so it travels with the repo
snapshot to GPU boxes (no JIT cache dependence).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join("deepof_amd", "ops", "hip")

ext = CUDAExtension(
    name="deepof_amd.ops.hip._deepof_hip",
    sources=[
        os.path.join(SRC, "ext.cpp"),
        os.path.join(SRC, "loss_kernels.hip"),
        os.path.join(SRC, "correlation.hip"),
        os.path.join(SRC, "adam.hip"),
        os.path.join(SRC, "conv_mfma.hip"),
        os.path.join(SRC, "conv_wrw2.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="deepof_amd",
    version="0.1.0",
    packages=["deepof_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
