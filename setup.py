"""Build the kakveda-amd HIP extension in-tree for gfx950 (MI355X).

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so lands next to kakveda_amd/ops/ so it travels with the repo
snapshot to GPU boxes (it is git-ignored; source of truth is the .hip).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="kakveda-amd-ext",
    ext_modules=[
        CUDAExtension(
            name="kakveda_amd.ops._kakveda_hip",
            sources=["kakveda_amd/ops/hip/kakveda_kernels.hip"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
