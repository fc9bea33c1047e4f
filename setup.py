"""Build the in-tree HIP extension for gfx950 (MI355X).

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built adaptdl_amd_hip*.so lands in the repo root so it travels with
repo snapshots (gpurun) and is importable without installation.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa

setup(
    name="adaptdl_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="adaptdl_amd_hip",
            sources=[
                "adaptdl_amd/ops/hip/bindings.cpp",
                "adaptdl_amd/ops/hip/gns_kernels.hip",
                "adaptdl_amd/ops/hip/bn_kernels.hip",
                "adaptdl_amd/ops/hip/conv_kernels.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        ),
    ],
    cmdclass={"build_ext": BuildExtension},
)
