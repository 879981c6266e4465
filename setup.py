"""Build the in-tree HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands at shifu_amd/ops/_shifu_hip*.so (in-tree, so it travels with
repo snapshots to GPU boxes).  hipcc cross-compiles without a GPU present.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CppExtension, CUDAExtension

setup(
    name="shifu_amd",
    version="0.1.0",
    packages=["shifu_amd"],
    ext_modules=[
        CUDAExtension(
            name="shifu_amd.ops._shifu_hip",
            sources=["shifu_amd/ops/hip/shifu_ops.hip"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        ),
        CppExtension(
            name="shifu_amd.io._shifu_io",
            sources=["shifu_amd/io/cpp/csv_reader.cpp"],
            libraries=["z"],
            extra_compile_args=["-O3", "-std=c++17"],
        ),
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
