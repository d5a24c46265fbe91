"""Build the gordo_amd package + the in-tree HIP extension for gfx950.

Usage (in-tree, so the .so travels with the repo snapshot):
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""
import os

from setuptools import find_packages, setup

ext_modules = []
cmdclass = {}

try:
    import torch
    from torch.utils.cpp_extension import BuildExtension, CUDAExtension

    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    hip_sources = [
        "gordo_amd/ops/hip/bindings.cpp",
        "gordo_amd/ops/hip/hourglass_kernels.hip",
        "gordo_amd/ops/hip/lstm_kernels.hip",
        "gordo_amd/ops/hip/lstm_h32_kernels.hip",
        "gordo_amd/ops/hip/lstm_wide_kernels.hip",
        "gordo_amd/ops/hip/lstm_wide256_kernels.hip",
    ]
    hip_sources = [s for s in hip_sources if os.path.exists(s)]
    if hip_sources and torch.version.hip is not None:
        ext_modules.append(
            CUDAExtension(
                name="gordo_amd_hip",
                sources=hip_sources,
                extra_compile_args={
                    "cxx": ["-O3", "-std=c++17"],
                    "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
                },
            )
        )
        cmdclass["build_ext"] = BuildExtension
except ImportError:
    pass

setup(
    name="gordo-amd",
    version="0.1.0",
    description="MI355X-native time-series anomaly-detection model framework",
    packages=find_packages(exclude=["tests", "tests.*"]),
    python_requires=">=3.10",
    ext_modules=ext_modules,
    cmdclass=cmdclass,
    entry_points={
        "console_scripts": ["gordo-amd = gordo_amd.cli.cli:gordo"],
    },
)
