# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Build the bluefog_amd native HIP extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces bluefog_amd/_C.*.so (gfx950-only; no CUDA path, no generic
multi-arch fatbin — this framework targets MI355X)."""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ext = CUDAExtension(
    name="bluefog_amd._C",
    sources=[
        "bluefog_amd/csrc/bindings.cpp",
        "bluefog_amd/csrc/bluefog_kernels.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="bluefog_amd",
    version="0.1.0",
    description="MI355X-native decentralized deep-learning training framework",
    packages=[
        "bluefog_amd",
        "bluefog_amd.ops",
        "bluefog_amd.parallel",
        "bluefog_amd.utils",
        "bluefog_amd.models",
        "bluefog_amd.run",
        "bluefog_amd.torch",
    ],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
