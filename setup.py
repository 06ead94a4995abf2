# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Build the in-tree HIP extension for gfx950:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces mlrun_amd/_hip_ops*.so (travels with the repo snapshot to the
GPU box; never installed into site-packages).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

setup(
    name="mlrun_amd",
    version="0.1.0",
    packages=["mlrun_amd"],
    ext_modules=[
        CUDAExtension(
            name="mlrun_amd._hip_ops",
            sources=[
                "mlrun_amd/ops/hip/bindings.cpp",
                "mlrun_amd/ops/hip/kernels.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
