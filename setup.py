"""Build the gfx950 native extension IN-TREE.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands next to the package (spark_sklearn_amd/_hip_ops*.so) so it
travels with the repo snapshot to the GPU box (no JIT cache dependence).
"""

import os
import sys

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ext = CUDAExtension(
    name="spark_sklearn_amd._hip_ops",
    sources=["spark_sklearn_amd/ops/csrc/ska_ops.hip"],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="spark_sklearn_amd",
    version="0.1.0",
    packages=["spark_sklearn_amd", "spark_sklearn_amd.models",
              "spark_sklearn_amd.ops", "spark_sklearn_amd.parallel"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
