"""Build the bodo_amd native HIP extension for gfx950 (MI355X), in-tree.

    python setup.py build_ext --inplace
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="bodo_amd_kernels",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="bodo_amd_kernels",
            sources=["csrc/kernels.hip", "csrc/gemm.hip", "csrc/parquet.hip"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
