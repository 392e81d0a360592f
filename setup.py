"""Build the gfx950 HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces distributedtraining_amd/_dta_hip*.so (git-ignored; travels with
the repo snapshot to GPU boxes).
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

SRC = [
    "distributedtraining_amd/ops/hip/bindings.cpp",
    "distributedtraining_amd/ops/hip/lt_fused.cpp",
    "distributedtraining_amd/ops/hip/elementwise.hip",
    "distributedtraining_amd/ops/hip/adamw.hip",
    "distributedtraining_amd/ops/hip/norms.hip",
    "distributedtraining_amd/ops/hip/ce.hip",
    "distributedtraining_amd/ops/hip/embedding.hip",
    "distributedtraining_amd/ops/hip/rope.hip",
    "distributedtraining_amd/ops/hip/gemv.hip",
    "distributedtraining_amd/ops/hip/merge.hip",
    "distributedtraining_amd/ops/hip/attention.hip",
]

setup(
    name="distributedtraining_amd",
    version="0.2.0",
    packages=find_packages(include=["distributedtraining_amd*"]),
    ext_modules=[
        CUDAExtension(
            name="distributedtraining_amd._dta_hip",
            sources=SRC,
            libraries=["hipblaslt"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
