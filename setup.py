"""In-tree build of the tiny_deepspeed_amd HIP extension for gfx950 (MI355X).

Build:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
(the arch is also defaulted below so a plain invocation works on the CPU-only
build container — hipcc cross-compiles gfx950 without a GPU).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
# keep hipcc from probing for a GPU
os.environ.setdefault("HCC_AMDGPU_TARGET", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

SOURCES = [
    "csrc/bind.cpp",
    "csrc/kernels/layernorm.hip",
    "csrc/kernels/elementwise.hip",
    "csrc/kernels/embedding.hip",
    "csrc/kernels/cross_entropy.hip",
    "csrc/kernels/optim.hip",
    "csrc/kernels/attention.hip",
    "csrc/kernels/gemm_tn.hip",
    "csrc/kernels/debug.hip",
]

setup(
    name="tiny_deepspeed_amd_C",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="tiny_deepspeed_amd._C",
            sources=SOURCES,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
