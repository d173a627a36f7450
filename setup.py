"""Build the in-tree gfx950 HIP kernel extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands in easydist_amd/ops/ and travels with the repo snapshot to
GPU boxes (it is git-ignored but NOT gpurun-ignored).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

SRC = [
    "easydist_amd/ops/csrc/bind.cpp",
    "easydist_amd/ops/csrc/norm_kernels.hip",
    "easydist_amd/ops/csrc/ce_kernels.hip",
    "easydist_amd/ops/csrc/optim_kernels.hip",
    "easydist_amd/ops/csrc/gemm_kernels.hip",
    "easydist_amd/ops/csrc/attn_kernels.hip",
]

setup(
    name="easydist_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="easydist_amd.ops._hip_ops",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        ),
        CUDAExtension(
            name="easydist_amd.memory._mem_alloc",
            sources=["easydist_amd/memory/csrc/profiling_allocator.cpp"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3"],
            },
        ),
    ],
    cmdclass={"build_ext": BuildExtension},
)
