"""Build the in-tree HIP extension realhf_amd._C for gfx950 (MI355X).

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so stays in-tree (realhf_amd/_C*.so) so it travels with the
repo snapshot to GPU boxes.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

SRC = [
    "realhf_amd/ops/csrc/bind.cpp",
    "realhf_amd/ops/csrc/mcmc_search.cpp",
    "realhf_amd/ops/csrc/rmsnorm.hip",
    "realhf_amd/ops/csrc/elementwise.hip",
    "realhf_amd/ops/csrc/gae.hip",
    "realhf_amd/ops/csrc/interval.hip",
    "realhf_amd/ops/csrc/attn_decode.hip",
    "realhf_amd/ops/csrc/rope_decode.hip",
    "realhf_amd/ops/csrc/grouped_gemm.hip",
    "realhf_amd/ops/csrc/grouped_gemm_bwd.hip",
    "realhf_amd/ops/csrc/skinny_gemm.hip",
    "realhf_amd/ops/csrc/attn_varlen.hip",
    "realhf_amd/ops/csrc/all_reduce.hip",
    "realhf_amd/ops/csrc/attn_bwd.hip",
]

setup(
    name="realhf_amd_C",
    ext_modules=[
        CUDAExtension(
            name="realhf_amd._C",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
