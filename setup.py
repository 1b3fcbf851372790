"""Build the in-tree CDNA4 HIP extension: python setup.py build_ext --inplace

Targets gfx950 (MI355X) only — no fat binaries, no CUDA path.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

setup(
    name="oktopk_amd",
    version="0.1.0",
    packages=["oktopk_amd"],
    ext_modules=[
        CUDAExtension(
            name="oktopk_amd._hip_ops",
            sources=[
                "oktopk_amd/ops/csrc/bindings.cpp",
                "oktopk_amd/ops/csrc/kernels.hip",
                "oktopk_amd/ops/csrc/linear_gelu.hip",
                "oktopk_amd/ops/csrc/add_layernorm.hip",
                "oktopk_amd/ops/csrc/attention.hip",
                "oktopk_amd/ops/csrc/attention_fa.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
