"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands at pipegoose_amd/ops/_C*.so and travels with the repo
snapshot to GPU boxes.
"""
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

SRC = [
    "pipegoose_amd/ops/csrc/bindings.cpp",
    "pipegoose_amd/ops/csrc/layer_norm.hip",
    "pipegoose_amd/ops/csrc/bias_gelu.hip",
    "pipegoose_amd/ops/csrc/cross_entropy.hip",
    "pipegoose_amd/ops/csrc/attention.hip",
    "pipegoose_amd/ops/csrc/rms_norm.hip",
    "pipegoose_amd/ops/csrc/rope.hip",
    "pipegoose_amd/ops/csrc/router.hip",
    "pipegoose_amd/ops/csrc/silu_mul.hip",
    "pipegoose_amd/ops/csrc/adamw.hip",
    "pipegoose_amd/ops/csrc/gemm.hip",
    "pipegoose_amd/ops/csrc/fp8_quant.hip",
]

setup(
    name="pipegoose_amd",
    version="0.1.0",
    packages=find_packages(include=["pipegoose_amd*"]),
    ext_modules=[
        CUDAExtension(
            name="pipegoose_amd.ops._C",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
