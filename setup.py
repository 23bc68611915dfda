"""In-tree build of the fmda_amd HIP extension for MI355X (gfx950).

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so lands at fmda_amd/ops/_fmda_hip*.so (git-ignored; travels with
the repo snapshot to GPU boxes).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="fmda_amd",
    version="0.1.0",
    packages=["fmda_amd", "fmda_amd.data", "fmda_amd.models", "fmda_amd.ops",
              "fmda_amd.parallel", "fmda_amd.runtime"],
    ext_modules=[
        CUDAExtension(
            name="fmda_amd.ops._fmda_hip",
            sources=[
                "fmda_amd/ops/csrc/bindings.cpp",
                "fmda_amd/ops/csrc/gru_kernels.hip",
                "fmda_amd/ops/csrc/optim_kernels.hip",
                "fmda_amd/ops/csrc/checkpoint.cpp",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
