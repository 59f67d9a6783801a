"""Build the HIP kernel extension in-tree (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting sail_amd/ops/_sail_kernels*.so travels to GPU boxes with the
repo snapshot; there is no JIT path.
"""
import os
import glob

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

sources = ["sail_amd/ops/csrc/module.cpp"] + sorted(glob.glob("sail_amd/ops/csrc/*.hip"))

setup(
    name="sail_amd_kernels",
    ext_modules=[
        CUDAExtension(
            name="sail_amd.ops._sail_kernels",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
