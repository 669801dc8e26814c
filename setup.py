"""Build the MI355X-native actuator extension in-tree.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces fma_amd/_C.*.so next to the package sources so the built artifact
travels with any snapshot of the tree (no JIT cache dependency).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="fma_amd",
    version="0.2.0",
    description="MI355X-native fast model actuation",
    packages=["fma_amd"],
    ext_modules=[
        CUDAExtension(
            name="fma_amd._C",
            sources=[
                "fma_amd/csrc/actuator.cpp",
                "fma_amd/csrc/kernels.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
