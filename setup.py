"""Build the in-tree HIP extension: ``python setup.py build_ext --inplace``.

Targets gfx950 (MI355X) only — set by PYTORCH_ROCM_ARCH below.  The built
``bigclam/_C*.so`` is git-ignored but ships with the gpurun snapshot.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import pybind11
from setuptools import Extension, setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

setup(
    name="bigclam",
    version="0.1.0",
    packages=["bigclam"],
    ext_modules=[
        Extension(
            name="bigclam._io_native",
            sources=["bigclam/kernels/io_native.cpp"],
            include_dirs=[pybind11.get_include()],
            extra_compile_args=["-O3", "-std=c++17"],
            language="c++",
        ),
        CUDAExtension(
            name="bigclam._C",
            sources=[
                "bigclam/kernels/bindings.cpp",
                "bigclam/kernels/bigclam_kernels.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
