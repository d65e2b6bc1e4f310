"""Build the mi355ml HIP extension in-tree (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built mi355ml/_hip_ext*.so stays inside the package so the repo snapshot
carries it to GPU boxes.
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ext = CUDAExtension(
    name="mi355ml._hip_ext",
    sources=[
        "mi355ml/_hip/bindings.cpp",
        "mi355ml/_hip/kernels.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="mi355ml",
    version="0.1.0",
    packages=find_packages(include=["mi355ml", "mi355ml.*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
