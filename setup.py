"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands at fastfp_amd/ops/_fastfp_hip*.so and travels with the
repo snapshot to GPU boxes (it is git-ignored but NOT gpurun-ignored).
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

here = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="fastfp_amd.ops._fastfp_hip",
    sources=[
        "fastfp_amd/ops/hip/bindings.cpp",
        "fastfp_amd/ops/hip/fastfp_kernels.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="fastfp_amd",
    version="0.1.0",
    packages=find_packages(include=["fastfp_amd*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
