"""Build the gfx950 HIP extension in-tree: `python setup.py build_ext --inplace`.

The resulting aigw_hip*.so lands at the repo root so it travels with the
source snapshot to GPU boxes (no JIT cache dependence).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import pybind11  # noqa: E402
from setuptools import Extension  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="aigw",
    version="0.1.0",
    packages=["aigw"],
    ext_modules=[
        CUDAExtension(
            name="aigw_hip",
            sources=["csrc/aigw_kernels.hip"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        ),
        Extension(
            name="aigw_native",
            sources=["csrc/aigw_native.cpp"],
            include_dirs=[pybind11.get_include()],
            extra_compile_args=["-O3", "-std=c++17"],
            language="c++",
        ),
        # CUDAExtension so admission.hip (the in-process HIP BPE counting
        # path) compiles with hipcc for gfx950 alongside the C++ server.
        CUDAExtension(
            name="aigw_fast",
            sources=[
                "csrc/aigw_fast_module.cpp",
                "csrc/fastpath.cpp",
                "csrc/admission.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
            extra_link_args=["-lpthread"],
        ),
    ],
    cmdclass={"build_ext": BuildExtension},
)
