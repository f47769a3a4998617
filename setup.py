"""Build the gfx950 HIP extensions in-tree.

python setup.py build_ext --inplace   (PYTORCH_ROCM_ARCH=gfx950)
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "beta9_amd", "ops", "csrc")

ext_flags = ["-O3", "-std=c++17"]

setup(
    name="beta9-amd",
    version="0.1.0",
    packages=find_packages(include=["beta9_amd", "beta9_amd.*"]),
    entry_points={"console_scripts": ["b9 = beta9_amd.cli.main:main"]},
    ext_modules=[
        CUDAExtension(
            name="beta9_amd.ops._hipsnap_ext",
            sources=[os.path.join(CSRC, "hipsnap.hip")],
            extra_compile_args={"cxx": ext_flags, "nvcc": ext_flags},
        ),
        CUDAExtension(
            name="beta9_amd.ops._gemm_ext",
            sources=[os.path.join(CSRC, "gemm.hip")],
            extra_compile_args={"cxx": ext_flags, "nvcc": ext_flags},
        ),
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
