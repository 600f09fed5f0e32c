"""Build the native CDNA4 (gfx950) kernel extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting kolibrie_amd/ops/_native*.so ships with the source tree (it is
git-ignored but travels with repo snapshots).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="kolibrie_amd",
    version="0.1.0",
    packages=["kolibrie_amd"],
    ext_modules=[
        CUDAExtension(
            name="kolibrie_amd.ops._native",
            sources=["kolibrie_amd/ops/csrc/kernels.hip"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
