"""Build the pipegcn_amd native extension in-tree.

Usage:  python setup.py build_ext --inplace

Compiles the C++ host graph core and the gfx950 HIP kernels into
pipegcn_amd/_C*.so (single offload arch: gfx950 / MI355X — no dual path).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="pipegcn_amd._C",
    sources=[
        "pipegcn_amd/csrc/bindings.cpp",
        "pipegcn_amd/csrc/graph_core.cpp",
        "pipegcn_amd/csrc/partitioner.cpp",
        "pipegcn_amd/csrc/hip/kernels.hip",
        "pipegcn_amd/csrc/hip/dual_gemm.hip",
        "pipegcn_amd/csrc/hip/elementwise.hip",
        "pipegcn_amd/csrc/hip/wgrad.hip",
        "pipegcn_amd/csrc/hip/dual_dgrad.hip",
    ],
    extra_compile_args={
        # -fopenmp: at::parallel_for is a header template — its OpenMP
        # pragmas expand in THIS translation unit and silently serialize
        # without the flag (measured: 8-thread partitioner phases ran 1x)
        "cxx": ["-O3", "-std=c++17", "-fopenmp"],
        "nvcc": ["-O3", "-std=c++17"],
    },
    extra_link_args=["-fopenmp"],
)

setup(
    name="pipegcn_amd",
    version="0.1.0",
    packages=["pipegcn_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
