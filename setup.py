"""In-tree build of the rbg_amd HIP extension for gfx950 (MI355X).

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The resulting rbg_amd/ops/_hip_ops*.so travels with the repo snapshot to GPU
boxes (no JIT cache involved).
"""
import os
import sys

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, "rbg_amd", "ops", "hip")

sources = [os.path.join(HIP_DIR, f) for f in sorted(os.listdir(HIP_DIR))
           if f.endswith((".hip", ".cpp"))]

ext = CUDAExtension(
    name="rbg_amd.ops._hip_ops",
    sources=sources,
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="rbg-mi355x",
    version="0.1.0",
    packages=["rbg_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
