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

# Canonical sources, listed explicitly: the glob previously also picked up
# hipify build artifacts (*_hip.hip) that torch's cpp_extension emits next to
# the originals, risking duplicate extern-C symbols from drifted copies.
HIP_SOURCES = [
    "attn_decode.hip",
    "attn_prefill.hip",
    "mfma_probe.hip",
    "rmsnorm.hip",
    "rope_kv.hip",
    "silu_mul.hip",
    "skinny_gemm.hip",
    "kv_peer_copy.hip",
    "allreduce.hip",
    "bindings.cpp",
]
sources = [os.path.join(HIP_DIR, f) for f in HIP_SOURCES]

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
