"""Build the dynamo_amd native extensions in-tree (gfx950 / MI355X only).

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so files land inside dynamo_amd/ so they travel with the repo
snapshot to GPU boxes (JIT caches under ~/.cache do not).
"""
import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CppExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "dynamo_amd", "csrc")

hip_sources = [
    os.path.join(CSRC, f)
    for f in (
        "bindings.cpp",
        "norm.hip",
        "rope.hip",
        "activation.hip",
        "cache.hip",
        "attention_decode.hip",
        "attention_prefill.hip",
        "sampling.hip",
        "moe.hip",
        "ipc.hip",
        "probe.hip",
    )
]

ext_modules = [
    CUDAExtension(
        name="dynamo_amd._hip",
        sources=hip_sources,
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
        },
    ),
    CppExtension(
        name="dynamo_amd._core",
        sources=[os.path.join(CSRC, "core", "core.cpp")],
        extra_compile_args=["-O3", "-std=c++17"],
    ),
]

setup(
    name="dynamo_amd",
    version="0.1.0",
    packages=["dynamo_amd"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
