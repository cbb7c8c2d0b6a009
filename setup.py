"""In-tree build of the _kvidx_C native extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Builds for gfx950 (MI355X) only - no multi-arch fatbins, no CUDA paths.
The .so lands inside llmd_kvcache_amd/ops/ so the repo snapshot carries it
to GPU hosts.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import torch  # noqa: E402
from torch.utils import cpp_extension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "llmd_kvcache_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "cpu_ops.cpp"),
    os.path.join(CSRC, "wirefront.cpp"),
]

extra_defs = []
if os.environ.get("KVIDX_DEBUG", "0") == "1":
    # hipMemcheck-style kernel/table invariant traps (kvidx_common.h)
    extra_defs.append("-DKVIDX_DEBUG_ASSERTS")

with_hip = torch.version.hip is not None
if with_hip:
    sources.append(os.path.join(CSRC, "hip_ops.hip"))
    ext = cpp_extension.CUDAExtension(
        name="llmd_kvcache_amd.ops._kvidx_C",
        sources=sources,
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17", "-DKVIDX_WITH_HIP"] + extra_defs,
            "nvcc": ["-O3", "-std=c++17", "-DKVIDX_WITH_HIP"] + extra_defs,
        },
    )
else:  # CPU-only fallback (non-ROCm torch)
    ext = cpp_extension.CppExtension(
        name="llmd_kvcache_amd.ops._kvidx_C",
        sources=sources,
        extra_compile_args=["-O3", "-std=c++17"] + extra_defs,
    )

setup(
    name="llmd_kvcache_amd",
    version="0.1.0",
    packages=[
        "llmd_kvcache_amd",
        "llmd_kvcache_amd.kvblock",
        "llmd_kvcache_amd.kvevents",
        "llmd_kvcache_amd.tokenization",
        "llmd_kvcache_amd.tokenization.prefixstore",
        "llmd_kvcache_amd.preprocessing",
        "llmd_kvcache_amd.metrics",
        "llmd_kvcache_amd.parallel",
        "llmd_kvcache_amd.service",
        "llmd_kvcache_amd.ops",
        "llmd_kvcache_amd.utils",
    ],
    ext_modules=[ext],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
