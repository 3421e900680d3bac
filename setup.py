"""Build script for ant_ray_amd native extensions.

Two tiers:
  * `_shm_store` — plain pybind11/C++ (no torch, no GPU): the shared-memory
    object store used by the core runtime.
  * `_hip_ops`, `_gpu_ipc` — HIP/CDNA4 extensions built via
    torch.utils.cpp_extension with PYTORCH_ROCM_ARCH=gfx950 (cross-compiles
    without a GPU). Built by `python setup.py build_ext --inplace` or by
    `__graft_entry__.build()`.
"""
import os
import sys

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import pybind11
from setuptools import Extension

ROOT = os.path.dirname(os.path.abspath(__file__))

ext_modules = [
    Extension(
        "ant_ray_amd._shm_store",
        sources=["ant_ray_amd/csrc/shm_store.cpp"],
        include_dirs=[pybind11.get_include()],
        extra_compile_args=["-O3", "-std=c++17", "-fvisibility=hidden"],
        extra_link_args=["-lpthread"],
        language="c++",
    ),
]

# HIP extensions (torch): compile only when requested, since importing torch
# costs ~2s and the HIP toolchain path is separate.
if os.environ.get("ANTRAY_BUILD_HIP", "1") == "1":
    try:
        from torch.utils.cpp_extension import CppExtension  # noqa: F401

        from torch.utils import cpp_extension as tce

        hip_sources = [
            "ant_ray_amd/csrc/hip_ops.cpp",
            "ant_ray_amd/csrc/kernels/elementwise.hip",
            "ant_ray_amd/csrc/kernels/norms.hip",
            "ant_ray_amd/csrc/kernels/rope.hip",
            "ant_ray_amd/csrc/kernels/cross_entropy.hip",
            "ant_ray_amd/csrc/kernels/adamw.hip",
            "ant_ray_amd/csrc/kernels/pack.hip",
        ]
        hip_sources = [s for s in hip_sources if os.path.exists(os.path.join(ROOT, s))]
        if hip_sources:
            ext_modules.append(
                tce.CUDAExtension(
                    name="ant_ray_amd._hip_ops",
                    sources=hip_sources,
                    extra_compile_args={
                        "cxx": ["-O3", "-std=c++17"],
                        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
                    },
                )
            )
        ipc_src = "ant_ray_amd/csrc/gpu_ipc.cpp"
        if os.path.exists(os.path.join(ROOT, ipc_src)):
            ext_modules.append(
                tce.CUDAExtension(
                    name="ant_ray_amd._gpu_ipc",
                    sources=[ipc_src],
                    extra_compile_args={
                        "cxx": ["-O3", "-std=c++17"],
                        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
                    },
                )
            )
        from torch.utils.cpp_extension import BuildExtension

        cmdclass = {"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)}
    except ImportError:
        cmdclass = {}
else:
    cmdclass = {}

setup(
    name="ant_ray_amd",
    version="0.1.0",
    description="MI355X-native distributed actor/task runtime with Ray-compatible APIs",
    packages=["ant_ray_amd"],
    ext_modules=ext_modules,
    cmdclass=cmdclass if "cmdclass" in dir() else {},
)
