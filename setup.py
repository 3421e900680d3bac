"""Build script for ant_ray_amd native extensions.

Two tiers:
  * `_shm_store` — plain pybind11/C++ (no torch, no GPU): the shared-memory
    object store used by the core runtime.
  * `_hip_ops`, `_gpu_ipc` — HIP/CDNA4 extensions: built ONLY by
    `ant_ray_amd/csrc/build.py` (hipcc --offload-arch=gfx950), called from
    `__graft_entry__.build()`. Not declared here.
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

# HIP extensions (_hip_ops, _gpu_ipc) are built EXCLUSIVELY by
# ant_ray_amd/csrc/build.py (hipcc --offload-arch=gfx950, explicit source
# list, incremental objects under csrc/_build). They are deliberately NOT
# setuptools extensions: an earlier stale source list here once clobbered
# the good in-tree .so with one missing the attention kernels.

setup(
    name="ant_ray_amd",
    version="0.1.0",
    description="MI355X-native distributed actor/task runtime with Ray-compatible APIs",
    packages=["ant_ray_amd"],
    ext_modules=ext_modules,
    entry_points={
        "console_scripts": [
            "ray=ant_ray_amd.scripts.cli:main",
            "serve=ant_ray_amd.serve.scripts:main",
        ],
    },
)
