"""Build the HIP/CDNA4 extensions in-tree (no JIT cache: the .so must travel
with the repo snapshot to the GPU box).

Pipeline:
  kernels/*.hip --hipcc --offload-arch=gfx950--> .o   (device code)
  hip_ops.cpp  --hipcc + torch includes-->       .o   (host binding)
  link --> ant_ray_amd/_hip_ops.<abi>.so

Incremental: objects are rebuilt only when sources are newer.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

CSRC = os.path.dirname(os.path.abspath(__file__))
PKG = os.path.dirname(CSRC)
BUILD = os.path.join(CSRC, "_build")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNELS = ["norms.hip", "rope.hip", "elementwise.hip", "cross_entropy.hip",
           "adamw.hip", "attention.hip", "attention_bwd.hip",
           "attention_decode.hip", "data_transform.hip"]


def _newer(src, obj):
    if not os.path.exists(obj):
        return True
    deps = [src, os.path.join(CSRC, "kernels", "common.hip.h")]
    return any(os.path.getmtime(d) > os.path.getmtime(obj) for d in deps if os.path.exists(d))


def _run(cmd):
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(
            f"command failed: {' '.join(cmd)}\n{r.stdout[-3000:]}\n{r.stderr[-3000:]}"
        )
    return r


def build(verbose=True):
    os.makedirs(BUILD, exist_ok=True)
    import torch

    tdir = os.path.dirname(torch.__file__)
    objs = []
    for k in KERNELS:
        src = os.path.join(CSRC, "kernels", k)
        obj = os.path.join(BUILD, k.replace(".hip", ".o"))
        objs.append(obj)
        if _newer(src, obj):
            if verbose:
                print(f"[hipcc] {k}")
            _run([HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
                  "-ffast-math", "-c", src, "-o", obj])

    bind_src = os.path.join(CSRC, "hip_ops.cpp")
    bind_obj = os.path.join(BUILD, "hip_ops.o")
    ext_suffix = sysconfig.get_config_var("EXT_SUFFIX")
    out_so = os.path.join(PKG, "_hip_ops" + ext_suffix)
    if _newer(bind_src, bind_obj):
        if verbose:
            print("[hipcc] hip_ops.cpp (torch binding)")
        py_inc = sysconfig.get_paths()["include"]
        _run([HIPCC, "-O2", "-std=c++17", "-fPIC", "-c", bind_src, "-o", bind_obj,
              f"-I{tdir}/include", f"-I{tdir}/include/torch/csrc/api/include",
              f"-I{py_inc}",
              "-D__HIP_PLATFORM_AMD__", "-DUSE_ROCM",
              "-DTORCH_EXTENSION_NAME=_hip_ops",
              "-D_GLIBCXX_USE_CXX11_ABI=1"])
    if _newer(bind_obj, out_so) or any(_newer(o, out_so) for o in objs):
        if verbose:
            print("[link] _hip_ops")
        _run([HIPCC, "-shared", bind_obj, *objs, "-o", out_so,
              f"-L{tdir}/lib", "-ltorch", "-ltorch_cpu", "-ltorch_python",
              "-lc10", "-ltorch_hip", "-lc10_hip", "-lamdhip64",
              f"-Wl,-rpath,{tdir}/lib"])

    # gpu_ipc extension (host-only hip runtime calls + torch)
    ipc_src = os.path.join(CSRC, "gpu_ipc.cpp")
    ipc_obj = os.path.join(BUILD, "gpu_ipc.o")
    ipc_so = os.path.join(PKG, "_gpu_ipc" + ext_suffix)
    if _newer(ipc_src, ipc_obj):
        if verbose:
            print("[hipcc] gpu_ipc.cpp")
        py_inc = sysconfig.get_paths()["include"]
        _run([HIPCC, "-O2", "-std=c++17", "-fPIC", "-c", ipc_src, "-o", ipc_obj,
              f"-I{tdir}/include", f"-I{tdir}/include/torch/csrc/api/include",
              f"-I{py_inc}",
              "-D__HIP_PLATFORM_AMD__", "-DUSE_ROCM",
              "-DTORCH_EXTENSION_NAME=_gpu_ipc",
              "-D_GLIBCXX_USE_CXX11_ABI=1"])
    if _newer(ipc_obj, ipc_so):
        if verbose:
            print("[link] _gpu_ipc")
        _run([HIPCC, "-shared", ipc_obj, "-o", ipc_so,
              f"-L{tdir}/lib", "-ltorch", "-ltorch_cpu", "-ltorch_python",
              "-lc10", "-ltorch_hip", "-lc10_hip", "-lamdhip64",
              f"-Wl,-rpath,{tdir}/lib"])
    return out_so


if __name__ == "__main__":
    print(build())
