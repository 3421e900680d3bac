"""Runtime environments: per-task/actor worker environment plugins.

Role parity: reference python/ray/_private/runtime_env/ (plugins: env_vars,
working_dir, py_modules, conda/pip/uv, containers, nsight,
rocprof_sys.py:20 — the AMD rocprof-sys wrapping of the worker command) and
the per-node runtime-env agent. Here the raylet applies plugins directly at
worker spawn (build_worker_spawn); a lease whose runtime_env has more than
env_vars gets a DEDICATED worker that dies on return (no pool pollution).

Supported keys:
  env_vars      {name: value}      — also applied to pooled workers at lease
  working_dir   path               — worker cwd + sys.path entry
  py_modules    [paths]            — prepended to PYTHONPATH
  pip / conda                      — offline image: validated but rejected
                                     with a clear error (no package index)
  rocprof_sys   {} or {out_dir}    — wrap worker in rocprof-sys-run / rocprofv3
  nsight        {...}              — NVIDIA-only: rejected on this platform
"""
from __future__ import annotations

import os
import shutil
from typing import Dict, List, Optional, Tuple


class RuntimeEnvSetupError(RuntimeError):
    pass


def validate(runtime_env: dict) -> None:
    known = {"env_vars", "working_dir", "py_modules", "pip", "conda", "uv",
             "rocprof_sys", "nsight", "container", "config"}
    unknown = set(runtime_env) - known
    if unknown:
        raise RuntimeEnvSetupError(f"unknown runtime_env keys: {unknown}")


def build_worker_spawn(cmd: List[str], env: Dict[str, str],
                       runtime_env: dict) -> Tuple[List[str], Dict[str, str],
                                                   Optional[str]]:
    """Apply runtime_env plugins to a worker spawn (cmd, env, cwd)."""
    validate(runtime_env)
    cwd = None
    # the worker must import ant_ray_amd regardless of its cwd
    import ant_ray_amd

    pkg_root = os.path.dirname(os.path.dirname(os.path.abspath(
        ant_ray_amd.__file__)))
    env["PYTHONPATH"] = pkg_root + os.pathsep + env.get("PYTHONPATH", "")
    ev = runtime_env.get("env_vars") or {}
    env.update({str(k): str(v) for k, v in ev.items()})

    wd = runtime_env.get("working_dir")
    if wd:
        if not os.path.isdir(wd):
            raise RuntimeEnvSetupError(f"working_dir {wd!r} does not exist")
        cwd = wd
        env["PYTHONPATH"] = wd + os.pathsep + env.get("PYTHONPATH", "")

    mods = runtime_env.get("py_modules") or []
    for m in mods:
        if not os.path.exists(m):
            raise RuntimeEnvSetupError(f"py_modules path {m!r} does not exist")
        env["PYTHONPATH"] = str(m) + os.pathsep + env.get("PYTHONPATH", "")

    if runtime_env.get("pip") or runtime_env.get("conda") or runtime_env.get("uv"):
        raise RuntimeEnvSetupError(
            "pip/conda/uv runtime envs need a package index; this deployment "
            "is air-gapped (install into the base image instead)")
    if runtime_env.get("nsight"):
        raise RuntimeEnvSetupError("nsight is NVIDIA-only; use rocprof_sys")

    prof = runtime_env.get("rocprof_sys")
    if prof is not None:
        # parity: runtime_env/rocprof_sys.py prefixes the worker command
        prof = prof if isinstance(prof, dict) else {}
        out_dir = prof.get("output_dir", "/tmp/antray/rocprof")
        os.makedirs(out_dir, exist_ok=True)
        wrapper = shutil.which("rocprof-sys-run")
        if wrapper:
            cmd = [wrapper, "--output", out_dir, "--"] + cmd
        else:
            rocprofv3 = shutil.which("rocprofv3") or "/opt/rocm/bin/rocprofv3"
            cmd = [rocprofv3, "--kernel-trace", "-d", out_dir, "--"] + cmd
    return cmd, env, cwd
