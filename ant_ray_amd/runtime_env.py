"""ray.runtime_env public types (parity: python/ray/runtime_env/).

RuntimeEnv is a dict subclass so every internal path that accepts a
plain dict keeps working; it validates the known keys this build
implements (env_vars / working_dir / py_modules / config + profiler
plugins) and raises on unsupported cloud-only fields, pointing at the
air-gapped limitation instead of failing later.
"""
from __future__ import annotations

_SUPPORTED = {
    "env_vars", "working_dir", "py_modules", "config",
    "nsight", "rocprof_sys",  # profiler wrapper plugins
}
_OFFLINE_ONLY = {"pip", "conda", "uv", "container", "image_uri"}


class RuntimeEnvConfig(dict):
    def __init__(self, setup_timeout_seconds: int = 600,
                 eager_install: bool = True):
        super().__init__(setup_timeout_seconds=setup_timeout_seconds,
                         eager_install=eager_install)


class RuntimeEnv(dict):
    def __init__(self, **kwargs):
        for k in kwargs:
            if k in _OFFLINE_ONLY:
                raise ValueError(
                    f"runtime_env field {k!r} needs package/image downloads; "
                    "this air-gapped build supports: "
                    + ", ".join(sorted(_SUPPORTED)))
            if k not in _SUPPORTED:
                raise ValueError(
                    f"unknown runtime_env field {k!r}; supported: "
                    + ", ".join(sorted(_SUPPORTED)))
        if "env_vars" in kwargs:
            ev = kwargs["env_vars"]
            if not isinstance(ev, dict) or not all(
                    isinstance(k, str) and isinstance(v, str)
                    for k, v in ev.items()):
                raise TypeError("env_vars must be Dict[str, str]")
        super().__init__(**{k: v for k, v in kwargs.items() if v is not None})
