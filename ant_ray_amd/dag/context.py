"""DAG global settings.

Role parity: reference python/ray/dag/context.py:38 (DAGContext —
RAY_CGRAPH_* env-overridable knobs; get_current singleton). The channel
implementation here reads buffer_size_bytes and the timeouts when a DAG
is compiled into mutable shm channels.
"""
import os
from dataclasses import dataclass

_default_context = None


def _env(name, default, cast):
    v = os.environ.get(f"RAY_CGRAPH_{name}")
    return cast(v) if v is not None else default


@dataclass
class DAGContext:
    submit_timeout: int = 10
    get_timeout: int = 10
    teardown_timeout: int = 30
    read_iteration_timeout: float = 0.1
    buffer_size_bytes: int = 1_000_000
    max_inflight_executions: int = 10
    max_buffered_results: int = 1000
    overlap_gpu_communication: bool = False

    def __post_init__(self):
        if self.read_iteration_timeout > self.get_timeout:
            raise ValueError(
                "read_iteration_timeout must be <= get_timeout")

    @staticmethod
    def get_current() -> "DAGContext":
        global _default_context
        if _default_context is None:
            _default_context = DAGContext(
                submit_timeout=_env("submit_timeout", 10, int),
                get_timeout=_env("get_timeout", 10, int),
                teardown_timeout=_env("teardown_timeout", 30, int),
                read_iteration_timeout=_env("read_iteration_timeout", 0.1,
                                            float),
                buffer_size_bytes=_env("buffer_size_bytes", 1_000_000, int),
                max_inflight_executions=_env("max_inflight_executions", 10,
                                             int),
                max_buffered_results=_env("max_buffered_results", 1000, int),
                overlap_gpu_communication=bool(
                    _env("overlap_gpu_communication", 0, int)),
            )
        return _default_context
