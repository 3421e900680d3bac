"""Tracing helper: OpenTelemetry-style spans around task/actor calls.

Role parity: reference python/ray/util/tracing/tracing_helper.py (lazy otel
import, function/actor wrappers installed by ray.init(_tracing_startup_hook)).
Offline image has no opentelemetry; spans fall back to the task-event ring
(visible in `ray timeline` / util.state.get_timeline). With opentelemetry
installed, real spans are emitted through the same API.
"""
from __future__ import annotations

import contextlib
import time
from typing import Optional

_tracer = None
_enabled = False


def setup_tracing(otlp_endpoint: Optional[str] = None):
    """Enable tracing (parity: _tracing_startup_hook). Uses otel when
    importable, else the built-in task-event sink."""
    global _tracer, _enabled
    _enabled = True
    try:
        from opentelemetry import trace
        from opentelemetry.sdk.trace import TracerProvider

        provider = TracerProvider()
        trace.set_tracer_provider(provider)
        _tracer = trace.get_tracer("ant_ray_amd")
    except ImportError:
        _tracer = None
    return _enabled


def is_tracing_enabled() -> bool:
    return _enabled


@contextlib.contextmanager
def span(name: str, attributes: Optional[dict] = None):
    """Trace span context manager; nests with otel when present."""
    if not _enabled:
        yield None
        return
    if _tracer is not None:
        with _tracer.start_as_current_span(name) as s:
            for k, v in (attributes or {}).items():
                s.set_attribute(k, v)
            yield s
        return
    t0 = time.time()
    try:
        yield None
    finally:
        from ant_ray_amd.util.insight import _emit

        _emit({"type": "span", "name": name, "state": "FINISHED",
               "start_ts": t0, "end_ts": time.time(),
               **({"attrs": attributes} if attributes else {})})
