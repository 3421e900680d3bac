"""Structured logging configuration for drivers and workers.

Role parity: reference python/ray/_private/ray_logging/logging_config.py
(LoggingConfig dataclass + DefaultLoggingConfigurator.configure) and the
CoreContextFilter that stamps runtime-context ids onto every record.
Passed as ``ray.init(logging_config=LoggingConfig(...))``; applied in the
driver immediately and re-applied in each spawned worker (the config is
tiny and JSON-serializable, so it rides the task/actor runtime env).
"""
from __future__ import annotations

import json
import logging
from dataclasses import dataclass, field
from typing import List

_STANDARD_ATTRS = (
    "name", "levelname", "pathname", "lineno", "threadName", "process",
    "funcName", "created",
)


class CoreContextFilter(logging.Filter):
    """Stamp job/worker/node/actor/task ids onto each record (best-effort:
    records emitted before ray.init simply omit them)."""

    def filter(self, record):
        try:
            from ant_ray_amd._private.worker import global_worker

            if global_worker.connected:
                cw = global_worker.core_worker
                jid = getattr(cw, "job_id", None)
                record.job_id = jid.hex() if isinstance(jid, bytes) else str(jid)
                record.worker_id = cw.worker_id.hex()[:16]
                node = getattr(cw, "node_id", None)
                if node:
                    record.node_id = node.hex()[:16]
                tid = getattr(cw, "current_task_id", None)
                if tid:
                    record.task_id = tid.hex()[:16]
                aid = getattr(cw, "actor_id", None)
                if aid:
                    record.actor_id = aid.hex()[:16]
        except Exception:
            pass
        return True


class TextFormatter(logging.Formatter):
    def __init__(self, additional_attrs: List[str]):
        super().__init__()
        self._attrs = list(additional_attrs)

    def format(self, record):
        ts = self.formatTime(record)
        parts = [f"{ts}\t{record.levelname}\t{record.name}:{record.lineno} -- "
                 f"{record.getMessage()}"]
        for a in ("job_id", "worker_id", "node_id", "actor_id", "task_id",
                  *self._attrs):
            v = getattr(record, a, None)
            if v is not None:
                parts.append(f"{a}={v}")
        out = " ".join(parts)
        if record.exc_info:
            out += "\n" + self.formatException(record.exc_info)
        return out


class JSONFormatter(logging.Formatter):
    def __init__(self, additional_attrs: List[str]):
        super().__init__()
        self._attrs = list(additional_attrs)

    def format(self, record):
        d = {
            "asctime": self.formatTime(record),
            "levelname": record.levelname,
            "message": record.getMessage(),
            "filename": record.filename,
            "lineno": record.lineno,
        }
        for a in ("job_id", "worker_id", "node_id", "actor_id", "task_id",
                  *self._attrs):
            v = getattr(record, a, None)
            if v is not None:
                d[a] = v
        if record.exc_info:
            d["exc_text"] = self.formatException(record.exc_info)
        return json.dumps(d)


@dataclass
class LoggingConfig:
    encoding: str = "TEXT"
    log_level: str = "INFO"
    additional_log_standard_attrs: List[str] = field(default_factory=list)

    def __post_init__(self):
        if self.encoding not in ("TEXT", "JSON"):
            raise ValueError(
                f"Invalid encoding {self.encoding!r}; supported: TEXT, JSON")
        for a in self.additional_log_standard_attrs:
            if a not in _STANDARD_ATTRS:
                raise ValueError(
                    f"Unknown standard attr {a!r}; supported: {_STANDARD_ATTRS}")

    def _apply(self):
        global _applied
        _applied = True
        fmt_cls = JSONFormatter if self.encoding == "JSON" else TextFormatter
        handler = logging.StreamHandler()
        handler.setLevel(self.log_level)
        handler.setFormatter(fmt_cls(self.additional_log_standard_attrs))
        handler.addFilter(CoreContextFilter())
        root = logging.getLogger()
        root.setLevel(self.log_level)
        root.addHandler(handler)
        ray_logger = logging.getLogger("ant_ray_amd")
        ray_logger.setLevel(self.log_level)
        for h in ray_logger.handlers[:]:
            ray_logger.removeHandler(h)
        ray_logger.addHandler(handler)
        ray_logger.propagate = False

    def _to_dict(self):
        return {"encoding": self.encoding, "log_level": self.log_level,
                "additional_log_standard_attrs":
                    list(self.additional_log_standard_attrs)}

    @classmethod
    def _from_dict(cls, d):
        return cls(**d)


_applied = False  # set once any config applies in this process
