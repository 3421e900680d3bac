"""Application metrics: Counter / Gauge / Histogram.

Role parity: reference python/ray/util/metrics.py (Cython Metric backed by
OpenCensus, exported via the per-node dashboard agent to Prometheus). Here
values are recorded into the GCS KV (namespace "metrics") keyed by
(name, tags); the dashboard head serves them in Prometheus exposition
format at /metrics.
"""
from __future__ import annotations

import json
import threading
import time
from typing import Dict, List, Optional, Tuple


def _publish(name: str, kind: str, value, tags: Dict[str, str],
             extra: Optional[dict] = None):
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    if cw is None or not cw.connected or getattr(cw, "gcs", None) is None:
        return
    key = f"{name}|{json.dumps(tags, sort_keys=True)}".encode()
    payload = {"name": name, "kind": kind, "value": value, "tags": tags,
               "ts": time.time()}
    if extra:
        payload.update(extra)
    global _pub_seq
    _pub_seq += 1
    try:
        cw.io.submit(cw.gcs.call("kv_put", {
            "ns": "metrics", "key": key,
            "value": json.dumps(payload).encode(), "overwrite": True,
            # ordered last-writer-wins: fire-and-forget publishes may be
            # handled out of order server-side. The seq is scoped to THIS
            # client (seq_id): another worker publishing the same metric
            # starts its own sequence and is never judged against ours
            "seq": _pub_seq,
            "seq_id": cw.worker_id,
        }))
    except Exception:
        pass


_pub_seq = 0


class _Metric:
    def __init__(self, name: str, description: str = "",
                 tag_keys: Optional[Tuple[str, ...]] = None):
        if not name:
            raise ValueError("metric name is required")
        self._name = name
        self._description = description
        self._tag_keys = tuple(tag_keys or ())
        self._default_tags: Dict[str, str] = {}
        self._lock = threading.Lock()

    def set_default_tags(self, tags: Dict[str, str]):
        self._default_tags = dict(tags)
        return self

    def _tags(self, tags):
        out = dict(self._default_tags)
        out.update(tags or {})
        unknown = set(out) - set(self._tag_keys)
        if unknown:
            raise ValueError(f"unknown tag keys {unknown}; declared "
                             f"{self._tag_keys}")
        return out

    @property
    def info(self):
        return {"name": self._name, "description": self._description,
                "tag_keys": self._tag_keys}


class Counter(_Metric):
    def __init__(self, name, description="", tag_keys=None):
        super().__init__(name, description, tag_keys)
        self._values: Dict[str, float] = {}

    def inc(self, value: float = 1.0, tags: Optional[Dict[str, str]] = None):
        if value <= 0:
            raise ValueError("Counter.inc value must be positive")
        t = self._tags(tags)
        k = json.dumps(t, sort_keys=True)
        with self._lock:
            self._values[k] = self._values.get(k, 0.0) + value
            v = self._values[k]
        _publish(self._name, "counter", v, t)


class Gauge(_Metric):
    def set(self, value: float, tags: Optional[Dict[str, str]] = None):
        _publish(self._name, "gauge", float(value), self._tags(tags))


class Histogram(_Metric):
    def __init__(self, name, description="", boundaries: List[float] = None,
                 tag_keys=None):
        super().__init__(name, description, tag_keys)
        if not boundaries:
            raise ValueError("Histogram requires boundaries")
        self.boundaries = list(boundaries)
        self._counts: Dict[str, List[int]] = {}
        self._sums: Dict[str, float] = {}

    def observe(self, value: float, tags: Optional[Dict[str, str]] = None):
        t = self._tags(tags)
        k = json.dumps(t, sort_keys=True)
        with self._lock:
            counts = self._counts.setdefault(k, [0] * (len(self.boundaries) + 1))
            i = 0
            while i < len(self.boundaries) and value > self.boundaries[i]:
                i += 1
            counts[i] += 1
            self._sums[k] = self._sums.get(k, 0.0) + value
            snapshot = list(counts)
            total = self._sums[k]
        _publish(self._name, "histogram", total, t,
                 {"buckets": self.boundaries, "counts": snapshot})


def prometheus_text(metric_rows: List[dict]) -> str:
    """Render recorded metrics in Prometheus exposition format."""
    lines = []
    for m in metric_rows:
        labels = ",".join(f'{k}="{v}"' for k, v in sorted(m["tags"].items()))
        label_s = "{" + labels + "}" if labels else ""
        if m["kind"] == "histogram":
            acc = 0
            for b, c in zip(m["buckets"] + ["+Inf"], m["counts"]):
                acc += c
                lb = labels + ("," if labels else "") + f'le="{b}"'
                lines.append(f'{m["name"]}_bucket{{{lb}}} {acc}')
            lines.append(f'{m["name"]}_sum{label_s} {m["value"]}')
            lines.append(f'{m["name"]}_count{label_s} {acc}')
        else:
            lines.append(f'{m["name"]}{label_s} {m["value"]}')
    return "\n".join(lines) + "\n"
