"""Flow Insight (lite): call-graph + object-flow observability events.

Role parity: ant fork python/ray/util/insight.py:1-787 (decorators emit
CallSubmit/CallBegin/CallEnd/ObjectGet/ObjectPut events to an insight
server; the dashboard proxies it at insight_head.py). Here events land in
the GCS task-event ring (source="insight") and `get_flow_graph()`
aggregates them into the caller->callee graph the Flow Insight UI renders.
"""
from __future__ import annotations

import functools
import time
from collections import defaultdict
from typing import Any, Dict, List, Optional


def _emit(event: Dict[str, Any]):
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    if cw is None or not cw.connected or getattr(cw, "gcs", None) is None:
        return
    event.setdefault("ts", time.time())
    event["source"] = "insight"
    try:
        cw.io.submit(cw.gcs.notify("task_events", {"events": [event]}))
    except Exception:
        pass


def record_call_submit(caller: str, callee: str):
    _emit({"type": "CallSubmit", "caller": caller, "callee": callee,
           "name": callee, "state": "SUBMITTED"})


def record_call_begin(callee: str):
    _emit({"type": "CallBegin", "name": callee, "state": "RUNNING",
           "start_ts": time.time()})


def record_call_end(callee: str, duration_s: float):
    _emit({"type": "CallEnd", "name": callee, "state": "FINISHED",
           "end_ts": time.time(), "duration_s": duration_s})


def record_object_put(name: str, nbytes: int):
    _emit({"type": "ObjectPut", "name": name, "nbytes": nbytes})


def record_object_get(name: str, nbytes: int):
    _emit({"type": "ObjectGet", "name": name, "nbytes": nbytes})


def insight_monitor(fn):
    """Decorator parity with the reference's @insight_monitor: wraps a
    function/method so begin/end events flow to the insight sink."""

    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        name = getattr(fn, "__qualname__", fn.__name__)
        record_call_begin(name)
        t0 = time.time()
        try:
            return fn(*args, **kwargs)
        finally:
            record_call_end(name, time.time() - t0)

    return wrapper


def get_flow_graph(limit: int = 20000) -> Dict[str, Any]:
    """Aggregate insight + task events into {nodes, edges} for the UI."""
    from ant_ray_amd.util.state import list_tasks

    events = list_tasks(limit=limit)
    nodes: Dict[str, Dict[str, Any]] = {}
    edges = defaultdict(int)
    for e in events:
        name = e.get("name") or "(anon)"
        n = nodes.setdefault(name, {"name": name, "calls": 0,
                                    "total_s": 0.0, "failed": 0})
        if e.get("type") == "CallSubmit":
            edges[(e.get("caller", "(driver)"), e.get("callee", name))] += 1
        elif e.get("state") in ("FINISHED", "FAILED"):
            n["calls"] += 1
            if e.get("state") == "FAILED":
                n["failed"] += 1
            if e.get("start_ts") and e.get("end_ts"):
                n["total_s"] += e["end_ts"] - e["start_ts"]
            elif e.get("duration_s"):
                n["total_s"] += e["duration_s"]
    return {
        "nodes": list(nodes.values()),
        "edges": [{"from": a, "to": b, "count": c}
                  for (a, b), c in edges.items()],
    }
