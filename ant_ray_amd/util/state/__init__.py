"""State API: programmatic cluster introspection.

Role parity: reference python/ray/util/state/ (list_actors/list_tasks/
list_nodes/... against GCS + agents; state_head.py aggregator). Here every
query goes straight to the GCS over the connected worker's channel.
"""
from __future__ import annotations

import os
from typing import Any, Dict, List, Optional


def _gcs_call(method: str, payload: Optional[dict] = None, timeout: float = 30):
    import ant_ray_amd as ray
    from ant_ray_amd._private.worker import global_worker

    if not ray.is_initialized():
        ray.init(ignore_reinit_error=True)
    cw = global_worker.core_worker
    return cw.io.run(cw.gcs.call(method, payload or {}, timeout=timeout),
                     timeout=timeout + 5)


def list_actors(filters=None, limit: int = 1000, **_) -> List[Dict[str, Any]]:
    out = []
    for a in _gcs_call("list_actors"):
        row = {
            "actor_id": a["actor_id"].hex() if isinstance(a.get("actor_id"), bytes) else a.get("actor_id"),
            "state": a.get("state"),
            "name": a.get("name") or "",
            "class_name": a.get("class_name") or "",
            "pid": a.get("pid"),
            "node_id": a.get("node_id").hex() if isinstance(a.get("node_id"), bytes) else a.get("node_id"),
        }
        if _match(row, filters):
            out.append(row)
    return out[:limit]


def list_tasks(filters=None, limit: int = 1000, **_) -> List[Dict[str, Any]]:
    evs = _gcs_call("list_task_events", {"limit": limit})
    out = [e for e in evs if _match(e, filters)]
    return out[:limit]


def list_nodes(filters=None, limit: int = 1000, **_) -> List[Dict[str, Any]]:
    out = []
    for n in _gcs_call("node_table"):
        row = dict(n)
        if isinstance(row.get("node_id"), bytes):
            row["node_id"] = row["node_id"].hex()
        if _match(row, filters):
            out.append(row)
    return out[:limit]


def list_cluster_events(source: str = None, limit: int = 1000, **_):
    """Structured lifecycle events (NODE/ACTOR REGISTERED/ALIVE/DEAD/...),
    parity: reference export events / RayEventRecorder."""
    return _gcs_call("list_cluster_events",
                     {"limit": limit, "source": source})


def list_workers(filters=None, limit: int = 1000, **_) -> List[Dict[str, Any]]:
    rows = [_hexify(w) for w in _gcs_call("list_workers")]
    return [r for r in rows if _match(r, filters)][:limit]


def list_jobs(filters=None, limit: int = 1000, **_) -> List[Dict[str, Any]]:
    rows = [_hexify(j) for j in _gcs_call("list_jobs")]
    return [r for r in rows if _match(r, filters)][:limit]


def list_placement_groups(filters=None, limit: int = 1000, **_):
    rows = []
    for pg in _gcs_call("list_placement_groups"):
        rows.append({
            "placement_group_id": pg["pg_id"].hex(),
            "state": pg.get("state"),
            "strategy": pg.get("strategy"),
            "bundles": pg.get("bundles"),
            "name": pg.get("name", ""),
        })
    return [r for r in rows if _match(r, filters)][:limit]


def list_objects(filters=None, limit: int = 1000, **_):
    """Summarized view: per-node object-store stats (the reference lists
    per-object entries from all core workers; our shm store keeps only
    aggregate counters)."""
    out = []
    for n in _gcs_call("node_table"):
        out.append({
            "node_id": n["node_id"].hex() if isinstance(n.get("node_id"), bytes) else n.get("node_id"),
            "object_store_memory_total": n.get("resources_total", {}).get("object_store_memory"),
            "object_store_memory_available": n.get("resources_available", {}).get("object_store_memory"),
        })
    return out[:limit]


def list_logs(node_id: str = None) -> List[str]:
    """Session log files (parity: ray.util.state.list_logs)."""
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    logs_dir = os.path.join(getattr(cw, "session_dir", "") or "", "logs")
    if not os.path.isdir(logs_dir):
        return []
    return sorted(os.listdir(logs_dir))


def get_log(filename: str, tail: int = 1000) -> str:
    """Tail of one session log file (parity: ray.util.state.get_log)."""
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    fp = os.path.join(getattr(cw, "session_dir", "") or "", "logs",
                      os.path.basename(filename))
    if not os.path.isfile(fp):
        raise FileNotFoundError(filename)
    with open(fp, errors="replace") as f:
        lines = f.readlines()
    return "".join(lines[-tail:])


def summarize_actors():
    from collections import Counter

    c = Counter(a.get("state") for a in list_actors(limit=20000))
    return [{"state": k, "count": v} for k, v in c.items()]


def summarize_objects():
    return store_stats()


def get_actor(actor_id: str) -> Optional[Dict[str, Any]]:
    """Single-actor view by hex id (parity: state.get_actor)."""
    for a in list_actors(limit=100000):
        if a["actor_id"] == actor_id:
            return a
    return None


def get_node(node_id: str) -> Optional[Dict[str, Any]]:
    for n in list_nodes(limit=100000):
        if n["node_id"] == node_id:
            return n
    return None


def get_placement_group(pg_id: str) -> Optional[Dict[str, Any]]:
    for g in list_placement_groups(limit=100000):
        if g["placement_group_id"] == pg_id:
            return g
    return None


def store_stats() -> List[Dict[str, Any]]:
    """Per-node shm object-store usage (backs `ray memory`)."""
    out = []
    for s in _gcs_call("store_stats"):
        out.append(_hexify(dict(s)))
    return out


def summarize_tasks():
    from collections import Counter

    c = Counter((e.get("name"), e.get("state")) for e in list_tasks(limit=20000))
    return [{"name": k[0], "state": k[1], "count": v} for k, v in c.items()]


def _hexify(d: dict) -> dict:
    return {k: (v.hex() if isinstance(v, bytes) else v) for k, v in d.items()}


def _match(row: dict, filters) -> bool:
    if not filters:
        return True
    for f in filters:
        key, op, val = f
        have = row.get(key)
        if op in ("=", "=="):
            if str(have) != str(val):
                return False
        elif op == "!=":
            if str(have) == str(val):
                return False
    return True


def get_timeline(limit: int = 20000) -> List[dict]:
    """Chrome-trace events from the GCS task-event buffer
    (parity: `ray timeline` / GcsTaskManager export)."""
    evs = _gcs_call("list_task_events", {"limit": limit})
    trace = []
    for e in evs:
        start = e.get("start_ts")
        end = e.get("end_ts")
        if start is None or end is None:
            continue
        trace.append({
            "name": e.get("name") or e.get("type"),
            "cat": e.get("type"),
            "ph": "X",
            "ts": start * 1e6,
            "dur": max((end - start) * 1e6, 1),
            "pid": e.get("pid"),
            "tid": e.get("pid"),
            "args": {"task_id": e.get("task_id"), "state": e.get("state")},
        })
    return trace


def get_task(task_id: str) -> Optional[Dict[str, Any]]:
    """Single task-event view by hex id (parity: state.get_task)."""
    for t in list_tasks(limit=100000):
        tid = t.get("task_id")
        if isinstance(tid, bytes):
            tid = tid.hex()
        if tid == task_id:
            return t
    return None


def get_worker(worker_id: str) -> Optional[Dict[str, Any]]:
    for w in list_workers(limit=100000):
        if w.get("worker_id") == worker_id:
            return w
    return None


def get_job(job_id: str) -> Optional[Dict[str, Any]]:
    for j in list_jobs(limit=100000):
        if str(j.get("job_id")) == str(job_id) \
                or j.get("submission_id") == job_id:
            return j
    return None


def get_objects(object_id: str = None, **_) -> List[Dict[str, Any]]:
    """Object views, optionally filtered to one hex id (parity:
    state.get_objects)."""
    rows = list_objects(limit=100000)
    if object_id is None:
        return rows
    return [r for r in rows if r.get("object_id") == object_id]


def list_runtime_envs(filters=None, limit: int = 1000, **_):
    """Runtime envs currently attached to live workers (parity:
    state.list_runtime_envs; this build materializes envs at worker
    spawn, so the distinct env specs of live workers ARE the active
    set)."""
    envs = []
    seen = set()
    for w in list_workers(limit=100000):
        env = w.get("runtime_env")
        key = repr(env)
        if env and key not in seen:
            seen.add(key)
            envs.append({"runtime_env": env, "ref_cnt": 1,
                         "success": True})
        elif env and key in seen:
            for e in envs:
                if repr(e["runtime_env"]) == key:
                    e["ref_cnt"] += 1
    out = [e for e in envs if _match(e, filters)]
    return out[:limit]


class StateApiClient:
    """Programmatic façade over the state functions (parity: reference
    util/state/api.py StateApiClient — list/get by resource name)."""

    def __init__(self, address: str = None, **_):
        self.address = address

    _LIST = {
        "actors": "list_actors", "tasks": "list_tasks",
        "nodes": "list_nodes", "workers": "list_workers",
        "jobs": "list_jobs", "placement_groups": "list_placement_groups",
        "objects": "list_objects", "runtime_envs": "list_runtime_envs",
        "cluster_events": "list_cluster_events",
    }
    _GET = {"actors": "get_actor", "nodes": "get_node",
            "tasks": "get_task", "workers": "get_worker",
            "jobs": "get_job",
            "placement_groups": "get_placement_group"}

    def list(self, resource: str, options=None, raise_on_missing_output=True,
             **kwargs):
        fn = self._LIST.get(resource)
        if fn is None:
            raise ValueError(f"unknown resource {resource!r}; "
                             f"available: {sorted(self._LIST)}")
        opts = dict(kwargs)
        if options is not None:
            for k in ("filters", "limit"):
                v = getattr(options, k, None)
                if v is not None:
                    opts[k] = v
        return globals()[fn](**opts)

    def get(self, resource: str, id: str, **kwargs):
        fn = self._GET.get(resource)
        if fn is None:
            raise ValueError(f"unknown resource {resource!r}; "
                             f"available: {sorted(self._GET)}")
        return globals()[fn](id)
