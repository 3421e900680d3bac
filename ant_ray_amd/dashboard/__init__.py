"""Dashboard-lite: REST head for cluster state + Prometheus metrics.

Role parity: reference python/ray/dashboard/ (head.py aiohttp server at
:8265 with module subprocesses: node/job/state/metrics/events heads). One
starlette/uvicorn app in a detached actor serving the same core endpoints:

  GET /api/cluster_status      nodes + resource totals
  GET /api/nodes /api/actors /api/tasks /api/jobs /api/placement_groups
  GET /api/virtual_clusters    (ant-fork virtual_cluster_head parity)
  GET /metrics                 Prometheus exposition (util.metrics)
  POST /api/jobs               submit a job {entrypoint, runtime_env}
  GET /api/jobs/{id}/logs
  GET /healthz

Start with ray.init(include_dashboard=True) or
dashboard.start_dashboard(port).
"""
from __future__ import annotations

import json
import os
import threading
from typing import Optional

DASHBOARD_ACTOR_NAME = "DASHBOARD_HEAD_ACTOR"


class DashboardHead:
    def __init__(self, host: str = "127.0.0.1", port: int = 8265):
        self.host = host
        self.port = port
        self._started = threading.Event()
        t = threading.Thread(target=self._serve, daemon=True, name="dashboard")
        t.start()

    def ready(self) -> int:
        if not self._started.wait(30):
            raise RuntimeError("dashboard failed to start")
        return self.port

    def _serve(self):
        import uvicorn

        async def app(scope, receive, send):
            if scope["type"] == "lifespan":
                while True:
                    msg = await receive()
                    if msg["type"] == "lifespan.startup":
                        await send({"type": "lifespan.startup.complete"})
                    else:
                        await send({"type": "lifespan.shutdown.complete"})
                        return
            if scope["type"] != "http":
                return
            try:
                status, ctype, body = await self._route(scope, receive)
            except Exception as e:
                status, ctype, body = 500, b"text/plain", str(e).encode()
            await send({"type": "http.response.start", "status": status,
                        "headers": [(b"content-type", ctype)]})
            await send({"type": "http.response.body", "body": body})

        config = uvicorn.Config(app, host=self.host, port=self.port,
                                log_level="warning", loop="asyncio")
        self._server = uvicorn.Server(config)
        import asyncio

        async def run():
            asyncio.get_event_loop().call_later(0.2, self._started.set)
            await self._server.serve()

        asyncio.new_event_loop().run_until_complete(run())

    async def _gcs(self, method, payload=None):
        from ant_ray_amd._private.worker import global_worker

        cw = global_worker.core_worker
        fut = cw.io.submit(cw.gcs.call(method, payload or {}, timeout=30))
        import asyncio

        return await asyncio.wrap_future(fut)

    async def _node_stats(self):
        """Latest per-node reporter samples from the GCS KV."""
        keys = (await self._gcs("kv_keys", {"ns": "node_stats",
                                            "prefix": b""}))["keys"]
        out = {}
        for k in keys:
            r = await self._gcs("kv_get", {"ns": "node_stats", "key": k})
            if r.get("value"):
                out[k.decode()] = json.loads(r["value"].decode())
        return out

    async def _route(self, scope, receive):
        path = scope["path"].rstrip("/")
        method = scope["method"]

        def js(obj, status=200):
            return status, b"application/json", json.dumps(
                obj, default=_js_default).encode()

        if path == "/healthz" or path == "/api/healthz":
            return 200, b"text/plain", b"ok"
        if path == "/api/version":
            import ant_ray_amd

            return js({"version": ant_ray_amd.__version__})
        if path == "/api/cluster_status":
            nodes = await self._gcs("node_table")
            total = await self._gcs("cluster_resources")
            return js({"nodes": nodes, "resources": total})
        if path == "/api/nodes":
            nodes = await self._gcs("node_table")
            stats = await self._node_stats()
            for n in nodes if isinstance(nodes, list) else []:
                nid = n.get("node_id")
                nid = nid.hex() if isinstance(nid, bytes) else str(nid)
                if nid in stats:
                    n["physical_stats"] = stats[nid]
            return js(nodes)
        if path == "/api/node_stats" or path == "/api/gpus":
            # per-node reporter samples (reference reporter_agent shape)
            stats = await self._node_stats()
            if path == "/api/gpus":
                return js({nid: s.get("gpus", [])
                           for nid, s in stats.items()})
            return js(stats)
        if path == "/api/actors":
            return js(await self._gcs("list_actors"))
        if path == "/api/tasks":
            return js(await self._gcs("list_task_events", {"limit": 1000}))
        if path == "/api/placement_groups":
            return js(await self._gcs("list_placement_groups"))
        if path == "/api/cluster_events":
            return js(await self._gcs("list_cluster_events", {"limit": 1000}))
        if path == "/api/memory":
            return js(await self._gcs("store_stats"))
        if path == "/api/logs" or path.startswith("/api/logs/"):
            # session log serving (reference dashboard log module:
            # log_agent.py file listing + tail)
            from ant_ray_amd._private.worker import global_worker

            cw = global_worker.core_worker
            logs_dir = os.path.join(
                getattr(cw, "session_dir", "") or "", "logs")
            if not os.path.isdir(logs_dir):
                return js([])
            if path == "/api/logs":
                return js(sorted(os.listdir(logs_dir)))
            name = os.path.basename(path.split("/api/logs/", 1)[1])
            fp = os.path.join(logs_dir, name)
            if not os.path.isfile(fp):
                return 404, b"text/plain", b"no such log"
            with open(fp, "rb") as f:
                f.seek(0, 2)
                size = f.tell()
                f.seek(max(0, size - 64 * 1024))
                return 200, b"text/plain", f.read()
        if path == "/api/serve/applications":
            # serve observability (reference dashboard serve head)
            try:
                from ant_ray_amd import serve as _serve

                return js(_serve.status())
            except Exception as e:
                return js({"error": str(e)}, 500)
        if path == "/api/virtual_clusters":
            return js(await self._gcs("list_virtual_clusters"))
        if path == "/api/display":
            # ray.show_in_dashboard messages (ns "dashboard_display")
            keys = (await self._gcs("kv_keys", {"ns": "dashboard_display",
                                                "prefix": b""}))["keys"]
            out = {}
            for k in keys:
                r = await self._gcs("kv_get", {"ns": "dashboard_display",
                                               "key": k})
                if r.get("value"):
                    out[k.decode()] = json.loads(r["value"].decode())
            return js(out)
        if path == "/metrics":
            from ant_ray_amd.util.metrics import prometheus_text

            keys = (await self._gcs("kv_keys", {"ns": "metrics",
                                                "prefix": b""}))["keys"]
            rows = []
            for k in keys:
                r = await self._gcs("kv_get", {"ns": "metrics", "key": k})
                if r.get("value"):
                    rows.append(json.loads(r["value"].decode()))
            return 200, b"text/plain; version=0.0.4", prometheus_text(rows).encode()
        if path == "/api/jobs" and method == "GET":
            from ant_ray_amd.job_submission import JobSubmissionClient

            return js(JobSubmissionClient().list_jobs())
        if path == "/api/jobs" and method == "POST":
            body = b""
            while True:
                msg = await receive()
                body += msg.get("body", b"")
                if not msg.get("more_body"):
                    break
            req = json.loads(body or b"{}")
            from ant_ray_amd.job_submission import JobSubmissionClient

            job_id = JobSubmissionClient().submit_job(
                entrypoint=req["entrypoint"],
                submission_id=req.get("submission_id"),
                runtime_env=req.get("runtime_env"))
            return js({"submission_id": job_id})
        if path.startswith("/api/jobs/") and path.endswith("/logs"):
            job_id = path.split("/")[3]
            from ant_ray_amd.job_submission import JobSubmissionClient

            return 200, b"text/plain", JobSubmissionClient().get_job_logs(
                job_id).encode()
        if path.startswith("/api/jobs/"):
            job_id = path.split("/")[3]
            from ant_ray_amd.job_submission import JobSubmissionClient

            info = JobSubmissionClient().get_job_info(job_id)
            return js(info or {"status": "PENDING"})
        return 404, b"text/plain", b"not found"


def _js_default(o):
    if isinstance(o, bytes):
        return o.hex()
    return str(o)


def start_dashboard(port: int = 8265, host: str = "127.0.0.1"):
    """Start (or fetch) the dashboard head actor; returns the bound port."""
    import ant_ray_amd as ray

    try:
        head = ray.get_actor(DASHBOARD_ACTOR_NAME)
    except Exception:
        Head = ray.remote(DashboardHead)
        head = Head.options(name=DASHBOARD_ACTOR_NAME, lifetime="detached",
                            num_cpus=0, max_concurrency=100).remote(host, port)
    return ray.get(head.ready.remote(), timeout=60)
