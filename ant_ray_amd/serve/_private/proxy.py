"""HTTP proxy actor: uvicorn/starlette ASGI → DeploymentHandle calls.

Role parity: reference python/ray/serve/_private/proxy.py:709 (HTTPProxy,
per-node ProxyActor :1153). Request path: uvicorn → route match on app
route_prefix → ingress DeploymentHandle.remote(HTTPRequestData) → await →
ASGI response. Runs uvicorn on a daemon thread inside the actor.
"""
from __future__ import annotations

import asyncio
import json
import threading
import time
from typing import Any, Dict, Optional

from ant_ray_amd.serve._private.common import CONTROLLER_NAME
from ant_ray_amd.serve._private.replica import HTTPRequestData, _HTTPResponseData
from ant_ray_amd.serve.handle import DeploymentHandle

_ROUTE_REFRESH_S = 2.0


class HTTPProxy:
    def __init__(self, host: str = "127.0.0.1", port: int = 8000,
                 request_timeout_s: float = None):
        self.host = host
        self.port = port
        self.request_timeout_s = request_timeout_s  # None = no limit
        self._routes: Dict[str, str] = {}  # route_prefix -> app name
        self._handles: Dict[str, DeploymentHandle] = {}
        self._last_refresh = 0.0
        self._server = None
        self._started = threading.Event()
        t = threading.Thread(target=self._serve_thread, daemon=True,
                             name="serve-proxy")
        t.start()

    def ready(self) -> int:
        if not self._started.wait(timeout=30):
            raise RuntimeError("proxy failed to start")
        return self.port

    def _serve_thread(self):
        import uvicorn

        async def app(scope, receive, send):  # plain fn => detected as ASGI3
            await self._asgi(scope, receive, send)

        config = uvicorn.Config(app, host=self.host, port=self.port,
                                log_level="warning", loop="asyncio")
        self._server = uvicorn.Server(config)

        async def run():
            # signal readiness once the socket is bound
            asyncio.get_event_loop().call_later(0.2, self._started.set)
            await self._server.serve()

        asyncio.new_event_loop().run_until_complete(run())

    async def _refresh_routes(self, force: bool = False):
        now = time.monotonic()
        if not force and self._routes and now - self._last_refresh < _ROUTE_REFRESH_S:
            return
        import ant_ray_amd as ray

        controller = ray.get_actor(CONTROLLER_NAME)
        apps = await controller.list_applications.remote()
        routes = {}
        handles = {}
        streaming = {}
        for app, info in apps.items():
            prefix = info.get("route_prefix") or "/"
            ingress = info.get("ingress")
            if ingress:
                routes[prefix] = app
                handles[app] = self._handles.get(app) or DeploymentHandle(
                    ingress, app)
                streaming[app] = bool(info.get("ingress_streaming"))
        self._routes = routes
        self._handles = handles
        self._streaming = streaming
        self._last_refresh = now

    def _match(self, path: str) -> Optional[str]:
        best = None
        for prefix in self._routes:
            norm = prefix.rstrip("/") or ""
            if path == norm or path.startswith(norm + "/") or prefix == "/":
                if best is None or len(prefix) > len(best):
                    best = prefix
        return best

    async def _asgi(self, scope, receive, send):
        if scope["type"] == "lifespan":
            while True:
                msg = await receive()
                if msg["type"] == "lifespan.startup":
                    await send({"type": "lifespan.startup.complete"})
                elif msg["type"] == "lifespan.shutdown":
                    await send({"type": "lifespan.shutdown.complete"})
                    return
        if scope["type"] != "http":
            return
        path = scope["path"]
        if path == "/-/healthz":
            await _send_simple(send, 200, b"ok")
            return
        if path == "/-/routes":
            await self._refresh_routes()
            await _send_json(send, 200, self._routes)
            return
        try:
            await self._refresh_routes()
        except Exception as e:
            await _send_simple(send, 503, f"controller unavailable: {e}".encode())
            return
        prefix = self._match(path)
        if prefix is None:
            # the app may have been deployed within the refresh window
            await self._refresh_routes(force=True)
            prefix = self._match(path)
        if prefix is None:
            await _send_simple(send, 404, b"no app at this route")
            return
        app = self._routes[prefix]
        body = b""
        while True:
            msg = await receive()
            if msg["type"] == "http.request":
                body += msg.get("body", b"")
                if not msg.get("more_body"):
                    break
            else:
                break
        sub_path = path[len(prefix.rstrip("/")):] if prefix != "/" else path
        req = HTTPRequestData(
            method=scope["method"], path=sub_path or "/",
            query_string=scope.get("query_string", b""),
            headers=[(k.decode(), v.decode()) for k, v in scope.get("headers", [])],
            body=body, route_prefix=prefix,
        )
        if getattr(self, "_streaming", {}).get(app):
            await self._proxy_stream(send, app, req)
            return
        try:
            coro = self._handles[app].remote(req)
            if self.request_timeout_s:
                import asyncio as _a

                result = await _a.wait_for(
                    _aw_response(coro), timeout=self.request_timeout_s)
            else:
                result = await coro
        except TimeoutError:
            await _send_simple(send, 408,
                               b"request timed out (request_timeout_s)")
            return
        except Exception as e:
            import asyncio as _a

            if isinstance(e, _a.TimeoutError):
                await _send_simple(send, 408,
                                   b"request timed out (request_timeout_s)")
                return
            await _send_simple(send, 500, f"error: {e}".encode())
            return
        await _send_result(send, result)

    async def _proxy_stream(self, send, app: str, req):
        """Chunked transfer of a generator ingress: each yielded item is
        flushed to the client as it arrives from the replica."""
        import ant_ray_amd as ray

        handle = self._handles[app]
        router = handle._get_router()
        try:
            gen, idx = router.submit_stream(None, (req,), {})
        except Exception as e:
            await _send_simple(send, 503, f"error: {e}".encode())
            return
        started = False
        try:
            async for ref in gen:
                chunk = _encode_chunk(ray.get(ref))
                if not started:
                    started = True
                    await send({"type": "http.response.start", "status": 200,
                                "headers": [(b"content-type",
                                             b"text/plain; charset=utf-8"),
                                            (b"transfer-encoding",
                                             b"chunked")]})
                await send({"type": "http.response.body", "body": chunk,
                            "more_body": True})
            if not started:
                await send({"type": "http.response.start", "status": 200,
                            "headers": [(b"content-type", b"text/plain")]})
            await send({"type": "http.response.body", "body": b""})
        except Exception as e:
            if not started:
                await _send_simple(send, 500, f"error: {e}".encode())
            else:
                # mid-stream failure: terminate the body
                await send({"type": "http.response.body", "body": b""})
        finally:
            router.complete(idx)


async def _aw_response(resp):
    return await resp


def _encode_chunk(v) -> bytes:
    if isinstance(v, (bytes, bytearray)):
        return bytes(v)
    if isinstance(v, str):
        return v.encode()
    return (json.dumps(v) + "\n").encode()


async def _send_simple(send, status: int, body: bytes,
                       ctype: bytes = b"text/plain"):
    await send({"type": "http.response.start", "status": status,
                "headers": [(b"content-type", ctype)]})
    await send({"type": "http.response.body", "body": body})


async def _send_json(send, status: int, obj: Any):
    await _send_simple(send, status, json.dumps(obj).encode(),
                       b"application/json")


async def _send_result(send, result: Any):
    if isinstance(result, _HTTPResponseData):
        await send({"type": "http.response.start", "status": result.status,
                    "headers": result.raw_headers})
        await send({"type": "http.response.body", "body": result.body})
    elif isinstance(result, (bytes, bytearray)):
        await _send_simple(send, 200, bytes(result),
                           b"application/octet-stream")
    elif isinstance(result, str):
        await _send_simple(send, 200, result.encode())
    else:
        await _send_json(send, 200, result)
