"""Replica actor: hosts one copy of a deployment's user callable.

Role parity: reference python/ray/serve/_private/replica.py (2,201 LoC —
UserCallableWrapper, request handling, health checks). Async-concurrency:
the actor runs with max_concurrency = max_ongoing_requests; async user
callables interleave on the actor's event loop (our runtime's async-actor
fibers), sync callables run on the thread pool.
"""
from __future__ import annotations

import asyncio
import inspect
import pickle
from typing import Any, Dict, Optional


class HTTPRequestData:
    """Picklable HTTP request (the proxy can't ship a live ASGI scope)."""

    def __init__(self, method: str, path: str, query_string: bytes,
                 headers: list, body: bytes, route_prefix: str = "/"):
        self.method = method
        self.path = path
        self.query_string = query_string
        self.headers = headers
        self.body = body
        self.route_prefix = route_prefix

    def to_starlette(self):
        from starlette.requests import Request

        scope = {
            "type": "http",
            "method": self.method,
            "path": self.path,
            "raw_path": self.path.encode(),
            "query_string": self.query_string,
            "headers": [(k.encode() if isinstance(k, str) else k,
                         v.encode() if isinstance(v, str) else v)
                        for k, v in self.headers],
            "root_path": "",
        }
        body = self.body

        async def receive():
            return {"type": "http.request", "body": body, "more_body": False}

        return Request(scope, receive)


class ReplicaContext:
    """Runtime info for the current replica (parity: reference
    serve/context.py:37 ReplicaContext — app_name/deployment/replica_tag/
    servable_object/rank/world_size). Available inside a replica via
    serve.get_replica_context()."""

    def __init__(self, app_name, deployment, replica_tag, servable_object,
                 rank, world_size):
        self.app_name = app_name
        self.deployment = deployment
        self.replica_tag = replica_tag
        self.servable_object = servable_object
        self.rank = rank
        self.world_size = world_size

    def __repr__(self):
        return (f"ReplicaContext(app={self.app_name!r}, "
                f"deployment={self.deployment!r}, tag={self.replica_tag!r}, "
                f"rank={self.rank}/{self.world_size})")


_replica_context = None  # set once per replica actor process


def get_replica_context() -> ReplicaContext:
    if _replica_context is None:
        raise RuntimeError(
            "get_replica_context() may only be called from within a Ray "
            "Serve replica (deployment constructor or request handler).")
    return _replica_context


class Replica:
    """The actor. Created by the controller with the deployment's pickled
    callable + init args (inner DeploymentHandles arrive ready to use)."""

    def __init__(self, callable_bytes: bytes, init_args, init_kwargs,
                 user_config=None, context=None):
        global _replica_context
        target = pickle.loads(callable_bytes)
        if context:
            _replica_context = ReplicaContext(
                context.get("app_name", "default"),
                context.get("deployment", ""),
                context.get("replica_tag", ""),
                None, context.get("rank", 0), context.get("world_size", 1))
        self._is_function = inspect.isfunction(target)
        if self._is_function:
            self._callable = target
        else:
            self._callable = target(*init_args, **(init_kwargs or {}))
            if user_config is not None and hasattr(self._callable,
                                                   "reconfigure"):
                self._callable.reconfigure(user_config)
        if _replica_context is not None:
            _replica_context.servable_object = self._callable
        self._num_ongoing = 0

    def reconfigure(self, user_config):
        if hasattr(self._callable, "reconfigure"):
            self._callable.reconfigure(user_config)
        return True

    async def check_health(self):
        fn = getattr(self._callable, "check_health", None)
        if fn is not None:
            r = fn()
            if inspect.iscoroutine(r):
                # run_until_complete would blow up inside the actor's
                # already-running loop; await instead
                return await r
            return r if r is not None else True
        return True

    def num_ongoing_requests(self) -> int:
        return self._num_ongoing

    def handle_request_streaming(self, method_name: Optional[str], args,
                                 kwargs):
        """Streaming entry: the user callable is a (sync or async)
        generator; each produced chunk rides the runtime's streaming
        return path (num_returns='streaming' actor method) so the proxy /
        caller consumes chunks as they are generated (parity: reference
        Serve response streaming over ObjectRefGenerator)."""
        self._num_ongoing += 1
        try:
            if self._is_function:
                fn = self._callable
            else:
                fn = getattr(self._callable, method_name or "__call__")
            args = tuple(
                a.to_starlette() if isinstance(a, HTTPRequestData) else a
                for a in args
            )
            out = fn(*args, **(kwargs or {}))
            if hasattr(out, "__anext__"):
                loop = asyncio.new_event_loop()
                try:
                    while True:
                        try:
                            yield loop.run_until_complete(out.__anext__())
                        except StopAsyncIteration:
                            break
                finally:
                    loop.close()
            else:
                for chunk in out:
                    yield chunk
        finally:
            self._num_ongoing -= 1

    async def handle_request(self, method_name: Optional[str], args, kwargs):
        """Entry for handle calls AND HTTP (args[0] is HTTPRequestData)."""
        self._num_ongoing += 1
        try:
            if self._is_function:
                fn = self._callable
            else:
                fn = getattr(self._callable, method_name or "__call__")
            args = tuple(
                a.to_starlette() if isinstance(a, HTTPRequestData) else a
                for a in args
            )
            if inspect.iscoroutinefunction(fn):
                out = await fn(*args, **(kwargs or {}))
            else:
                # sync callables run on a thread pool so one blocking
                # handler never stalls the replica's event loop (health
                # probes, concurrent requests) — parity with the
                # reference's sync-handler thread execution
                import functools

                out = await asyncio.get_event_loop().run_in_executor(
                    None, functools.partial(fn, *args, **(kwargs or {})))
                if inspect.iscoroutine(out):
                    out = await out
            return _encode_response(out)
        finally:
            self._num_ongoing -= 1


def _encode_response(out: Any):
    """Starlette Responses are flattened to a picklable triple; everything
    else passes through (handle-to-handle calls see raw python values)."""
    try:
        from starlette.responses import Response
    except ImportError:
        return out
    if isinstance(out, Response):
        return _HTTPResponseData(out.status_code, list(out.raw_headers),
                                 out.body)
    return out


class _HTTPResponseData:
    def __init__(self, status: int, raw_headers, body: bytes):
        self.status = status
        self.raw_headers = raw_headers
        self.body = body


Replica.handle_request_streaming.__ray_method_opts__ = {
    "num_returns": "streaming"}
