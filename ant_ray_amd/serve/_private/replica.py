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


class Replica:
    """The actor. Created by the controller with the deployment's pickled
    callable + init args (inner DeploymentHandles arrive ready to use)."""

    def __init__(self, callable_bytes: bytes, init_args, init_kwargs,
                 user_config=None):
        target = pickle.loads(callable_bytes)
        self._is_function = inspect.isfunction(target)
        if self._is_function:
            self._callable = target
        else:
            self._callable = target(*init_args, **(init_kwargs or {}))
            if user_config is not None and hasattr(self._callable,
                                                   "reconfigure"):
                self._callable.reconfigure(user_config)
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
