"""gRPC ingress proxy (parity: reference serve/_private/proxy.py:533
gRPCProxy — a per-node gRPC server routing RPCs to deployment handles).

No grpc_tools in this image, so instead of compiled protobuf stubs the
proxy registers a GENERIC service `ray.serve.UserApplicationService`
with raw-bytes unary methods:

    /ray.serve.UserApplicationService/<app_name>

The request bytes are handed to the app's ingress deployment (after a
pickle attempt — a pickled python object arrives as itself, anything
else as bytes) and the response is pickled back. Clients call it with
grpc.Channel.unary_unary and identity (de)serializers, so any gRPC
client in any language can reach a deployment without generated stubs.
"""
from __future__ import annotations

import logging
import pickle
import threading
from concurrent import futures
from typing import Optional

logger = logging.getLogger("antray.serve.grpc")


class GRPCProxy:
    """Detached actor: gRPC server + controller route refresh."""

    def __init__(self, host: str = "127.0.0.1", port: int = 9000):
        import grpc

        self._handles = {}
        self._lock = threading.Lock()
        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=32))

        proxy = self

        class _Generic(grpc.GenericRpcHandler):
            def service(self, handler_call_details):
                method = handler_call_details.method
                if not method.startswith("/ray.serve.UserApplicationService/"):
                    return None
                app = method.rsplit("/", 1)[1]

                def unary(request_bytes, context):
                    return proxy._dispatch(app, request_bytes, context)

                return grpc.unary_unary_rpc_method_handler(
                    unary,
                    request_deserializer=None,   # raw bytes in
                    response_serializer=None,    # raw bytes out
                )

        self._server.add_generic_rpc_handlers((_Generic(),))
        self._port = self._server.add_insecure_port(f"{host}:{port}")
        self._server.start()

    def ready(self) -> int:
        return self._port

    def _get_handle(self, app: str):
        with self._lock:
            h = self._handles.get(app)
        if h is not None:
            return h
        import ant_ray_amd as ray
        from ant_ray_amd.serve.handle import DeploymentHandle

        controller = ray.get_actor("SERVE_CONTROLLER_ACTOR")
        apps = ray.get(controller.list_applications.remote(), timeout=30)
        info = apps.get(app)
        if info is None or not info.get("ingress"):
            return None
        h = DeploymentHandle(info["ingress"], app)
        with self._lock:
            self._handles[app] = h
        return h

    def _dispatch(self, app: str, request_bytes: bytes, context):
        import grpc

        h = self._get_handle(app)
        if h is None:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"no serve application {app!r}")
        try:
            payload = pickle.loads(request_bytes)
        except Exception:
            payload = request_bytes
        try:
            out = h.remote(payload).result(timeout_s=60)
        except Exception as e:
            logger.exception("grpc dispatch to %s failed", app)
            context.abort(grpc.StatusCode.INTERNAL, str(e))
        return pickle.dumps(out)


def start_grpc_proxy(host: str = "127.0.0.1", port: int = 9000,
                     name: str = "SERVE_GRPC_PROXY_ACTOR"):
    """Start (or fetch) the gRPC ingress actor; returns its bound port."""
    import ant_ray_amd as ray

    try:
        proxy = ray.get_actor(name)
    except Exception:
        Proxy = ray.remote(GRPCProxy)
        proxy = Proxy.options(name=name, lifetime="detached", num_cpus=0,
                              max_concurrency=100,
                              max_restarts=-1).remote(host, port)
    return ray.get(proxy.ready.remote(), timeout=60)
