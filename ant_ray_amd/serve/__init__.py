"""ant_ray_amd.serve — Ray Serve parity: scalable model serving on actors.

Role parity: reference python/ray/serve/ (~102k LoC).
Surface: @serve.deployment / .bind() / serve.run / serve.start /
serve.shutdown / serve.delete / serve.status / get_app_handle /
get_deployment_handle / @serve.batch, DeploymentHandle composition.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional, Union

from ant_ray_amd.serve._private.common import (
    CONTROLLER_NAME,
    DEFAULT_APP_NAME,
    AutoscalingConfig,
)
from ant_ray_amd.serve.batching import batch
from ant_ray_amd.serve.multiplex import get_multiplexed_model_id, multiplexed
from ant_ray_amd.serve.handle import DeploymentHandle, DeploymentResponse

PROXY_NAME = "SERVE_PROXY_ACTOR"

__all__ = [
    "Application", "AutoscalingConfig", "Deployment", "DeploymentHandle",
    "DeploymentResponse", "HTTPOptions", "RunTarget", "batch", "delete",
    "deployment", "get_app_handle", "get_deployment_handle",
    "get_multiplexed_model_id", "get_replica_context", "ingress",
    "multiplexed", "run", "run_many", "shutdown", "shutdown_async",
    "start", "status",
]

from ant_ray_amd.serve._private.replica import (  # noqa: E402
    get_replica_context,
)


@dataclass
class HTTPOptions:
    """HTTP proxy options (parity: reference serve/config.py:620).
    Accepted by serve.start(http_options=...) as an instance or dict."""

    host: str = "127.0.0.1"
    port: int = 8000
    root_path: str = ""
    request_timeout_s: Optional[float] = None
    keep_alive_timeout_s: float = 5.0
    location: str = "HeadOnly"


@dataclass
class Application:
    """A deployment bound to its constructor args (possibly nested)."""

    deployment: "Deployment"
    args: tuple = ()
    kwargs: dict = field(default_factory=dict)


class Deployment:
    def __init__(self, target: Union[Callable, type], name: str,
                 num_replicas: int = 1, ray_actor_options: Optional[dict] = None,
                 max_ongoing_requests: int = 100, user_config: Any = None,
                 autoscaling_config: Optional[Union[dict, AutoscalingConfig]] = None,
                 health_check_period_s: float = 10.0,
                 graceful_shutdown_timeout_s: float = 20.0, **_extra):
        self._target = target
        self.name = name
        self.num_replicas = num_replicas
        self.ray_actor_options = ray_actor_options or {}
        self.max_ongoing_requests = max_ongoing_requests
        self.user_config = user_config
        self.autoscaling_config = autoscaling_config
        self.health_check_period_s = health_check_period_s
        self.graceful_shutdown_timeout_s = graceful_shutdown_timeout_s

    def options(self, **kwargs) -> "Deployment":
        merged = dict(
            name=self.name, num_replicas=self.num_replicas,
            ray_actor_options=self.ray_actor_options,
            max_ongoing_requests=self.max_ongoing_requests,
            user_config=self.user_config,
            autoscaling_config=self.autoscaling_config,
            health_check_period_s=self.health_check_period_s,
            graceful_shutdown_timeout_s=self.graceful_shutdown_timeout_s,
        )
        merged.update(kwargs)
        return Deployment(self._target, **merged)

    def bind(self, *args, **kwargs) -> Application:
        return Application(self, args, kwargs)

    def __call__(self, *a, **kw):
        raise RuntimeError(
            "deployments are not callable directly; use .bind() + serve.run "
            "then handle.remote()"
        )

    def __repr__(self):
        return f"Deployment(name={self.name!r})"


def deployment(_target=None, **kwargs):
    """@serve.deployment decorator (parity serve/api.py:321)."""

    def wrap(target):
        name = kwargs.pop("name", None) or target.__name__
        return Deployment(target, name=name, **kwargs)

    if _target is not None:
        return wrap(_target)
    return wrap


def ingress(asgi_app):
    """@serve.ingress(fastapi_app_or_factory): class decorator mounting the
    deployment's methods behind a FastAPI/starlette app (parity
    serve/api.py ingress). A zero-arg FACTORY may be passed instead of an
    app: it is called lazily inside the replica — required when the app
    is built in a function scope (a function-local FastAPI app pickles by
    VALUE, and starlette State's __getattr__ recurses infinitely during
    unpickling)."""

    def wrap(cls):
        class ASGIWrapped(cls):
            __name__ = cls.__name__

            async def __call__(self, request):
                app = getattr(self, "_resolved_asgi_app", None)
                if app is None:
                    app = asgi_app
                    if callable(app) and not hasattr(app, "router"):
                        app = app()  # factory
                    self._resolved_asgi_app = app
                # run one request through the ASGI app
                scope = dict(request.scope)
                body = await request.body()
                scope["app_root_path"] = ""
                messages = [{"type": "http.request", "body": body,
                             "more_body": False}]
                sent: List[dict] = []

                async def receive():
                    return messages.pop(0) if messages else {
                        "type": "http.disconnect"}

                async def send(msg):
                    sent.append(msg)

                # make `self` reachable from route functions via app state
                app.state.serve_self = self
                await app(scope, receive, send)
                status = 200
                headers: List = []
                chunks = []
                for m in sent:
                    if m["type"] == "http.response.start":
                        status = m["status"]
                        headers = m.get("headers", [])
                    elif m["type"] == "http.response.body":
                        chunks.append(m.get("body", b""))
                from starlette.responses import Response

                return Response(
                    content=b"".join(chunks), status_code=status,
                    headers={k.decode(): v.decode() for k, v in headers},
                )

        return ASGIWrapped

    return wrap


# ------------------------------------------------------------- lifecycle


def _get_or_create_controller(http_host="127.0.0.1", http_port=8000):
    import ant_ray_amd as ray

    from ant_ray_amd.serve._private.controller import ServeController

    if not ray.is_initialized():
        ray.init()
    try:
        return ray.get_actor(CONTROLLER_NAME)
    except Exception:
        pass
    Controller = ray.remote(ServeController)
    c = Controller.options(
        name=CONTROLLER_NAME, lifetime="detached", num_cpus=0,
        max_concurrency=1000, max_restarts=-1,
    ).remote(http_port, http_host)
    return c


def start(detached: bool = True, http_options=None,
          grpc_options: Optional[dict] = None, **_):
    """Start Serve system actors (controller + HTTP proxy)."""
    import ant_ray_amd as ray

    if isinstance(http_options, HTTPOptions):
        http_options = http_options.__dict__
    http_options = http_options or {}
    host = http_options.get("host", "127.0.0.1")
    port = http_options.get("port", 8000)
    controller = _get_or_create_controller(host, port)
    try:
        proxy = ray.get_actor(PROXY_NAME)
    except Exception:
        from ant_ray_amd.serve._private.proxy import HTTPProxy

        Proxy = ray.remote(HTTPProxy)
        proxy = Proxy.options(
            name=PROXY_NAME, lifetime="detached", num_cpus=0,
            max_concurrency=1000, max_restarts=-1,
        ).remote(host, port, http_options.get("request_timeout_s"))
        ray.get(proxy.ready.remote(), timeout=60)
    if grpc_options:
        from ant_ray_amd.serve._private.grpc_proxy import start_grpc_proxy

        start_grpc_proxy(grpc_options.get("host", "127.0.0.1"),
                         grpc_options.get("port", 9000))
    return controller


def _collect_deployments(app: Application, out: Dict[str, dict],
                         app_name: str):
    """Topological flatten: nested bound apps become handles."""
    from ant_ray_amd._private.serialization import dumps_by_value as _dumps

    def resolve(v):
        if isinstance(v, Application):
            _collect_deployments(v, out, app_name)
            return DeploymentHandle(v.deployment.name, app_name)
        return v

    d = app.deployment
    init_args = tuple(resolve(a) for a in app.args)
    init_kwargs = {k: resolve(v) for k, v in app.kwargs.items()}
    auto = d.autoscaling_config
    if isinstance(auto, AutoscalingConfig):
        auto = auto.__dict__
    import inspect as _inspect

    t = d._target
    call = t if (_inspect.isfunction(t) or _inspect.ismethod(t)) \
        else getattr(t, "__call__", None)
    is_streaming = bool(call) and (
        _inspect.isgeneratorfunction(call)
        or _inspect.isasyncgenfunction(call))
    out[d.name] = {
        "name": d.name,
        "callable_bytes": _dumps(d._target),
        "init_args": init_args,
        "init_kwargs": init_kwargs,
        "num_replicas": d.num_replicas,
        "max_ongoing_requests": d.max_ongoing_requests,
        "ray_actor_options": d.ray_actor_options,
        "autoscaling_config": auto,
        "user_config": d.user_config,
        "is_streaming": is_streaming,
        "graceful_shutdown_timeout_s": d.graceful_shutdown_timeout_s,
        "health_check_period_s": d.health_check_period_s,
    }


def run(target: Application, *, name: str = DEFAULT_APP_NAME,
        route_prefix: str = "/", blocking: bool = False,
        _local_testing_mode: bool = False, **_) -> DeploymentHandle:
    """Deploy an application; returns the ingress DeploymentHandle
    (parity serve/api.py:686)."""
    import ant_ray_amd as ray

    if not isinstance(target, Application):
        raise TypeError("serve.run expects a bound deployment "
                        "(Deployment.bind(...))")
    controller = start()
    deployments: Dict[str, dict] = {}
    _collect_deployments(target, deployments, name)
    ingress_name = target.deployment.name
    ray.get(controller.deploy_application.remote(
        name, route_prefix, list(deployments.values()), ingress_name,
    ), timeout=300)
    handle = DeploymentHandle(ingress_name, name)
    if blocking:
        try:
            while True:
                time.sleep(1)
        except KeyboardInterrupt:
            pass
    return handle


@dataclass(frozen=True)
class RunTarget:
    """One application for serve.run_many (parity serve/api.py:520)."""

    target: Application
    name: str = DEFAULT_APP_NAME
    route_prefix: Optional[str] = "/"
    logging_config: Optional[dict] = None


def run_many(targets, blocking: bool = False, **_) -> List[DeploymentHandle]:
    """Deploy several applications; returns their ingress handles
    (parity serve/api.py:645)."""
    if not targets:
        raise ValueError("No applications provided.")
    handles = []
    for t in targets:
        if not t.name:
            raise ValueError("Application name must be a non-empty string.")
        handles.append(run(t.target, name=t.name,
                           route_prefix=t.route_prefix or f"/{t.name}"))
    if blocking:
        try:
            while True:
                time.sleep(1)
        except KeyboardInterrupt:
            pass
    return handles


_run = run
_run_many = run_many


async def shutdown_async():
    """Async serve.shutdown (parity serve/api.py:130): awaitable from an
    async context (e.g. inside a deployment), no blocking ray.get."""
    import asyncio

    await asyncio.get_event_loop().run_in_executor(None, shutdown)


def delete(name: str, _blocking: bool = True):
    import ant_ray_amd as ray

    try:
        controller = ray.get_actor(CONTROLLER_NAME)
    except Exception:
        return
    ray.get(controller.delete_application.remote(name), timeout=120)


def status() -> dict:
    import ant_ray_amd as ray

    try:
        controller = ray.get_actor(CONTROLLER_NAME)
    except Exception:
        return {"applications": {}}
    return {"applications": ray.get(controller.list_applications.remote(),
                                    timeout=60)}


def shutdown():
    import ant_ray_amd as ray

    try:
        controller = ray.get_actor(CONTROLLER_NAME)
    except Exception:
        return
    try:
        ray.get(controller.graceful_shutdown.remote(), timeout=120)
    except Exception:
        pass
    for actor_name in (PROXY_NAME, CONTROLLER_NAME):
        try:
            ray.kill(ray.get_actor(actor_name))
        except Exception:
            pass
    # wait for the names to actually free: a serve.start right after
    # shutdown must not resolve a dying controller (the kill is processed
    # asynchronously by the GCS)
    import time as _time

    deadline = _time.time() + 15
    for actor_name in (PROXY_NAME, CONTROLLER_NAME):
        while _time.time() < deadline:
            try:
                ray.get_actor(actor_name)
            except Exception:
                break
            _time.sleep(0.1)


def get_app_handle(name: str = DEFAULT_APP_NAME) -> DeploymentHandle:
    import ant_ray_amd as ray

    controller = ray.get_actor(CONTROLLER_NAME)
    cfg = ray.get(controller.get_app_config.remote(name), timeout=60)
    if cfg is None:
        raise RuntimeError(f"no application named {name!r}")
    return DeploymentHandle(cfg["ingress"], name)


def get_deployment_handle(deployment_name: str,
                          app_name: str = DEFAULT_APP_NAME) -> DeploymentHandle:
    return DeploymentHandle(deployment_name, app_name)
