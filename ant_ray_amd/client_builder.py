"""ray.client(...) builder API.

Role parity: reference python/ray/client_builder.py:89 (ClientBuilder:
.env/.namespace/.connect returning a ClientContext usable as a context
manager) and :348 (ray.client entry point). Here a "client" connection is
this package's ray-client-lite driver mode (`ray.init(address="ray://…")`,
see tests/test_ray_client.py) — there is no separate gRPC proxy process to
speak to, so connect() is a thin, validated ray.init call.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, Optional


@dataclass
class ClientContext:
    dashboard_url: Optional[str]
    python_version: str
    ray_version: str
    ray_commit: str

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.disconnect()

    def disconnect(self):
        import ant_ray_amd as ray

        if ray.is_initialized():
            ray.shutdown()


class ClientBuilder:
    def __init__(self, address: Optional[str]):
        self.address = address
        self._namespace: Optional[str] = None
        self._runtime_env: Optional[Dict[str, Any]] = None
        self._connected = False

    def env(self, env: Dict[str, Any]) -> "ClientBuilder":
        """Set the runtime environment for the session."""
        self._runtime_env = env
        return self

    def namespace(self, namespace: str) -> "ClientBuilder":
        """Set the namespace for the session."""
        self._namespace = namespace
        return self

    def connect(self) -> ClientContext:
        import platform

        import ant_ray_amd as ray

        if self._connected:
            raise RuntimeError("connect() called twice on this builder")
        self._connected = True
        addr = self.address
        if addr is not None and not addr.startswith("ray://"):
            addr = f"ray://{addr}"
        ray.init(address=addr, namespace=self._namespace,
                 runtime_env=self._runtime_env)
        return ClientContext(
            dashboard_url=None,
            python_version=platform.python_version(),
            ray_version=ray.__version__,
            ray_commit="unknown",
        )


def client(address: Optional[str] = None) -> ClientBuilder:
    """Build a client connection: ``ray.client("host:port").connect()``.

    With ``address=None`` connects to (or starts) a local cluster, like
    the reference's _LocalClientBuilder.
    """
    return ClientBuilder(address)
