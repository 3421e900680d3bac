"""ray.util.connect / disconnect — explicit client-style connection.

Role parity: reference python/ray/util/client_connect.py. Thin wrappers
over this package's ray-client-lite driver mode (see client_builder.py).
"""
from typing import Optional


def connect(conn_str: str, namespace: Optional[str] = None, **kwargs):
    import ant_ray_amd as ray

    addr = conn_str
    if addr and not addr.startswith("ray://"):
        addr = f"ray://{addr}"
    return ray.init(address=addr, namespace=namespace, **kwargs)


def disconnect():
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()
