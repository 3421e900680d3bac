"""ray.util parity namespace."""
from typing import List, Optional

from ant_ray_amd.util.placement_group import (  # noqa: F401
    get_placement_group,
    placement_group,
    placement_group_table,
    remove_placement_group,
    get_current_placement_group,
)
from ant_ray_amd.util.client_connect import connect, disconnect  # noqa: F401
from ant_ray_amd.util.debug import (  # noqa: F401
    disable_log_once_globally,
    enable_periodic_logging,
    log_once,
)
from ant_ray_amd.util.helpers import as_completed, map_unordered  # noqa: F401
from ant_ray_amd.util.serialization import (  # noqa: F401
    deregister_serializer,
    register_serializer,
)


def get_node_ip_address(address: str = "8.8.8.8") -> str:
    """This node's primary IP (parity: _private/services.py:632). The
    single-node-class deployment binds everything to 127.0.0.1."""
    import socket

    try:
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.settimeout(0)
        s.connect((address, 80))
        ip = s.getsockname()[0]
        s.close()
        return ip
    except Exception:
        return "127.0.0.1"


def get_node_instance_id() -> str:
    """Cloud instance id of this node (parity: services.py:742); empty in
    this air-gapped deployment, falling back to the runtime node id."""
    try:
        import ant_ray_amd as ray

        return ray.get_runtime_context().get_node_id()
    except Exception:
        return ""


def list_named_actors(all_namespaces: bool = False) -> List:
    """Names of live named actors (parity: util/__init__.py
    list_named_actors): current namespace by default; with
    all_namespaces=True, dicts of {name, namespace} across all."""
    from ant_ray_amd import _check_connected
    from ant_ray_amd._private.worker import global_worker

    _check_connected()
    cw = global_worker.core_worker
    # names register under the namespace given at creation ("" unless the
    # actor passed one — same default get_actor resolves against)
    r = cw.io.run(cw.gcs.call("list_named_actors", {
        "all_namespaces": all_namespaces,
        "namespace": "",
    }, timeout=30), timeout=35)
    rows = r.get("actors", [])
    if all_namespaces:
        return [{"name": a["name"], "namespace": a["namespace"]} for a in rows]
    return [a["name"] for a in rows]


def __getattr__(name):
    import importlib

    if name in ("collective", "state", "queue", "metrics", "scheduling_strategies",
                "actor_pool", "accelerators", "iter", "debug", "helpers",
                "serialization", "multiprocessing"):
        return importlib.import_module(f"ant_ray_amd.util.{name}")
    if name == "pdb":
        return importlib.import_module("ant_ray_amd.util.rpdb")
    if name == "ray_debugpy":
        return importlib.import_module("ant_ray_amd.util.debugpy")
    if name == "ActorPool":
        return importlib.import_module("ant_ray_amd.util.actor_pool").ActorPool
    raise AttributeError(name)

from ant_ray_amd.util.inspect_serializability import (  # noqa: F401,E402
    inspect_serializability,
)
