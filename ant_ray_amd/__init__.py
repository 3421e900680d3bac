"""ant_ray_amd — MI355X-native distributed actor/task runtime with
Ray-compatible public APIs.

Role parity: the `ray` top-level API (reference python/ray/__init__.py;
ray.init at python/ray/_private/worker.py:1431, ray.get :2853, ray.put :3032).
A compatibility alias package `ray` re-exports everything here.
"""
from __future__ import annotations

import atexit
import os
import time
from typing import Any, List, Optional, Sequence, Union

from ant_ray_amd import exceptions
from ant_ray_amd._private.object_ref import ObjectRef
from ant_ray_amd._private.worker import ObjectRefGenerator  # noqa: F401
from ant_ray_amd._private.worker import (
    DRIVER_MODE,
    LOCAL_MODE,
    WORKER_MODE,
    CoreWorker,
    global_worker,
)
from ant_ray_amd.actor import ActorClass, ActorHandle, method
from ant_ray_amd.remote_function import RemoteFunction

__version__ = "0.1.0"

__all__ = [
    "init", "shutdown", "is_initialized", "remote", "get", "put", "wait",
    "kill", "cancel", "get_actor", "get_gpu_ids", "get_runtime_context",
    "nodes", "cluster_resources", "available_resources", "method",
    "ObjectRef", "ObjectRefGenerator", "ActorHandle", "exceptions", "actor", "remote_function",
    "util", "train", "serve", "data", "tune",
    "LoggingConfig", "ClientBuilder", "client", "Language", "SCRIPT_MODE",
    "show_in_dashboard", "java_function", "java_actor_class", "cpp_function",
]


def init(
    address: Optional[str] = None,
    *,
    num_cpus: Optional[int] = None,
    num_gpus: Optional[int] = None,
    resources: Optional[dict] = None,
    object_store_memory: Optional[int] = None,
    local_mode: bool = False,
    namespace: Optional[str] = None,
    ignore_reinit_error: bool = False,
    runtime_env: Optional[dict] = None,
    log_to_driver: bool = True,
    dashboard_host: str = "127.0.0.1",
    dashboard_port: Optional[int] = None,
    include_dashboard: Optional[bool] = None,
    _node_ip_address: str = "127.0.0.1",
    _system_config: Optional[dict] = None,
    **kwargs,
):
    """Connect to (or start) a cluster. Parity: ray.init (worker.py:1431)."""
    if global_worker.connected:
        if ignore_reinit_error:
            return RuntimeContext(global_worker)
        raise RuntimeError("ray.init() called twice; pass ignore_reinit_error=True")

    lc = kwargs.pop("logging_config", None)
    if lc is not None:
        lc._apply()

    global_worker.namespace = namespace or f"ns-{os.getpid()}"

    if local_mode:
        cw = CoreWorker(LOCAL_MODE)
        cw.connect_local_mode()
        global_worker.core_worker = cw
        global_worker.mode = LOCAL_MODE
        atexit.register(shutdown)
        return RuntimeContext(global_worker)

    address = address or os.environ.get("ANTRAY_ADDRESS") or os.environ.get("RAY_ADDRESS")
    head = None
    if address in (None, "local"):
        from ant_ray_amd._private.node import start_head

        head = start_head(
            num_cpus=num_cpus,
            num_gpus=num_gpus,
            object_store_memory=object_store_memory,
            resources=resources,
            host=_node_ip_address,
        )
        gcs_addr = head.info["gcs_addr"]
        global_worker.session_dir = head.info["session_dir"]
    elif address == "auto":
        # find the most recent session on this machine
        base = "/tmp/antray"
        candidates = []
        if os.path.isdir(base):
            for d in os.listdir(base):
                p = os.path.join(base, d, "head.json")
                if os.path.exists(p):
                    candidates.append(p)
        if not candidates:
            raise ConnectionError("address='auto' but no running cluster found")
        import json as _json

        with open(sorted(candidates)[-1]) as f:
            info = _json.load(f)
        gcs_addr = info["gcs_addr"]
        global_worker.session_dir = info["session_dir"]
    else:
        gcs_addr = address
    # Ray-Client scheme (parity ray://host:port — util/client): a remote
    # driver that proxies big objects through the raylet data plane
    if gcs_addr.startswith("ray://"):
        gcs_addr = gcs_addr[len("ray://"):]

    host, port = gcs_addr.rsplit(":", 1)
    cw = CoreWorker(DRIVER_MODE, node_ip=_node_ip_address, session_dir=global_worker.session_dir)
    # ant-fork parity: pin this driver's job to a virtual cluster
    # (reference: job submission carries virtual_cluster_id; scheduling is
    # then restricted to that cluster's nodes)
    vc = kwargs.pop("_virtual_cluster_id", None) or os.environ.get(
        "ANTRAY_VIRTUAL_CLUSTER")
    if vc:
        cw.virtual_cluster_id = vc
    try:
        cw.connect((host, int(port)), is_driver=True)
    except Exception:
        if head is not None:
            head.terminate()
        raise
    global_worker.core_worker = cw
    global_worker.mode = DRIVER_MODE
    global_worker._head_proc = head
    if lc is not None:
        # propagate to workers: every spawned worker applies this at boot
        import json as _json

        try:
            cw.io.run(cw.gcs.call("kv_put", {
                "ns": "_cluster", "key": b"logging_config",
                "value": _json.dumps(lc._to_dict()).encode(),
                "overwrite": True}, timeout=10), timeout=15)
        except Exception:
            pass
    atexit.register(shutdown)
    if include_dashboard:
        from ant_ray_amd.dashboard import start_dashboard

        start_dashboard(port=dashboard_port or 8265, host=dashboard_host)
    return RuntimeContext(global_worker)


def shutdown(_exiting_interpreter: bool = False):
    cw = global_worker.core_worker
    if cw is None:
        return
    head = global_worker._head_proc
    if head is not None and cw.connected and cw.mode == DRIVER_MODE:
        try:
            cw.io.run(cw.gcs.call("shutdown", {}, timeout=5), timeout=6)
        except Exception:
            pass
    try:
        cw.shutdown()
    except Exception:
        pass
    if head is not None:
        # give the head a moment to exit cleanly, then make sure
        for _ in range(20):
            if head.proc.poll() is not None:
                break
            time.sleep(0.05)
        head.terminate()
    global_worker.core_worker = None
    global_worker.mode = None
    global_worker._head_proc = None


def is_initialized() -> bool:
    return global_worker.connected


def _check_connected():
    if not global_worker.connected:
        raise RuntimeError("ant-ray has not been started; call ray.init() first")


def remote(*args, **kwargs):
    """@ray.remote decorator for functions and classes."""

    def make(obj):
        import inspect

        if inspect.isclass(obj):
            return ActorClass(obj, kwargs)
        return RemoteFunction(obj, kwargs)

    if len(args) == 1 and not kwargs and (callable(args[0]) or isinstance(args[0], type)):
        return make(args[0])
    if args:
        raise TypeError("@ray.remote accepts only keyword options")
    return make


def put(value: Any, *, _owner=None, _tensor_transport: Optional[str] = None) -> ObjectRef:
    _check_connected()
    if isinstance(value, ObjectRef):
        raise TypeError("Calling 'put' on an ObjectRef is not allowed")
    return global_worker.core_worker.put(value, tensor_transport=_tensor_transport)


def get(
    object_refs: Union[ObjectRef, Sequence[ObjectRef]],
    *,
    timeout: Optional[float] = None,
):
    _check_connected()
    from ant_ray_amd.dag.node import CompiledDAGRef

    if isinstance(object_refs, CompiledDAGRef):
        return object_refs.get(timeout)
    single = isinstance(object_refs, ObjectRef)
    refs = [object_refs] if single else list(object_refs)
    for r in refs:
        if not isinstance(r, ObjectRef):
            raise TypeError(f"ray.get takes ObjectRefs, got {type(r)}")
    values = global_worker.core_worker.get(refs, timeout)
    return values[0] if single else values


def wait(
    object_refs: Sequence[ObjectRef],
    *,
    num_returns: int = 1,
    timeout: Optional[float] = None,
    fetch_local: bool = True,
):
    _check_connected()
    if isinstance(object_refs, ObjectRef):
        raise TypeError("ray.wait takes a list of ObjectRefs")
    if num_returns > len(object_refs):
        raise ValueError("num_returns > number of refs")
    return global_worker.core_worker.wait(
        list(object_refs), num_returns=num_returns, timeout=timeout, fetch_local=fetch_local
    )


def kill(actor: ActorHandle, *, no_restart: bool = True):
    _check_connected()
    if not isinstance(actor, ActorHandle):
        raise TypeError("ray.kill takes an ActorHandle")
    actor._ray_kill(no_restart=no_restart)


def cancel(object_ref: ObjectRef, *, force: bool = False, recursive: bool = True):
    """Cancel the task that produces object_ref (parity: ray.cancel /
    CoreWorker::CancelTask). Queued tasks are dropped; running tasks get
    KeyboardInterrupt at their next Python bytecode boundary (force=False
    — a task blocked in one long C call needs force) or their worker
    killed outright (force=True);
    ray.get on the ref then raises TaskCancelledError. recursive is
    accepted for API parity (children are not chased yet)."""
    _check_connected()
    cw = global_worker.core_worker
    if cw.mode == LOCAL_MODE:
        return  # local-mode tasks run synchronously at submit
    cw.cancel_task(object_ref, force=force)


def get_actor(name: str, namespace: Optional[str] = None) -> ActorHandle:
    _check_connected()
    cw = global_worker.core_worker
    if cw.mode == LOCAL_MODE:
        raise ValueError("get_actor is not supported in local mode")
    view = cw.io.run(
        cw.gcs.call(
            "get_actor_by_name",
            {"name": name, "namespace": namespace or ""},
            timeout=30,
        ),
        timeout=35,
    )
    if view is None or view["state"] == "DEAD":
        raise ValueError(f"Failed to look up actor with name '{name}'")
    return ActorHandle(view["actor_id"])


def get_gpu_ids() -> List[int]:
    _check_connected()
    return list(global_worker.core_worker.gpu_ids)


def nodes() -> List[dict]:
    _check_connected()
    cw = global_worker.core_worker
    if cw.mode == LOCAL_MODE:
        return [{
            "NodeID": "local", "Alive": True,
            "Resources": {"CPU": float(os.cpu_count() or 1)},
        }]
    table = cw.io.run(cw.gcs.call("node_table", {}, timeout=30), timeout=35)
    return [
        {
            "NodeID": n["node_id"].hex(),
            "Alive": n["alive"],
            "NodeManagerAddress": n["addr"][0],
            "NodeManagerPort": n["addr"][1],
            "Resources": n["resources_total"],
            "Available": n["resources_available"],
            "ObjectStoreSocketName": n["store_path"],
            "Labels": n.get("labels") or {},
        }
        for n in table
    ]


def cluster_resources() -> dict:
    _check_connected()
    cw = global_worker.core_worker
    if cw.mode == LOCAL_MODE:
        return {"CPU": float(os.cpu_count() or 1)}
    return cw.io.run(cw.gcs.call("cluster_resources", {}, timeout=30), timeout=35)["total"]


def available_resources() -> dict:
    _check_connected()
    cw = global_worker.core_worker
    if cw.mode == LOCAL_MODE:
        return {"CPU": float(os.cpu_count() or 1)}
    return cw.io.run(cw.gcs.call("cluster_resources", {}, timeout=30), timeout=35)["available"]


class RuntimeContext:
    """Parity: ray.runtime_context.RuntimeContext."""

    def __init__(self, worker):
        self._worker = worker

    @property
    def address_info(self):
        head = self._worker._head_proc
        return dict(head.info) if head else {}

    def get_job_id(self):
        return str(self._worker.core_worker.job_id)

    def get_node_id(self):
        nid = self._worker.core_worker.node_id
        return nid.hex() if nid else "local"

    def get_actor_id(self):
        aid = self._worker.core_worker.actor_id
        return aid.hex() if aid else None

    def get_task_id(self):
        tid = self._worker.core_worker.current_task_id
        return tid.hex() if tid else None

    def get_worker_id(self):
        return self._worker.core_worker.worker_id.hex()

    def get_accelerator_ids(self):
        return {"GPU": [str(g) for g in self._worker.core_worker.gpu_ids]}

    @property
    def namespace(self):
        return self._worker.namespace

    @property
    def gcs_address(self):
        cw = self._worker.core_worker
        return f"{cw.gcs_addr[0]}:{cw.gcs_addr[1]}" if getattr(cw, "gcs_addr", None) else None

    def get_runtime_env_string(self):
        return "{}"


def timeline(filename: Optional[str] = None):
    """Chrome-trace of recorded task events (parity: ray.timeline);
    returns the event list, and writes JSON when filename is given."""
    _check_connected()
    from ant_ray_amd.util.state import get_timeline

    trace = get_timeline()
    if filename:
        import json as _json

        with open(filename, "w") as f:
            _json.dump(trace, f)
    return trace


def get_runtime_context() -> RuntimeContext:
    _check_connected()
    return RuntimeContext(global_worker)


from ant_ray_amd._private.logging_config import LoggingConfig  # noqa: E402
from ant_ray_amd.client_builder import ClientBuilder, client  # noqa: E402

SCRIPT_MODE = DRIVER_MODE  # legacy alias (reference worker.py:111)


class Language:
    """Cross-language markers (reference Language proto enum). Only PYTHON
    tasks execute in this build; JAVA/CPP exist so code that inspects the
    enum imports cleanly."""

    PYTHON = "PYTHON"
    JAVA = "JAVA"
    CPP = "CPP"


def _xlang_unsupported(kind):
    def fn(*_a, **_k):
        raise NotImplementedError(
            f"cross-language ({kind}) is not supported in this MI355X build; "
            "see PARITY.md §2.2")

    return fn


java_function = _xlang_unsupported("java_function")
java_actor_class = _xlang_unsupported("java_actor_class")
cpp_function = _xlang_unsupported("cpp_function")


class _Config:
    """ray._config parity: read access to the env-overridable system-config
    registry (`_private/ray_constants.py`)."""

    def __getattr__(self, name):
        from ant_ray_amd._private import ray_constants

        def get(default=None):
            sysc = getattr(ray_constants, "_system_config", {}) or {}
            if name in sysc:
                return sysc[name]
            return os.environ.get(f"RAY_{name}", default)

        return get


_config = _Config()


def show_in_dashboard(message: str, key: str = "", dtype: str = "text"):
    """Display a message for the current task/actor in the dashboard
    (parity: reference worker.py:2821). Stored in the GCS KV under the
    "dashboard_display" namespace; served at GET /api/display."""
    _check_connected()
    assert dtype in ("text", "html"), f"dtype accepts only: text, html"
    import json as _json

    cw = global_worker.core_worker
    owner = (cw.actor_id or cw.worker_id).hex()[:16]
    cw.io.submit(cw.gcs.call("kv_put", {
        "ns": "dashboard_display", "key": f"{owner}|{key}".encode(),
        "value": _json.dumps({"message": message, "dtype": dtype}).encode(),
        "overwrite": True,
    }))


# Submodules are imported lazily to keep `import ant_ray_amd` light.
def __getattr__(name):
    import importlib

    if name in ("util", "train", "serve", "data", "tune", "dag", "experimental",
                "autoscaler", "cluster_utils", "ops", "models", "parallel"):
        return importlib.import_module(f"ant_ray_amd.{name}")
    raise AttributeError(f"module 'ant_ray_amd' has no attribute '{name}'")
