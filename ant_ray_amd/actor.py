"""Actor API: ActorClass / ActorHandle / ActorMethod.

Role parity: reference python/ray/actor.py (ActorClass :1228, _remote :1538,
ActorHandle :1942, _actor_method_call :2138). Handles serialize by actor id +
owner address; any process deserializing one can call the actor (address
resolution via GCS).
"""
from __future__ import annotations

import hashlib
from typing import Any, Dict, Optional

from ant_ray_amd._private.ids import ActorID, JobID
from ant_ray_amd._private.worker import LOCAL_MODE, global_worker
from ant_ray_amd.exceptions import RayActorError
from ant_ray_amd.remote_function import _normalize_opts

_VALID_ACTOR_OPTIONS = {
    "num_cpus", "num_gpus", "resources", "max_restarts", "max_task_retries",
    "max_concurrency", "name", "namespace", "lifetime", "runtime_env",
    "scheduling_strategy", "get_if_exists", "memory", "concurrency_groups",
    "max_pending_calls", "accelerator_type", "label_selector", "_metadata",
}


class ActorExit(SystemExit):
    """Raised by exit_actor(); the executor exits the worker gracefully
    after replying to the in-flight call."""


def exit_actor():
    """Terminate the current actor from inside one of its methods
    (parity: ray.actor.exit_actor) — pending queued calls fail with
    the actor-died error, the reply for THIS call still reaches the
    caller."""
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    if cw is None or cw.executor is None or cw.executor.actor_instance is None:
        raise RuntimeError("exit_actor() called outside an actor method")
    raise ActorExit(0)


def method(**kwargs):
    """@ray.method decorator (num_returns, concurrency_group...)."""

    def annotate(m):
        m.__ray_method_opts__ = kwargs
        return m

    return annotate


class ActorMethod:
    def __init__(self, handle: "ActorHandle", name: str, opts: Dict[str, Any]):
        self._handle = handle
        self._name = name
        self._opts = dict(opts)

    def options(self, **kwargs):
        merged = dict(self._opts)
        merged.update(kwargs)
        return ActorMethod(self._handle, self._name, merged)

    def remote(self, *args, **kwargs):
        return self._handle._actor_method_call(self._name, args, kwargs, self._opts)

    def bind(self, *args, **kwargs):
        """DAG node calling this method on the EXISTING actor (parity:
        actor method .bind in python/ray/dag)."""
        from ant_ray_amd.dag.node import ClassMethodNode, _ExistingActorShim

        return ClassMethodNode(_ExistingActorShim(self._handle), self._name,
                               args, kwargs)

    def __call__(self, *args, **kwargs):
        raise TypeError(
            f"Actor method '{self._name}' cannot be called directly; use "
            f"'.{self._name}.remote()'."
        )


class ActorHandle:
    def __init__(self, actor_id: bytes, method_opts: Dict[str, Dict] = None,
                 _owned: bool = False):
        self._ray_actor_id = actor_id
        self._method_opts = method_opts or {}
        cw = global_worker.core_worker
        if cw is not None and cw.mode != LOCAL_MODE and cw.connected:
            cw.actor_handle_added(actor_id)
            self._registered = True
        else:
            self._registered = False

    @property
    def _actor_id(self):
        return ActorID(self._ray_actor_id)

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        return ActorMethod(self, name, self._method_opts.get(name, {}))

    def _actor_method_call(self, method_name, args, kwargs, opts):
        cw = global_worker.core_worker
        if cw is None or not cw.connected:
            raise RuntimeError("ray.init() must be called first")
        n_returns = opts.get("num_returns", 1)
        if cw.mode == LOCAL_MODE:
            refs = cw.executor.submit_actor_task(
                self._ray_actor_id, method_name, args, kwargs, opts
            )
        else:
            refs = cw.submit_actor_task(self._ray_actor_id, method_name, args, kwargs, opts)
        if n_returns in ("streaming", "dynamic"):
            return refs  # ObjectRefGenerator
        if n_returns == 1:
            return refs[0]
        return refs

    def __reduce__(self):
        return (_rehydrate_handle, (self._ray_actor_id, self._method_opts))

    def __repr__(self):
        return f"ActorHandle({self._ray_actor_id.hex()[:16]})"

    def __del__(self):
        try:
            if self._registered and global_worker.core_worker is not None:
                global_worker.core_worker.actor_handle_removed(self._ray_actor_id)
        except Exception:
            pass

    def _ray_kill(self, no_restart=True):
        cw = global_worker.core_worker
        if cw.mode == LOCAL_MODE:
            cw.executor.kill_actor(self._ray_actor_id, no_restart)
        else:
            cw.kill_actor(self._ray_actor_id, no_restart)


def _rehydrate_handle(actor_id, method_opts):
    return ActorHandle(actor_id, method_opts)


class ActorOptionWrapper:
    def __init__(self, actor_cls: "ActorClass", opts):
        self._actor_cls = actor_cls
        self._opts = opts

    def remote(self, *args, **kwargs):
        return self._actor_cls._remote(args, kwargs, self._opts)

    def bind(self, *args, **kwargs):
        from ant_ray_amd.dag.class_node import ClassNode

        return ClassNode(self._actor_cls, args, kwargs, self._opts)


class ActorClass:
    def __init__(self, cls, default_opts: Dict[str, Any] = None):
        self._cls = cls
        self._default_opts = _normalize_opts(default_opts or {})
        self.__name__ = cls.__name__
        self.__doc__ = cls.__doc__
        self.__ray_actor_class__ = cls

    def __call__(self, *args, **kwargs):
        raise TypeError(
            f"Actor class '{self.__name__}' cannot be instantiated directly; "
            f"use '{self.__name__}.remote()'."
        )

    def options(self, **opts):
        merged = dict(self._default_opts)
        merged.update(_normalize_opts(opts))
        return ActorOptionWrapper(self, merged)

    def remote(self, *args, **kwargs):
        return self._remote(args, kwargs, self._default_opts)

    def bind(self, *args, **kwargs):
        from ant_ray_amd.dag.class_node import ClassNode

        return ClassNode(self, args, kwargs, self._default_opts)

    def _collect_method_opts(self):
        out = {}
        for name in dir(self._cls):
            try:
                m = getattr(self._cls, name)
            except Exception:
                continue
            opts = getattr(m, "__ray_method_opts__", None)
            if opts:
                out[name] = dict(opts)
        return out

    def _remote(self, args, kwargs, opts):
        if not global_worker.connected:
            raise RuntimeError("ray.init() must be called before .remote()")
        cw = global_worker.core_worker
        actor_id = ActorID.of(JobID.from_int(cw.job_id or 0)).binary()
        method_opts = self._collect_method_opts()
        if cw.mode == LOCAL_MODE:
            cw.executor.create_actor(self._cls, actor_id, args, kwargs, opts)
            return ActorHandle(actor_id, method_opts, _owned=True)
        reply = cw.create_actor(self._cls, actor_id, args, kwargs, opts)
        if reply.get("existing"):
            actor_id = reply["actor_id"]
        handle = ActorHandle(actor_id, method_opts, _owned=True)
        st = cw._get_actor_state(actor_id)
        st.is_owner = True
        st.detached = opts.get("lifetime") == "detached"
        if opts.get("max_pending_calls"):
            st.max_pending_calls = int(opts["max_pending_calls"])
        return handle
