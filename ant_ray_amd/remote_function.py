"""@ray.remote functions.

Role parity: reference python/ray/remote_function.py:41 (class RemoteFunction,
_remote at :314). Options subset mirrors python/ray/_common/ray_option_utils.py.
"""
from __future__ import annotations

import hashlib
from typing import Any, Dict

from ant_ray_amd._private.worker import LOCAL_MODE, global_worker

_VALID_OPTIONS = {
    "num_cpus", "num_gpus", "num_returns", "resources", "max_retries",
    "name", "runtime_env", "memory", "scheduling_strategy", "max_calls",
    "retry_exceptions", "accelerator_type", "label_selector", "_metadata",
}


def _normalize_opts(opts: Dict[str, Any]) -> Dict[str, Any]:
    out = dict(opts)
    strategy = out.pop("scheduling_strategy", None)
    if isinstance(strategy, str):
        if strategy == "SPREAD":
            out["_spread"] = True
        # "DEFAULT" falls through
    if strategy is not None and not isinstance(strategy, str):
        from ant_ray_amd.util.scheduling_strategies import (
            NodeAffinitySchedulingStrategy,
            PlacementGroupSchedulingStrategy,
        )

        from ant_ray_amd.util.scheduling_strategies import (
            NodeLabelSchedulingStrategy,
        )

        if isinstance(strategy, NodeLabelSchedulingStrategy):
            out["_label_selector"] = {"hard": strategy.hard,
                                      "soft": strategy.soft}
        elif isinstance(strategy, PlacementGroupSchedulingStrategy):
            pg = strategy.placement_group
            out["placement_group"] = {
                "pg_id": pg.id.binary() if hasattr(pg.id, "binary") else pg.id,
                "bundle_index": strategy.placement_group_bundle_index,
            }
        elif isinstance(strategy, NodeAffinitySchedulingStrategy):
            nid = strategy.node_id
            out["_node_affinity"] = bytes.fromhex(nid) if isinstance(nid, str) else nid
    # Ray 2.x plain label_selector option: dict of hard constraints
    sel = out.pop("label_selector", None)
    if sel:
        cur = out.setdefault("_label_selector", {"hard": {}, "soft": {}})
        cur["hard"] = {**cur.get("hard", {}), **sel}
    return out


class RemoteFunction:
    def __init__(self, fn, default_opts: Dict[str, Any] = None):
        if not callable(fn):
            raise TypeError("@ray.remote requires a callable")
        self._function = fn
        self._opts = _normalize_opts(default_opts or {})
        self.__name__ = getattr(fn, "__name__", "remote_function")
        self.__doc__ = getattr(fn, "__doc__", None)
        import cloudpickle

        try:
            blob = cloudpickle.dumps(fn)
        except Exception:
            blob = (getattr(fn, "__module__", "") + "." + self.__name__).encode()
        self._fn_id = hashlib.blake2b(blob, digest_size=20).digest()

    def __call__(self, *args, **kwargs):
        raise TypeError(
            f"Remote function '{self.__name__}' cannot be called directly; "
            f"use '{self.__name__}.remote()'."
        )

    def options(self, **opts):
        merged = dict(self._opts)
        merged.update(_normalize_opts(opts))
        rf = RemoteFunction.__new__(RemoteFunction)
        rf._function = self._function
        rf._opts = merged
        rf.__name__ = self.__name__
        rf.__doc__ = self.__doc__
        rf._fn_id = self._fn_id
        return rf

    def remote(self, *args, **kwargs):
        if not global_worker.connected:
            raise RuntimeError("ray.init() must be called before .remote()")
        cw = global_worker.core_worker
        opts = self._opts
        n_returns = opts.get("num_returns", 1)
        if cw.mode == LOCAL_MODE:
            refs = cw.executor.submit_task(self._function, args, kwargs, opts)
        else:
            refs = cw.submit_task(self._function, self._fn_id, args, kwargs, opts)
        if n_returns in ("streaming", "dynamic"):
            return refs  # ObjectRefGenerator
        if n_returns == 1:
            return refs[0]
        return refs

    @property
    def _function_name(self):
        return self.__name__

    def bind(self, *args, **kwargs):
        from ant_ray_amd.dag.function_node import FunctionNode

        return FunctionNode(self, args, kwargs)
