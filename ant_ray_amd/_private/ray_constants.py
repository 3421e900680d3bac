"""Config registry: typed, env-overridable knobs.

Role parity: reference src/ray/common/ray_config_def.h (~470
RAY_CONFIG(type, name, default) knobs overridable via RAY_<name> env vars
or ray.init(_system_config=...)) and python/ray/_private/ray_constants.py.
Every tunable the runtime consults is declared here with the SAME override
convention (RAY_<name>); apply_system_config merges _system_config dicts.
"""
from __future__ import annotations

import os
from typing import Any, Dict

_REGISTRY: Dict[str, Any] = {}


def RAY_CONFIG(name: str, default, typ=None):
    typ = typ or type(default)
    raw = os.environ.get(f"RAY_{name}")
    if raw is not None:
        if typ is bool:
            val = raw.lower() in ("1", "true", "yes")
        else:
            val = typ(raw)
    else:
        val = default
    _REGISTRY[name] = val
    return val


def get(name: str, default=None):
    return _REGISTRY.get(name, default)


def apply_system_config(cfg: Dict[str, Any]):
    """ray.init(_system_config={...}) parity: overrides after env."""
    for k, v in (cfg or {}).items():
        _REGISTRY[k] = v


def all_config() -> Dict[str, Any]:
    return dict(_REGISTRY)


# ---- core knobs (names follow the reference's where one exists) ----------
# object store
object_store_memory_fraction = RAY_CONFIG("object_store_memory_fraction", 0.6)
max_direct_call_object_size = RAY_CONFIG("max_direct_call_object_size",
                                         100 * 1024)
object_manager_default_chunk_size = RAY_CONFIG(
    "object_manager_default_chunk_size", 5 * 1024 * 1024)
# scheduling / leases
pipeline_depth = RAY_CONFIG("worker_lease_pipeline_depth", 8)
lease_idle_release_s = RAY_CONFIG("worker_lease_timeout_milliseconds", 2000) / 1000.0
worker_prestart_cap = RAY_CONFIG("worker_prestart_cap", 16)
idle_worker_killing_time_threshold_ms = RAY_CONFIG(
    "idle_worker_killing_time_threshold_ms", 60_000)
# health
health_check_period_ms = RAY_CONFIG("health_check_period_ms", 1000)
health_check_failure_threshold = RAY_CONFIG("health_check_failure_threshold", 5)
# gcs
gcs_task_events_max = RAY_CONFIG("task_events_max_num_task_in_gcs", 20000)
# ant-fork
virtual_cluster_enabled = RAY_CONFIG("virtual_cluster_enabled", True)
# chaos/testing (parity: asio_chaos.cc RAY_testing_asio_delay_us)
testing_rpc_delay_us = RAY_CONFIG("testing_asio_delay_us", "", str)

# python-side constants (ray_constants.py parity)
DEFAULT_DASHBOARD_PORT = 8265
DEFAULT_PORT = 6379
ID_SIZE = 28
