"""`ray` compatibility alias — re-exports ant_ray_amd under the reference's
package name so user code written for the reference (import ray;
ray.init(); @ray.remote; from ray import serve, train, data, tune; ...)
runs unchanged.
"""
import sys as _sys

import ant_ray_amd as _impl
from ant_ray_amd import *  # noqa: F401,F403
from ant_ray_amd import (  # noqa: F401
    __version__,
    actor,
    exceptions,
    remote_function,
)

# submodule aliases: make `import ray.serve`, `from ray.train import ...`,
# `from ray.util.queue import Queue` etc resolve to the ant_ray_amd modules
_ALIASES = [
    "train", "train.torch", "serve", "data", "tune", "dag", "util",
    "util.collective", "util.state", "util.queue", "util.metrics",
    "util.actor_pool", "util.placement_group", "util.scheduling_strategies",
    "util.virtual_cluster", "cluster_utils", "job_submission", "dashboard",
    "exceptions", "actor", "remote_function", "internal", "experimental",
    "scripts", "scripts.cli",
]


def _alias(name):
    import importlib

    try:
        mod = importlib.import_module(f"ant_ray_amd.{name}")
    except ImportError:
        return
    _sys.modules[f"ray.{name}"] = mod
    parts = name.split(".")
    if len(parts) == 1:
        globals()[parts[0]] = mod


for _name in _ALIASES:
    _alias(_name)

# ray.cloudpickle parity (reference vendors cloudpickle at this path)
import cloudpickle as _cp  # noqa: E402

_sys.modules["ray.cloudpickle"] = _cp
cloudpickle = _cp
