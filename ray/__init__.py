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

# Any other `ray.X.Y` import resolves to `ant_ray_amd.X.Y` via a meta-path
# finder, so arbitrary-depth reference imports (ray.runtime_context,
# ray.workflow, ray.serve.handle, ray._private.worker, ...) work without
# enumerating them.
import importlib as _importlib  # noqa: E402
import importlib.abc as _ilabc  # noqa: E402
import importlib.util as _ilutil  # noqa: E402


class _AliasLoader(_ilabc.Loader):
    def __init__(self, impl):
        self._impl = impl
        # module_from_spec clobbers __name__/__spec__/__loader__ on the
        # (shared) impl module; save them to restore in exec_module so the
        # canonical ant_ray_amd identity survives (pickle-by-reference of
        # its functions depends on __name__)
        self._saved = {k: getattr(impl, k, None)
                       for k in ("__name__", "__spec__", "__loader__",
                                 "__package__")}

    def create_module(self, spec):
        return self._impl

    def exec_module(self, module):
        for k, v in self._saved.items():
            try:
                setattr(module, k, v)
            except Exception:
                pass


class _RayAliasFinder(_ilabc.MetaPathFinder):
    def find_spec(self, fullname, path=None, target=None):
        if not fullname.startswith("ray."):
            return None
        try:
            impl = _importlib.import_module("ant_ray_amd." + fullname[4:])
        except ImportError:
            return None
        return _ilutil.spec_from_loader(fullname, _AliasLoader(impl))


_sys.meta_path.insert(0, _RayAliasFinder())


def __getattr__(name):
    # `from ray import workflow` etc without a prior submodule import
    try:
        return _importlib.import_module(f"ant_ray_amd.{name}")
    except ImportError:
        raise AttributeError(f"module 'ray' has no attribute {name!r}")
