"""ray.experimental namespace (parity: reference
python/ray/experimental/__init__.py). Heavier members (channel,
gpu_object_manager, tqdm_ray, ...) import lazily."""


def __getattr__(name):
    import importlib

    if name in ("channel", "collective", "gpu_object_manager",
                "internal_kv", "tqdm_ray", "compiled_dag_ref", "queue",
                "locations"):
        return importlib.import_module(f"ant_ray_amd.experimental.{name}")
    if name in ("get_object_locations", "get_local_object_locations"):
        mod = importlib.import_module("ant_ray_amd.experimental.locations")
        return getattr(mod, name)
    raise AttributeError(f"module 'ray.experimental' has no attribute {name!r}")
