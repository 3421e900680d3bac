"""ray.util parity namespace."""
from ant_ray_amd.util.placement_group import (  # noqa: F401
    placement_group,
    placement_group_table,
    remove_placement_group,
    get_current_placement_group,
)


def __getattr__(name):
    import importlib

    if name in ("collective", "state", "queue", "metrics", "scheduling_strategies",
                "actor_pool"):
        return importlib.import_module(f"ant_ray_amd.util.{name}")
    if name == "ActorPool":
        return importlib.import_module("ant_ray_amd.util.actor_pool").ActorPool
    raise AttributeError(name)

from ant_ray_amd.util.inspect_serializability import (  # noqa: F401,E402
    inspect_serializability,
)
