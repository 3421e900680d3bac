"""Placement groups. Parity: python/ray/util/placement_group.py (GCS-side
manager reference src/ray/gcs/gcs_placement_group_manager.h:55, 2PC scheduler
gcs_placement_group_scheduler.h:281 — our GCS does reserve/rollback per
bundle, see gcs.py _schedule_pg)."""
from __future__ import annotations

import time
from typing import Dict, List, Optional

from ant_ray_amd._private.ids import PlacementGroupID
from ant_ray_amd._private.worker import LOCAL_MODE, global_worker
from ant_ray_amd.exceptions import GetTimeoutError


class PlacementGroup:
    def __init__(self, pg_id: bytes, bundles: List[Dict[str, float]] = None):
        self._id = pg_id
        self.bundle_specs = bundles or []

    @property
    def id(self):
        return PlacementGroupID(self._id)

    def ready(self):
        """Returns an ObjectRef resolving when the PG is placed."""
        import ant_ray_amd as ray

        pg = self

        @ray.remote(num_cpus=0)
        def _pg_ready_waiter():
            return True

        self.wait(timeout_seconds=None)
        return _pg_ready_waiter.remote()

    def wait(self, timeout_seconds: Optional[float] = 30) -> bool:
        cw = global_worker.core_worker
        if cw.mode == LOCAL_MODE:
            return True
        deadline = None if timeout_seconds is None else time.monotonic() + timeout_seconds
        while True:
            view = cw.io.run(
                cw.gcs.call("get_placement_group", {"pg_id": self._id}, timeout=30),
                timeout=35,
            )
            if view and view["state"] == "CREATED":
                return True
            if view and view["state"] == "FAILED":
                return False
            if deadline is not None and time.monotonic() > deadline:
                return False
            time.sleep(0.05)

    def __reduce__(self):
        return (PlacementGroup, (self._id, self.bundle_specs))


def placement_group(
    bundles: List[Dict[str, float]],
    strategy: str = "PACK",
    name: str = "",
    lifetime: Optional[str] = None,
    _max_cpu_fraction_per_node: Optional[float] = None,
) -> PlacementGroup:
    cw = global_worker.core_worker
    if cw is None or not cw.connected:
        raise RuntimeError("ray.init() must be called first")
    if strategy not in ("PACK", "SPREAD", "STRICT_PACK", "STRICT_SPREAD"):
        raise ValueError(f"invalid placement strategy {strategy}")
    for b in bundles:
        if not b or any(v < 0 for v in b.values()):
            raise ValueError("bundles must be non-empty dicts of >=0 amounts")
    pg_id = PlacementGroupID.from_random().binary()
    if cw.mode == LOCAL_MODE:
        return PlacementGroup(pg_id, bundles)
    cw.io.run(
        cw.gcs.call(
            "create_placement_group",
            {"pg_id": pg_id, "bundles": bundles, "strategy": strategy, "name": name},
            timeout=30,
        ),
        timeout=35,
    )
    return PlacementGroup(pg_id, bundles)


def remove_placement_group(pg: PlacementGroup):
    cw = global_worker.core_worker
    if cw.mode == LOCAL_MODE:
        return
    cw.io.run(
        cw.gcs.call("remove_placement_group", {"pg_id": pg._id}, timeout=30),
        timeout=35,
    )


def placement_group_table(pg: Optional[PlacementGroup] = None):
    cw = global_worker.core_worker
    if cw.mode == LOCAL_MODE:
        return {}
    pgs = cw.io.run(cw.gcs.call("list_placement_groups", {}, timeout=30), timeout=35)
    out = {}
    for view in pgs:
        out[view["pg_id"].hex()] = {
            "placement_group_id": view["pg_id"].hex(),
            "state": view["state"],
            "strategy": view["strategy"],
            "bundles": {i: b for i, b in enumerate(view["bundles"])},
            "name": view.get("name", ""),
        }
    if pg is not None:
        return out.get(pg._id.hex(), {})
    return out


def get_placement_group(placement_group_name: str) -> PlacementGroup:
    """Look up a placement group by the name it was created with
    (parity: reference util/placement_group.py get_placement_group)."""
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    views = cw.io.run(cw.gcs.call("list_placement_groups", {}, timeout=30),
                      timeout=35)
    for v in views:
        if v.get("name") == placement_group_name:
            return PlacementGroup(v["pg_id"], v.get("bundles"))
    raise ValueError(
        f"Failed to look up placement group with name {placement_group_name!r}")


def get_current_placement_group() -> Optional[PlacementGroup]:
    return None
