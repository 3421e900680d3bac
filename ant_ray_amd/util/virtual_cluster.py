"""Virtual clusters — partition one physical cluster into named slices.

Role parity: ant fork's virtual clusters (reference
src/ray/gcs/gcs_virtual_cluster.h:29-610, design
doc/source/virtual-cluster/design-overview.rst:16-100): a virtual cluster
owns a subset of nodes; jobs created with
`ray.init(_virtual_cluster_id=...)` (or env ANTRAY_VIRTUAL_CLUSTER)
schedule actors/placement groups only on those nodes; count-based clusters
replenish from the unassigned pool when a member node dies.
"""
from __future__ import annotations

from typing import List, Optional


def _call(method: str, payload: dict, timeout: float = 30):
    import ant_ray_amd as ray
    from ant_ray_amd._private.worker import global_worker

    if not ray.is_initialized():
        ray.init(ignore_reinit_error=True)
    cw = global_worker.core_worker
    return cw.io.run(cw.gcs.call(method, payload, timeout=timeout),
                     timeout=timeout + 5)


def create_or_update_virtual_cluster(
    virtual_cluster_id: str,
    *,
    node_count: Optional[int] = None,
    node_ids: Optional[List[str]] = None,
    divisible: bool = False,
) -> dict:
    """Create (or resize) a virtual cluster. Give either a node_count (the
    GCS picks unassigned nodes, and replaces dead ones) or explicit
    node_ids (hex strings from ray.nodes())."""
    if (node_count is None) == (node_ids is None):
        raise ValueError("pass exactly one of node_count / node_ids")
    r = _call("create_or_update_virtual_cluster", {
        "virtual_cluster_id": virtual_cluster_id,
        "node_count": node_count,
        "node_ids": node_ids,
        "divisible": divisible,
    })
    if not r.get("ok"):
        raise RuntimeError(r.get("error", "virtual cluster update failed"))
    return r["view"]


def remove_virtual_cluster(virtual_cluster_id: str) -> bool:
    return _call("remove_virtual_cluster",
                 {"virtual_cluster_id": virtual_cluster_id})["ok"]


def list_virtual_clusters() -> List[dict]:
    return _call("list_virtual_clusters", {})


def get_virtual_cluster(virtual_cluster_id: str) -> Optional[dict]:
    return _call("get_virtual_cluster",
                 {"virtual_cluster_id": virtual_cluster_id})
