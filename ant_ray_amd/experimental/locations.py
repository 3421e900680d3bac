"""Object location introspection.

Role parity: reference python/ray/experimental/locations.py
(get_object_locations / get_local_object_locations). The owner's
location table stores holder ADDRESSES (host:port of the raylet data
plane) rather than raw node ids — they identify the node just as well
on this single-node-class deployment and are returned as the node_ids
entries.
"""
from typing import Dict, List


def get_object_locations(obj_refs: List, timeout_ms: int = -1) -> Dict:
    """{ref: {"node_ids": ["host:port"], "object_size": int|None}}."""
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    out = {}
    for ref in obj_refs:
        oid = ref.binary()
        holder = cw._object_locations.get(oid)
        meta = cw._owned.get(oid)
        if holder is None and meta is not None:
            # owned object still resident in this node's shm store
            holder = getattr(cw, "raylet_addr", None) or cw.addr
        node_ids = [f"{holder[0]}:{holder[1]}"] if holder else []
        size = meta.get("size") if isinstance(meta, dict) else None
        out[ref] = {"node_ids": node_ids, "object_size": size}
    return out


def get_local_object_locations(obj_refs: List) -> Dict:
    return get_object_locations(obj_refs)
