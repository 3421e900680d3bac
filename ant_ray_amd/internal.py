"""Internal APIs. Parity: ray._private.internal_api (free at
internal_api.py in the reference)."""
from __future__ import annotations

from typing import List, Sequence, Union

from ant_ray_amd._private.object_ref import ObjectRef
from ant_ray_amd._private.worker import global_worker


def free(object_refs: Union[ObjectRef, Sequence[ObjectRef]], local_only: bool = False):
    """Eagerly free objects (shm + GPU object store) at every known holder."""
    if isinstance(object_refs, ObjectRef):
        object_refs = [object_refs]
    cw = global_worker.core_worker
    oids = [r.binary() for r in object_refs]
    cw.store.free(oids)
    try:
        from ant_ray_amd.experimental.gpu_object_manager import gpu_object_store

        gpu_object_store.free(oids)
    except Exception:
        pass
    if local_only or cw.io is None:
        return
    targets = set()
    for r in object_refs:
        oid = r.binary()
        holder = cw._object_locations.get(oid)
        if holder:
            targets.add(tuple(holder))
        if r.owner_addr and tuple(r.owner_addr) != cw.addr:
            targets.add(tuple(r.owner_addr))

    async def _send():
        for addr in targets:
            try:
                conn = await cw._get_worker_conn_async_cached(addr)
                await conn.call("free_objects", {"oids": oids}, timeout=10)
            except Exception:
                pass

    cw.io.run(_send(), timeout=30)
