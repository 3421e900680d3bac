"""ObjectRef — the future/handle for a stored or pending object.

Role parity: reference python/ray/_raylet.pyx ObjectRef + ownership metadata
(owner address travels with the ref so any holder can resolve it; reference
src/ray/core_worker/reference_counter.h:44 tracks distributed counts). Local
refcounts are maintained via __del__ -> worker.remove_local_ref.
"""
from __future__ import annotations

from typing import Optional, Tuple

from ant_ray_amd._private.ids import ObjectID


class ObjectRef:
    __slots__ = ("_id", "owner_addr", "_worker", "call_site", "__weakref__")

    def __init__(self, id_bytes: bytes, owner_addr: Optional[Tuple[str, int]] = None,
                 worker=None, call_site: str = "", skip_adding_local_ref: bool = False):
        self._id = id_bytes
        self.owner_addr = tuple(owner_addr) if owner_addr else None
        self._worker = worker
        self.call_site = call_site
        if worker is not None and not skip_adding_local_ref:
            worker.add_local_ref(id_bytes)

    @classmethod
    def _rehydrate(cls, id_bytes: bytes, owner_addr):
        # Called during deserialization in a (possibly) different process.
        from ant_ray_amd._private.worker import global_worker

        w = global_worker if global_worker is not None and global_worker.connected else None
        ref = cls(id_bytes, owner_addr, worker=w)
        return ref

    def binary(self) -> bytes:
        return self._id

    def hex(self) -> str:
        return self._id.hex()

    @property
    def id(self) -> ObjectID:
        return ObjectID(self._id)

    def task_id(self):
        return None

    def future(self):
        """Return a concurrent.futures.Future resolving to the value."""
        from ant_ray_amd._private.worker import global_worker

        return global_worker.core_worker.get_async(self)

    def __await__(self):
        import asyncio

        from ant_ray_amd._private.worker import global_worker

        cf = global_worker.core_worker.get_async(self)
        return asyncio.wrap_future(cf).__await__()

    def __hash__(self):
        return hash(self._id)

    def __eq__(self, other):
        return isinstance(other, ObjectRef) and other._id == self._id

    def __repr__(self):
        return f"ObjectRef({self._id.hex()})"

    def __reduce__(self):
        # Plain pickle path (e.g. msgpack'd through our RPC): no borrow
        # bookkeeping — the runtime uses serialization.py for values, which
        # records contained refs explicitly.
        return (ObjectRef._rehydrate, (self._id, self.owner_addr))

    def __del__(self):
        w = self._worker
        if w is not None:
            try:
                w.remove_local_ref(self._id)
            except Exception:
                pass
