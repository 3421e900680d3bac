"""Object storage for a worker process: in-process memory store + shm store.

Role parity: reference CoreWorker's two-tier store — MemoryStore for inlined
small objects (src/ray/core_worker/store_provider/memory_store/memory_store.h:47,
<=100 KiB per ray_config_def.h:200) and the plasma provider for large ones
(store_provider/plasma_store_provider.cc). Here the large tier is our
direct-mapped shm store (csrc/shm_store.cpp).
"""
from __future__ import annotations

import threading
import time
from typing import Any, Dict, List, Optional

from ant_ray_amd._private import serialization
from ant_ray_amd.exceptions import GetTimeoutError, ObjectLostError

# Objects <= this are inlined in RPC replies / memory store (parity with
# reference max_direct_call_object_size, ray_config_def.h:200).
INLINE_OBJECT_MAX = 100 * 1024
PULL_CHUNK_BYTES = 5 << 20  # cross-node pull slice size (parity: reference
                            # ObjectManagerConfig object_chunk_size = 5 MiB)


class _InPlasma:
    """Memory-store marker: the value lives in the shm store."""

    __slots__ = ()


IN_PLASMA = _InPlasma()


class _Pending:
    """Memory-store marker: a task will produce this object."""

    __slots__ = ("event",)

    def __init__(self):
        self.event = threading.Event()


class MemoryStore:
    """In-process object map with blocking waits."""

    def __init__(self):
        self._lock = threading.Lock()
        self._objects: Dict[bytes, Any] = {}
        self._cv = threading.Condition(self._lock)

    def put(self, oid: bytes, value: Any):
        with self._cv:
            prev = self._objects.get(oid)
            self._objects[oid] = value
            self._cv.notify_all()
            if isinstance(prev, _Pending):
                prev.event.set()

    def mark_pending(self, oid: bytes):
        with self._cv:
            if oid not in self._objects:
                self._objects[oid] = _Pending()

    def get_now(self, oid: bytes):
        """Non-blocking; returns (found, value)."""
        with self._lock:
            v = self._objects.get(oid)
        if v is None or isinstance(v, _Pending):
            return False, None
        return True, v

    def contains(self, oid: bytes) -> bool:
        found, _ = self.get_now(oid)
        return found

    def is_pending(self, oid: bytes) -> bool:
        with self._lock:
            return isinstance(self._objects.get(oid), _Pending)

    def wait(self, oid: bytes, timeout: Optional[float]) -> bool:
        """Wait until oid is resolved (not pending/absent). True if resolved."""
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._cv:
            while True:
                v = self._objects.get(oid)
                if v is not None and not isinstance(v, _Pending):
                    return True
                remaining = None if deadline is None else deadline - time.monotonic()
                if remaining is not None and remaining <= 0:
                    return False
                self._cv.wait(remaining if remaining is not None else 1.0)

    def delete(self, oid: bytes):
        with self._lock:
            self._objects.pop(oid, None)


class ObjectStore:
    """The per-worker facade over memory store + shm store."""

    def __init__(self, shm_store=None):
        self.memory = MemoryStore()
        self.shm = shm_store  # ant_ray_amd._shm_store.ShmStore or None (local mode)

    # --- writes ---------------------------------------------------------------

    def put_serialized_to_shm(self, oid: bytes,
                              sobj: serialization.SerializedObject,
                              pin: bool = False):
        """pin=True keeps the creator's refcount (the owner holds the
        primary copy pinned while local refs exist — parity with the
        reference's primary-copy pinning; unpinned objects are LRU-evictable
        the moment readers drop them)."""
        try:
            off = self.shm.create_object(oid, sobj.total_size, sobj.metadata)
        except ValueError:
            # a re-executed task (lineage reconstruction / retry) rewrites
            # its own return object: drop the stale copy and recreate
            try:
                self.shm.delete(oid)
            except Exception:
                pass
            off = self.shm.create_object(oid, sobj.total_size, sobj.metadata)
        try:
            mv = self.shm.view_at(off, sobj.total_size, writable=True)
            sobj.write_into(mv)
            del mv
            self.shm.seal(oid)
        except Exception:
            self.shm.abort(oid)
            self.shm.release(oid)
            raise
        if not pin:
            self.shm.release(oid)

    def put_value(self, oid: bytes, value: Any, *, force_shm: bool = False) -> int:
        """Serialize and store; returns serialized size. Small values stay in
        the memory store unless force_shm (ray.put semantics)."""
        sobj = serialization.serialize(value)
        if force_shm or sobj.total_size > INLINE_OBJECT_MAX or self.shm is None:
            if self.shm is not None:
                self.put_serialized_to_shm(oid, sobj)
                self.memory.put(oid, IN_PLASMA)
            else:
                self.memory.put(oid, value)
        else:
            self.memory.put(oid, value)
        return sobj.total_size

    def put_local(self, oid: bytes, value: Any):
        self.memory.put(oid, value)

    # --- reads ----------------------------------------------------------------

    def get_from_shm(self, oid: bytes, timeout: Optional[float]):
        t = -1.0 if timeout is None else timeout
        buf, meta = self.shm.get_buffer(oid, t)
        if buf is None:
            raise GetTimeoutError(f"object {oid.hex()} not available after {timeout}s")
        value = serialization.deserialize(memoryview(buf), bytes(meta))
        return value, bytes(meta)

    def contains(self, oid: bytes) -> bool:
        found, v = self.memory.get_now(oid)
        if found and v is not IN_PLASMA:
            return True
        if self.shm is not None and self.shm.contains(oid):
            return True
        return found  # IN_PLASMA marker but evicted -> still claim; get may fail

    def free(self, oids: List[bytes]):
        for oid in oids:
            self.memory.delete(oid)
            if self.shm is not None:
                try:
                    self.shm.delete(oid)
                except Exception:
                    pass
