"""Mutable shared-memory channels for compiled DAGs.

Role parity: python/ray/experimental/channel/shared_memory_channel.py
(Channel :186 — mutable plasma objects with reader/writer refs). Here a
channel is a pinned slot in the node's C++ shm arena with a version futex
(`csrc/shm_store.cpp ChanHeader`): one writer, N readers, depth-1
backpressure. A hop costs a memcpy + futex wake instead of an actor RPC
round trip.

Single-node by design (shm): compiled DAGs fall back to the actor-RPC
path when the cluster spans nodes.
"""
from __future__ import annotations

import os
import struct
from typing import Any, Optional


class ChannelClosedError(Exception):
    """Read/write on a channel whose writer tore the DAG down."""


class _WrappedError:
    """An exception raised inside a DAG node, shipped through the channel
    so the driver re-raises it at get() time."""

    def __init__(self, exc: BaseException):
        self.exc = exc


class Channel:
    """Single-writer, num_readers-reader mutable shm channel."""

    def __init__(self, capacity: int = 1 << 20, num_readers: int = 1,
                 _handle: Optional[tuple] = None):
        if _handle is not None:
            self._path, self._oid, self._num_readers = _handle
            self._store = _open_store(self._path)
        else:
            from ant_ray_amd._private.worker import global_worker

            cw = global_worker.core_worker
            if cw is None or cw.store.shm is None:
                raise RuntimeError(
                    "channels need a local shm store (ray.init on this node)")
            self._store = cw.store.shm
            p = self._store.path
            self._path = p() if callable(p) else p
            self._oid = os.urandom(20)
            self._num_readers = num_readers
            self._store.channel_create(self._oid, capacity, num_readers)
        self._last_version = 0
        # hipIpc producer-side pin window: tensors exported for version N
        # stay pinned until write(N+2) is admitted. Admission of write(N)
        # means every (single-threaded) reader returned from read(N-1),
        # which strictly follows its fenced clone of version N-2 — so N-2's
        # device memory is provably unreferenced (see read()).
        self._gpu_pins: list = []

    def __reduce__(self):
        return (_attach, (self._path, self._oid, self._num_readers))

    # ------------------------------------------------------------- data plane
    def write(self, value: Any, timeout: Optional[float] = None):
        from ant_ray_amd._private import serialization

        pins = None
        if _may_hold_gpu(value):
            # GPU payloads ride the hipIpc tier: only the 64-byte handle
            # descriptor crosses the shm channel; the reader maps the
            # producer's HBM over xGMI and clones device-to-device — zero
            # host copies (vs the reference's CPU serialization through
            # shared_memory_channel.py for non-NCCL-annotated tensors).
            with serialization.gpu_transport_context("hip_ipc") as gctx:
                sobj = serialization.serialize(value)
            pins = gctx.pinned
        else:
            sobj = serialization.serialize(value)
        meta = sobj.metadata
        payload = struct.pack("<I", len(meta)) + meta + sobj.to_bytes()
        try:
            self._store.channel_write(
                self._oid, payload, -1.0 if timeout is None else timeout)
        except RuntimeError as e:
            raise _map_err(e) from None
        if pins is not None or self._gpu_pins:
            self._gpu_pins.append(pins or [])
            del self._gpu_pins[:-2]  # keep versions N and N-1 pinned

    def read(self, timeout: Optional[float] = None, unwrap: bool = True) -> Any:
        """unwrap=False returns _WrappedError values instead of raising —
        DAG node loops use it to forward upstream errors downstream."""
        from ant_ray_amd._private import serialization

        try:
            data, self._last_version = self._store.channel_read(
                self._oid, self._last_version,
                -1.0 if timeout is None else timeout)
        except RuntimeError as e:
            raise _map_err(e) from None
        (mlen,) = struct.unpack_from("<I", data)
        meta = data[4:4 + mlen]
        from ant_ray_amd.experimental.gpu_object_manager.gpu_object_store import (
            gpu_import_clone,
        )

        with gpu_import_clone() as imp:
            value = serialization.deserialize(
                memoryview(data)[4 + mlen:], meta)
        if imp.cloned:
            # fence the async device clones BEFORE the next channel_read
            # acks a newer version (the producer's pin-window proof relies
            # on clone-complete-before-next-ack)
            import torch

            torch.cuda.current_stream().synchronize()
        if unwrap and isinstance(value, _WrappedError):
            raise value.exc
        return value

    def close(self):
        self._gpu_pins.clear()
        try:
            self._store.channel_close(self._oid)
        except KeyError:
            pass

    def destroy(self):
        """Close and free the slot (creator only — drops the create pin)."""
        self.close()
        try:
            self._store.release(self._oid)
            self._store.delete(self._oid)
        except Exception:
            pass


def _may_hold_gpu(value: Any) -> bool:
    """Cheap check whether a channel payload can contain CUDA tensors
    (walks one container level; deeper nesting still works — the
    serializer just takes the default CPU-copy path for those)."""
    import sys

    if "torch" not in sys.modules:
        return False
    import torch

    if not torch.cuda.is_available():
        return False

    def is_gpu(v):
        return isinstance(v, torch.Tensor) and v.is_cuda

    if is_gpu(value):
        return True
    if isinstance(value, (list, tuple, set)):
        return any(is_gpu(v) for v in value)
    if isinstance(value, dict):
        return any(is_gpu(v) for v in value.values())
    return False


def _attach(path: str, oid: bytes, num_readers: int) -> Channel:
    return Channel(_handle=(path, oid, num_readers))


def _open_store(path: str):
    """Reuse this process's mapping of the node store when possible."""
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    if cw is not None and cw.store.shm is not None:
        p = cw.store.shm.path
        if (p() if callable(p) else p) == path:
            return cw.store.shm
    from ant_ray_amd._shm_store import ShmStore

    return ShmStore.open(path, 10.0)


def _map_err(e: RuntimeError) -> Exception:
    if "closed" in str(e):
        return ChannelClosedError(str(e))
    return e


def _project_input(inp: Any, key) -> Any:
    """Mirror InputAttributeNode._submit over a channel-delivered input."""
    from ant_ray_amd.dag.node import _DagInput

    if isinstance(inp, _DagInput):
        if isinstance(key, int):
            return inp.args[key]
        if key in inp.kwargs:
            return inp.kwargs[key]
        if len(inp.args) == 1:
            obj = inp.args[0]
            return obj[key] if isinstance(obj, dict) else getattr(obj, key)
        raise KeyError(key)
    if isinstance(key, int):
        if key == 0:
            return inp
        raise IndexError(key)
    return inp[key] if isinstance(inp, dict) else getattr(inp, key)
