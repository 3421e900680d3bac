"""GPU object store: tensors stay on device, moved actor-to-actor by
hipIpcMemHandle (zero host copies).

Role parity: reference Ray Direct Transport
(python/ray/experimental/gpu_object_manager/gpu_object_store.py holds
device tensors per owner actor; transfer orchestration
gpu_object_manager.py:419; wire enum common.proto:736 TensorTransport).
MI355X-native transport: instead of NCCL send/recv per tensor, the producer
exports a hipIpcMemHandle and the consumer maps the producer's HBM pages
directly over xGMI (hipIpcOpenMemHandle w/ lazy peer access). The producer
pins the tensor in its GPUObjectStore until the object is freed.

Supported transports (tensor_transport= on @ray.method / ray.put):
  * "object_store" (default) — GPU tensors are copied to CPU and travel
    through the shm object store (restored as CPU tensors, reference default)
  * "hip_ipc" — intra-node zero-copy device sharing (this module)
  * "collective" — RCCL send/recv via ray.util.collective groups (caller
    orchestrated, see util/collective)
"""
from __future__ import annotations

import threading
from typing import Dict, List

import torch


class GPUObjectStore:
    """Per-process registry pinning exported device tensors."""

    def __init__(self):
        self._lock = threading.Lock()
        self._objects: Dict[bytes, List[torch.Tensor]] = {}

    def add(self, key: bytes, tensors: List[torch.Tensor]):
        with self._lock:
            self._objects.setdefault(key, []).extend(tensors)

    def get(self, key: bytes):
        with self._lock:
            return list(self._objects.get(key, []))

    def has(self, key: bytes) -> bool:
        with self._lock:
            return key in self._objects

    def free(self, keys: List[bytes]):
        with self._lock:
            for k in keys:
                self._objects.pop(k, None)

    def num_objects(self) -> int:
        with self._lock:
            return len(self._objects)

    def total_bytes(self) -> int:
        with self._lock:
            return sum(
                t.numel() * t.element_size()
                for ts in self._objects.values()
                for t in ts
            )


gpu_object_store = GPUObjectStore()


def export_tensor(t: torch.Tensor) -> dict:
    """Produce a hip_ipc descriptor for a device tensor (pins it)."""
    from ant_ray_amd import _gpu_ipc

    t = t.detach()
    if not t.is_contiguous():
        t = t.contiguous()
    handle, offset, _size = _gpu_ipc.export_handle(t)
    return {
        "handle": bytes(handle),
        "offset": int(offset),
        "shape": tuple(t.shape),
        "dtype": str(t.dtype),
        "device": int(t.device.index or 0),
    }, t


def import_tensor(meta: dict) -> torch.Tensor:
    from ant_ray_amd import _gpu_ipc

    if not torch.cuda.is_available():
        raise RuntimeError(
            "cannot materialize a hip_ipc GPU object in a process without GPU "
            "access (cross-node hip_ipc is not supported; use "
            "tensor_transport='object_store' or 'collective')"
        )
    dtype = getattr(torch, meta["dtype"].replace("torch.", ""))
    device = torch.cuda.current_device()
    return _gpu_ipc.import_handle(
        meta["handle"], meta["offset"], list(meta["shape"]), dtype, device
    )


_import_mode = threading.local()


class gpu_import_clone:
    """While active, hip_ipc descriptors deserialized in this thread
    materialize as LOCAL device clones (and the peer mapping is released
    immediately) instead of zero-copy views. Compiled-DAG channel reads
    use this: the producer's buffer is mutable (rotating pin window), so
    the reader must own its copy. Sets `.cloned` when any import happened
    so the caller can fence the async device copies."""

    def __enter__(self):
        _import_mode.clone = True
        _import_mode.cloned = False
        return self

    def __exit__(self, *a):
        _import_mode.clone = False
        self.cloned = getattr(_import_mode, "cloned", False)
        _import_mode.cloned = False


def _rebuild_gpu_tensor(meta: dict):
    view = import_tensor(meta)
    if getattr(_import_mode, "clone", False):
        out = view.clone()
        del view  # drop the peer mapping refcount now
        _import_mode.cloned = True
        return out
    return view
