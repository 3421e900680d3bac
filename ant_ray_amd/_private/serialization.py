"""Object serialization: cloudpickle + protocol-5 out-of-band buffers.

Role parity: reference python/ray/_private/serialization.py (cloudpickle +
Arrow zero-copy buffers + out-of-band tensor hooks). Design here:

  * Values are pickled with `cloudpickle` at protocol 5; large contiguous
    buffers (numpy arrays, torch CPU tensor storage, bytes>threshold) are
    collected out-of-band via `buffer_callback`.
  * The stored wire format keeps buffers 64-byte aligned so a reader can map
    them zero-copy straight out of the shm segment (numpy arrays deserialized
    from the object store alias shm memory, read-only — same behavior as the
    reference's plasma-backed numpy reads).
  * ObjectRefs nested inside values are serialized by binary id + owner
    address and re-hydrated on read; the contained-ref list travels in the
    header so the runtime can track borrows.

Layout of a serialized object:
    u32 magic | u32 n_buffers | u64 pickle_size
    (u64 offset, u64 size) * n_buffers      # relative to start of data area
    pickle bytes | padding | buffer0 | padding | buffer1 ...
"""
from __future__ import annotations

import pickle
import struct
from typing import Any, List, Optional, Tuple

import cloudpickle

MAGIC = 0x41525A31  # "ARZ1"
_HDR = struct.Struct("<IIQ")
_BUF = struct.Struct("<QQ")
ALIGN = 64

# Metadata tags (stored in the shm object's metadata field).
META_PICKLE = b"py"
META_ERROR = b"err"
META_RAW = b"raw"  # plain bytes payload, no pickle
META_ACTOR_DIED = b"actor_died"
META_GPU = b"gpu"  # GPU object: payload is a descriptor, tensors live on-device


def _align(n: int) -> int:
    return (n + ALIGN - 1) & ~(ALIGN - 1)


class SerializedObject:
    """A pickled value plus its out-of-band buffers, ready to write."""

    __slots__ = ("pickle_bytes", "buffers", "contained_refs", "total_size", "metadata")

    def __init__(self, pickle_bytes: bytes, buffers: List[memoryview], contained_refs, metadata=META_PICKLE):
        self.pickle_bytes = pickle_bytes
        self.buffers = buffers
        self.contained_refs = contained_refs
        self.metadata = metadata
        hdr = _HDR.size + _BUF.size * len(buffers)
        off = _align(hdr + len(pickle_bytes))
        for b in buffers:
            off = _align(off + b.nbytes)
        self.total_size = off

    def write_into(self, dest: memoryview):
        """Write the full serialized form into `dest` (length >= total_size)."""
        n = len(self.buffers)
        hdr_size = _HDR.size + _BUF.size * n
        _HDR.pack_into(dest, 0, MAGIC, n, len(self.pickle_bytes))
        off = _align(hdr_size + len(self.pickle_bytes))
        pos = _HDR.size
        offsets = []
        for b in self.buffers:
            offsets.append((off, b.nbytes))
            _BUF.pack_into(dest, pos, off, b.nbytes)
            pos += _BUF.size
            off = _align(off + b.nbytes)
        dest[hdr_size : hdr_size + len(self.pickle_bytes)] = self.pickle_bytes
        for (boff, bsize), b in zip(offsets, self.buffers):
            dest[boff : boff + bsize] = b  # PickleBuffer.raw() views are 1-D bytes

    def to_bytes(self) -> bytes:
        out = bytearray(self.total_size)
        self.write_into(memoryview(out))
        return bytes(out)


def _torch_cpu_tensor_reducer(tensor):
    """Out-of-band reducer for CPU torch tensors (dense, contiguous path)."""
    import torch

    if tensor.device.type != "cpu" or not tensor.is_contiguous() or tensor.is_sparse:
        # fall back to torch's own reducer
        return NotImplemented
    # byte view works for every dtype incl. bf16 (numpy has no bfloat16)
    np_view = tensor.detach().reshape(-1).view(torch.uint8).numpy()
    return (
        _rebuild_torch_tensor,
        (pickle.PickleBuffer(np_view), str(tensor.dtype), tuple(tensor.shape)),
    )


def _rebuild_torch_tensor(buf, dtype_str, shape):
    import warnings

    import numpy as np
    import torch

    dtype = getattr(torch, dtype_str.replace("torch.", ""))
    mv = buf.raw() if isinstance(buf, pickle.PickleBuffer) else memoryview(buf)
    np_arr = np.frombuffer(mv, dtype=np.uint8)
    with warnings.catch_warnings():
        # zero-copy view over (possibly read-only) shm memory; writes would
        # corrupt the store, so the returned tensor must be treated immutable
        warnings.simplefilter("ignore")
        t = torch.from_numpy(np_arr)
    return t.view(dtype).reshape(shape)


# ray.util.register_serializer registry (exact-type match, like the
# reference's per-class cloudpickle dispatch registration)
_custom_serializers = {}


def _apply_custom_deserializer(deser, data):
    return deser(data)


class _Pickler(cloudpickle.CloudPickler):
    def __init__(self, file, buffer_callback=None):
        super().__init__(file, protocol=5, buffer_callback=buffer_callback)
        self.contained_refs = []

    def reducer_override(self, obj):
        cs = _custom_serializers.get(type(obj))
        if cs is not None:
            ser, deser = cs
            return (_apply_custom_deserializer, (deser, ser(obj)))
        from ant_ray_amd._private.object_ref import ObjectRef

        if isinstance(obj, ObjectRef):
            self.contained_refs.append(obj)
            return (ObjectRef._rehydrate, (obj.binary(), obj.owner_addr))
        try:
            import sys

            if "torch" not in sys.modules:
                raise ImportError("torch not loaded")  # don't pay the import
            import torch

            if isinstance(obj, torch.Tensor):
                if obj.is_cuda:
                    ctx = getattr(_transport_ctx, "ctx", None)
                    if ctx is not None and ctx.mode == "hip_ipc":
                        from ant_ray_amd.experimental.gpu_object_manager.gpu_object_store import (
                            _rebuild_gpu_tensor,
                            export_tensor,
                        )

                        meta, pinned = export_tensor(obj)
                        ctx.pinned.append(pinned)
                        return (_rebuild_gpu_tensor, (meta,))
                    obj = obj.detach().cpu()
                r = _torch_cpu_tensor_reducer(obj)
                if r is not NotImplemented:
                    return r
            # torch.ops / torch.ops.aten are pseudo-modules (_Ops/_OpNamespace)
            # cloudpickle's submodule heuristic drags in whenever a function
            # names both "torch" and "ops"; pickle them by import reference
            import torch._ops as _t_ops

            if isinstance(obj, _t_ops._Ops):
                return (_reimport_torch_ops, ())
            if isinstance(obj, _t_ops._OpNamespace):
                return (_reimport_torch_op_namespace, (obj.name,))
        except ImportError:
            pass
        return super().reducer_override(obj)


def _reimport_torch_ops():
    import torch

    return torch.ops


def _reimport_torch_op_namespace(name):
    import torch

    return getattr(torch.ops, name)


_BUFFER_THRESHOLD = 512  # buffers below this get pickled in-band


import threading as _threading

_transport_ctx = _threading.local()


class gpu_transport_context:
    """While active, CUDA tensors serialized in this thread are exported via
    the given transport ("hip_ipc") instead of copied to CPU. Exported
    tensors are collected into `pinned` — the caller must register them in
    the GPUObjectStore under the object id."""

    def __init__(self, mode: str):
        self.mode = mode
        self.pinned = []

    def __enter__(self):
        _transport_ctx.ctx = self
        return self

    def __exit__(self, *a):
        _transport_ctx.ctx = None


_PLAIN_TYPES = (type(None), bool, int, float, str, bytes, bytearray)


def _is_plain_data(v, depth=0) -> bool:
    """True when the C-pickler fast path is SAFE: only data types whose plain
    pickling round-trips in any process. Functions/classes/arbitrary objects
    force the cloudpickle path (plain pickle would serialize a __main__
    function by reference, which a worker whose __main__ is default_worker
    cannot resolve)."""
    if isinstance(v, _PLAIN_TYPES):
        return True
    t = type(v)
    mod = t.__module__
    if mod == "numpy" or mod.startswith("numpy."):
        return True
    if mod == "torch" or mod.startswith("torch."):
        return t.__name__ in ("Tensor", "Parameter", "dtype", "Size")
    from ant_ray_amd._private.object_ref import ObjectRef

    if t is ObjectRef:
        return True
    if depth > 6:
        return False
    if t in (list, tuple, set, frozenset):
        return all(_is_plain_data(x, depth + 1) for x in v)
    if t is dict:
        return all(_is_plain_data(k, depth + 1) and _is_plain_data(x, depth + 1)
                   for k, x in v.items())
    return False


def _make_dispatch(contained_refs):
    """copyreg-style dispatch table for the C pickler fast path."""
    from ant_ray_amd._private.object_ref import ObjectRef

    table = {}

    def reduce_ref(obj):
        contained_refs.append(obj)
        return (ObjectRef._rehydrate, (obj.binary(), obj.owner_addr))

    table[ObjectRef] = reduce_ref
    try:
        import sys as _sys

        if "torch" not in _sys.modules:
            raise ImportError("torch not loaded in this process")
        import torch

        def reduce_tensor(t):
            if t.is_cuda:
                ctx = getattr(_transport_ctx, "ctx", None)
                if ctx is not None and ctx.mode == "hip_ipc":
                    from ant_ray_amd.experimental.gpu_object_manager.gpu_object_store import (
                        _rebuild_gpu_tensor,
                        export_tensor,
                    )

                    meta, pinned = export_tensor(t)
                    ctx.pinned.append(pinned)
                    return (_rebuild_gpu_tensor, (meta,))
                # default object-store transport: device -> host copy; the
                # reader gets a CPU tensor (reference default behavior)
                t = t.detach().cpu()
            r = _torch_cpu_tensor_reducer(t)
            if r is NotImplemented:
                return t.__reduce_ex__(5)
            return r

        table[torch.Tensor] = reduce_tensor
        try:
            from torch.nn import Parameter

            table[Parameter] = lambda t: t.__reduce_ex__(5)
        except ImportError:
            pass
    except ImportError:
        pass
    return table


def dumps_by_value(obj: Any) -> bytes:
    """cloudpickle-by-value with our reducers (torch.ops workaround, refs);
    for shipping user callables outside the task-arg path (e.g. Serve
    deployment bodies)."""
    import io

    f = io.BytesIO()
    _Pickler(f).dump(obj)
    return f.getvalue()


def serialize(value: Any, metadata: bytes = META_PICKLE) -> SerializedObject:
    import io

    buffers: List[memoryview] = []

    def cb(pb: pickle.PickleBuffer):
        mv = pb.raw()
        if mv.nbytes < _BUFFER_THRESHOLD:
            return True  # keep in-band
        buffers.append(mv)
        return False

    # Fast path: C pickler for verified plain-data payloads (ObjectRef /
    # numpy / torch tensors / basic containers). Anything else — functions,
    # classes, user objects — goes through cloudpickle so __main__-defined
    # code ships by value.
    contained_refs: list = []
    f = io.BytesIO()
    if _is_plain_data(value):
        try:
            p = pickle.Pickler(f, protocol=5, buffer_callback=cb)
            p.dispatch_table = _make_dispatch(contained_refs)
            p.dump(value)
            return SerializedObject(f.getvalue(), buffers, contained_refs,
                                    metadata)
        except (pickle.PicklingError, TypeError, AttributeError):
            pass
    buffers.clear()
    f = io.BytesIO()
    p = _Pickler(f, buffer_callback=cb)
    p.dump(value)
    return SerializedObject(f.getvalue(), buffers, p.contained_refs, metadata)


def deserialize(data: memoryview, metadata: bytes = META_PICKLE) -> Any:
    if metadata == META_RAW:
        return bytes(data)
    magic, n, psize = _HDR.unpack_from(data, 0)
    if magic != MAGIC:
        raise ValueError("corrupt serialized object")
    hdr_size = _HDR.size + _BUF.size * n
    buffers = []
    pos = _HDR.size
    for _ in range(n):
        off, size = _BUF.unpack_from(data, pos)
        pos += _BUF.size
        buffers.append(data[off : off + size])
    pbytes = data[hdr_size : hdr_size + psize]
    value = pickle.loads(pbytes, buffers=buffers)
    if metadata in (META_ERROR, META_ACTOR_DIED):
        # value is the exception instance to raise at the get() site
        return value
    return value


def serialize_error(exc: BaseException) -> SerializedObject:
    try:
        return serialize(exc, metadata=META_ERROR)
    except Exception:
        from ant_ray_amd.exceptions import RayTaskError

        fallback = RayTaskError(
            function_name=getattr(exc, "function_name", "unknown"),
            traceback_str=str(exc),
            cause=None,
        )
        return serialize(fallback, metadata=META_ERROR)
