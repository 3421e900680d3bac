"""Custom serializer registration.

Role parity: reference python/ray/util/serialization.py
(register_serializer:7 / deregister_serializer:25). Registered pairs are
consulted by the cloudpickle reducer for EXACT type matches; the
deserializer is itself pickled into the stream, so workers need no
registration of their own.
"""


def register_serializer(cls: type, *, serializer, deserializer):
    """Serialize instances of `cls` with `serializer(obj) -> state` and
    rebuild them with `deserializer(state)` (which must be picklable)."""
    from ant_ray_amd._private.serialization import _custom_serializers

    _custom_serializers[cls] = (serializer, deserializer)


def deregister_serializer(cls: type):
    """Remove the custom serializer for `cls` (no-op if absent)."""
    from ant_ray_amd._private.serialization import _custom_serializers

    _custom_serializers.pop(cls, None)
