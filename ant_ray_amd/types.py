"""ray.types (parity: reference types.py — ObjectRef type export)."""
from ant_ray_amd._private.object_ref import ObjectRef  # noqa: F401
