"""ray.runtime_context module (parity: reference runtime_context.py)."""
from ant_ray_amd import RuntimeContext, get_runtime_context  # noqa: F401
