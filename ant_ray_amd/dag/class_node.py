"""Back-compat import site used by ActorClass.bind (parity
python/ray/dag/class_node.py)."""
from ant_ray_amd.dag.node import ClassNode  # noqa: F401
