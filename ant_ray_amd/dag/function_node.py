"""Back-compat import site used by RemoteFunction.bind (parity
python/ray/dag/function_node.py)."""
from ant_ray_amd.dag.node import FunctionNode  # noqa: F401
