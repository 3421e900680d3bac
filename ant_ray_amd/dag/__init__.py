"""ant_ray_amd.dag — Ray DAG API (lazy task/actor graphs + compiled graphs).

Role parity: reference python/ray/dag/ (dag_node.py, function_node.py,
class_node.py, input_node.py, compiled_dag_node.py:805 CompiledDAG).
`fn.bind()` / `Actor.bind()` / `actor.method.bind()` build the lazy graph;
`.execute(*args)` submits it; `experimental_compile()` pre-resolves the
actor topology + per-actor schedule so repeated executions skip graph
walking (the reference additionally swaps RPCs for mutable-plasma channels;
our actor path already goes through the shm store + hipIpc GPU tier, so the
compiled form reuses it).
"""
from ant_ray_amd.dag.node import (
    ClassMethodNode,
    ClassNode,
    DAGNode,
    FunctionNode,
    InputAttributeNode,
    InputNode,
    MultiOutputNode,
)

__all__ = [
    "ClassMethodNode", "ClassNode", "DAGNode", "FunctionNode",
    "InputAttributeNode", "InputNode", "MultiOutputNode",
]
