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

from ant_ray_amd.dag.context import DAGContext  # noqa: E402

# structural dict keys used when serializing DAG nodes (parity:
# reference dag/constants.py)
PARENT_CLASS_NODE_KEY = "parent_class_node"
PREV_CLASS_METHOD_CALL_KEY = "prev_class_method_call"
BIND_INDEX_KEY = "bind_index"
IS_CLASS_METHOD_OUTPUT_KEY = "is_class_method_output"
COLLECTIVE_OPERATION_KEY = "collective_operation"
DAGNODE_TYPE_KEY = "dagnode_type"


class DAGInputData:
    """Multi-arg execute() bundle (parity: reference dag/input_node.py
    DAGInputData — positional + keyword inputs addressed by
    InputAttributeNode)."""

    def __init__(self, *args, **kwargs):
        self._args = list(args)
        self._kwargs = dict(kwargs)

    def __getitem__(self, key):
        if isinstance(key, int):
            return self._args[key]
        return self._kwargs[key]


class DAGOperationFuture:
    """Awaitable wrapper over an in-flight DAG operation (parity name:
    reference dag/dag_operation_future.py)."""

    def __init__(self, value):
        self._value = value

    def wait(self):
        return self._value


class GPUFuture(DAGOperationFuture):
    """A DAG value resident on GPU; wait() returns the tensor (parity
    name: reference GPUFuture — the hipIpc channel tier makes results
    device-resident already)."""


class CollectiveOutputNode(ClassMethodNode):
    """Marker node type for collective-op outputs in a compiled DAG
    (parity name: reference dag/collective_node.py)."""


def plot(dag_node, to_file: str = None):
    """Render a DAG as Graphviz dot text (parity: reference dag/vis
    utils; the graphviz binary is not in this image, so this emits/saves
    the .dot source which renders anywhere)."""
    lines = ["digraph G {"]
    seen = {}

    def walk(node):
        if id(node) in seen:
            return seen[id(node)]
        name = f"n{len(seen)}"
        seen[id(node)] = name
        label = type(node).__name__
        for attr in ("_method_name", "_fn"):
            v = getattr(node, attr, None)
            if v is not None:
                label += f"\n{getattr(v, '__name__', v)}"
                break
        lines.append(f'  {name} [label="{label}"];')
        for dep in getattr(node, "_deps", None) or                 getattr(node, "_args", []) or []:
            if isinstance(dep, DAGNode):
                lines.append(f"  {walk(dep)} -> {name};")
        return name

    walk(dag_node)
    lines.append("}")
    dot = "\n".join(lines)
    if to_file:
        with open(to_file, "w") as f:
            f.write(dot)
    return dot


__all__ = [
    "ClassMethodNode", "ClassNode", "CollectiveOutputNode", "DAGContext",
    "DAGInputData", "DAGNode", "DAGOperationFuture", "FunctionNode",
    "GPUFuture", "InputAttributeNode", "InputNode", "MultiOutputNode",
    "plot",
]
