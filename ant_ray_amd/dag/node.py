"""DAG node types + execution.

Role parity: reference python/ray/dag/dag_node.py (DAGNode base),
function_node.py, class_node.py, input_node.py, output_node.py
(MultiOutputNode), compiled_dag_node.py (CompiledDAG; execute loop). Each
execute() walks the graph bottom-up, memoizing per-node results so diamond
dependencies submit once; results flow between tasks as ObjectRefs (never
materialized on the driver).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple


class DAGNode:
    def __init__(self, args: tuple, kwargs: dict):
        self._bound_args = args
        self._bound_kwargs = kwargs or {}

    # -------------------------------------------------------------- execute

    def execute(self, *input_args, **input_kwargs):
        """Submit the whole DAG; returns ObjectRef(s) of this node."""
        cache: Dict[int, Any] = {}
        return self._resolve(cache, input_args, input_kwargs)

    def _resolve(self, cache, input_args, input_kwargs):
        key = id(self)
        if key not in cache:
            cache[key] = self._submit(cache, input_args, input_kwargs)
        return cache[key]

    def _submit(self, cache, input_args, input_kwargs):
        raise NotImplementedError

    def _resolve_args(self, cache, input_args, input_kwargs) -> Tuple[tuple, dict]:
        def r(v):
            if isinstance(v, DAGNode):
                return v._resolve(cache, input_args, input_kwargs)
            return v

        return (tuple(r(a) for a in self._bound_args),
                {k: r(v) for k, v in self._bound_kwargs.items()})

    def experimental_compile(self, **kwargs) -> "CompiledDAG":
        return CompiledDAG(self)

    # --------------------------------------------------------- introspection

    def _upstream(self) -> List["DAGNode"]:
        out = []
        for v in list(self._bound_args) + list(self._bound_kwargs.values()):
            if isinstance(v, DAGNode):
                out.append(v)
        return out


class InputNode(DAGNode):
    """The DAG's runtime input placeholder (context-manager per parity)."""

    def __init__(self):
        super().__init__((), {})

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        return False

    def _submit(self, cache, input_args, input_kwargs):
        if len(input_args) == 1 and not input_kwargs:
            return input_args[0]
        if not input_args and not input_kwargs:
            return None
        return _DagInput(input_args, input_kwargs)

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        return InputAttributeNode(self, name)

    def __getitem__(self, key):
        return InputAttributeNode(self, key)


class _DagInput:
    def __init__(self, args, kwargs):
        self.args = args
        self.kwargs = kwargs


class InputAttributeNode(DAGNode):
    """inp.x / inp[0] — projects a field of the runtime input."""

    def __init__(self, input_node: InputNode, key):
        super().__init__((), {})
        self._input_node = input_node
        self._key = key

    def _submit(self, cache, input_args, input_kwargs):
        if isinstance(self._key, int):
            return input_args[self._key]
        if self._key in input_kwargs:
            return input_kwargs[self._key]
        # attribute access on a single positional object
        if len(input_args) == 1:
            obj = input_args[0]
            if isinstance(obj, dict):
                return obj[self._key]
            return getattr(obj, self._key)
        raise KeyError(self._key)


class FunctionNode(DAGNode):
    def __init__(self, remote_fn, args, kwargs):
        super().__init__(args, kwargs)
        self._fn = remote_fn

    def _submit(self, cache, input_args, input_kwargs):
        args, kwargs = self._resolve_args(cache, input_args, input_kwargs)
        return self._fn.remote(*args, **kwargs)


class ClassNode(DAGNode):
    """Actor-to-be: materialized once per DAG (memoized across executes)."""

    def __init__(self, actor_cls, args, kwargs, opts):
        super().__init__(args, kwargs)
        self._actor_cls = actor_cls
        self._opts = opts or {}
        self._actor = None

    def _get_actor(self):
        if self._actor is None:
            cls = self._actor_cls
            if self._opts:
                self._actor = cls.options(**self._opts).remote(
                    *self._bound_args, **self._bound_kwargs)
            else:
                self._actor = cls.remote(*self._bound_args, **self._bound_kwargs)
        return self._actor

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        return _ClassMethodStub(self, name)

    def _submit(self, cache, input_args, input_kwargs):
        return self._get_actor()


class _ExistingActorShim:
    """Adapts a live ActorHandle to the ClassNode._get_actor protocol (for
    actor_handle.method.bind)."""

    def __init__(self, handle):
        self._handle = handle
        self._actor = handle

    def _get_actor(self):
        return self._handle

    def _upstream(self):
        return []


class _ClassMethodStub:
    def __init__(self, class_node: ClassNode, method: str):
        self._class_node = class_node
        self._method = method

    def bind(self, *args, **kwargs) -> "ClassMethodNode":
        return ClassMethodNode(self._class_node, self._method, args, kwargs)


class ClassMethodNode(DAGNode):
    def __init__(self, class_node: ClassNode, method: str, args, kwargs):
        super().__init__(args, kwargs)
        self._class_node = class_node
        self._method = method

    def _submit(self, cache, input_args, input_kwargs):
        actor = self._class_node._get_actor()
        args, kwargs = self._resolve_args(cache, input_args, input_kwargs)
        return getattr(actor, self._method).remote(*args, **kwargs)

    def _upstream(self):
        return super()._upstream() + [self._class_node]


class MultiOutputNode(DAGNode):
    def __init__(self, outputs: List[DAGNode]):
        super().__init__(tuple(outputs), {})

    def _submit(self, cache, input_args, input_kwargs):
        args, _ = self._resolve_args(cache, input_args, input_kwargs)
        return list(args)


class CompiledDAG:
    """Parity: compiled_dag_node.py:805. Pre-creates every ClassNode's actor
    and freezes the node topology; execute() then only submits actor calls /
    tasks in topological order. Returns a single ObjectRef (or list for
    MultiOutputNode) like the reference's CompiledDAGRef."""

    def __init__(self, root: DAGNode):
        self._root = root
        # materialize all actors up front
        seen = set()
        stack = [root]
        while stack:
            n = stack.pop()
            if id(n) in seen:
                continue
            seen.add(id(n))
            if isinstance(n, ClassNode):
                n._get_actor()
            if isinstance(n, ClassMethodNode):
                n._class_node._get_actor()
            stack.extend(n._upstream())

    def execute(self, *args, **kwargs):
        return self._root.execute(*args, **kwargs)

    def teardown(self):
        import ant_ray_amd as ray

        seen = set()
        stack = [self._root]
        while stack:
            n = stack.pop()
            if id(n) in seen:
                continue
            seen.add(id(n))
            if isinstance(n, ClassNode) and n._actor is not None:
                try:
                    ray.kill(n._actor)
                except Exception:
                    pass
                n._actor = None
            stack.extend(n._upstream())
