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


def _now():
    import time

    return time.monotonic()


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
        return CompiledDAG(
            self,
            _buffer_size_bytes=kwargs.get("_buffer_size_bytes", 1 << 20))

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


class CompiledDAGRef:
    """Result handle of a channel-mode execute (parity: CompiledDAGRef).
    ray.get() on it drains the output channel in execution order."""

    def __init__(self, dag: "CompiledDAG", seq: int):
        self._dag = dag
        self._seq = seq

    def get(self, timeout=None):
        return self._dag._result_for(self._seq, timeout)


class CompiledDAG:
    """Parity: compiled_dag_node.py:805 CompiledDAG.

    Channel mode (the aDAG fast path): when the graph is actor-method
    nodes on one node, every edge becomes a mutable shm channel
    (`experimental/channel.py`) and each actor runs a resident
    `__adag_loop__` (task_executor._adag_loop) — execute() is then one
    channel write + one channel read instead of per-node RPCs. Falls back
    to re-submitting actor calls per execute (the pre-compile path) for
    graphs with task nodes, multi-node clusters, or no local shm.
    """

    def __init__(self, root: DAGNode, _buffer_size_bytes: int = 1 << 20):
        self._root = root
        self._buffer_size = _buffer_size_bytes
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
        self._channel_mode = False
        self._torn_down = False
        self._seq = 0
        self._read_seq = 0
        self._results: dict = {}
        import threading

        # serializes the output-channel drain: concurrent ray.get on two
        # CompiledDAGRefs (execute_async) must not interleave channel reads
        self._result_lock = threading.Lock()
        try:
            self._channel_mode = self._compile_channels()
        except Exception:
            self._channel_mode = False

    # ------------------------------------------------------- channel compile
    def _compile_channels(self) -> bool:
        import ant_ray_amd as ray
        from ant_ray_amd._private.worker import global_worker
        from ant_ray_amd.experimental.channel import Channel

        cw = global_worker.core_worker
        if cw is None or cw.store.shm is None:
            return False
        if len([n for n in ray.nodes() if n.get("Alive")]) != 1:
            return False  # shm channels are single-node

        # classify the graph: method nodes + input (+ MultiOutputNode root)
        outputs: List[DAGNode]
        if isinstance(self._root, MultiOutputNode):
            outputs = [a for a in self._root._bound_args]
        else:
            outputs = [self._root]
        method_nodes: List[ClassMethodNode] = []
        input_chan_nodes: list = []  # InputNode / InputAttributeNode
        seen = set()
        stack = list(outputs)
        while stack:
            n = stack.pop()
            if id(n) in seen:
                continue
            seen.add(id(n))
            if isinstance(n, ClassMethodNode):
                method_nodes.append(n)
                for a in n._bound_args:
                    if isinstance(a, DAGNode):
                        stack.append(a)
                if n._bound_kwargs and any(
                        isinstance(v, DAGNode)
                        for v in n._bound_kwargs.values()):
                    return False
            elif isinstance(n, (InputNode, InputAttributeNode)):
                input_chan_nodes.append(n)
            else:
                return False
        if not method_nodes:
            return False
        if any(not isinstance(o, ClassMethodNode) for o in outputs):
            return False

        # every node must consume at least one channel/upstream value, else
        # its resident loop would produce unboundedly — leave such on RPC
        for m in method_nodes:
            if not any(isinstance(a, DAGNode) for a in m._bound_args):
                return False

        # topological order of method nodes (DFS postorder from outputs)
        order: List[ClassMethodNode] = []
        marked: set = set()

        def visit(n):
            if id(n) in marked or not isinstance(n, ClassMethodNode):
                return
            marked.add(id(n))
            for a in n._bound_args:
                if isinstance(a, DAGNode):
                    visit(a)
            order.append(n)

        for o in outputs:
            visit(o)

        def actor_key(m: ClassMethodNode):
            return bytes(m._class_node._get_actor()._ray_actor_id)

        # which ACTORS (or the driver) consume each produced value — a
        # channel is only needed for cross-actor/driver edges; same-actor
        # edges pass through the loop's local value map
        out_ids = {id(o) for o in outputs}
        readers: dict = {}  # "input" | id(node) -> set(actor_key | b"driver")
        for m in order:
            ak = actor_key(m)
            for a in m._bound_args:
                if isinstance(a, (InputNode, InputAttributeNode)):
                    readers.setdefault("input", set()).add(ak)
                elif isinstance(a, ClassMethodNode):
                    if actor_key(a) != ak:
                        readers.setdefault(id(a), set()).add(ak)
        for o in outputs:
            readers.setdefault(id(o), set()).add(b"driver")

        n_input_readers = len(readers.get("input", ()))
        self._in_chan = (Channel(self._buffer_size, n_input_readers)
                         if n_input_readers else None)
        chan_of: dict = {}
        for m in order:
            n_readers = len(readers.get(id(m), ()))
            chan_of[id(m)] = (Channel(self._buffer_size, n_readers)
                              if n_readers else None)

        # ONE resident loop per actor running all its ops in topo order
        # (parity: compiled_dag_node do_exec_tasks — an actor is dedicated
        # to the DAG and executes its bound tasks every iteration)
        specs_by_actor: dict = {}
        handles: dict = {}
        for m in order:
            ak = actor_key(m)
            handles[ak] = m._class_node._get_actor()
            ins = []
            for a in m._bound_args:
                if isinstance(a, InputNode):
                    ins.append(("chan", self._in_chan))
                elif isinstance(a, InputAttributeNode):
                    ins.append(("chan_key", self._in_chan, a._key))
                elif isinstance(a, ClassMethodNode):
                    if actor_key(a) == ak:
                        ins.append(("local", id(a)))
                    else:
                        ins.append(("chan", chan_of[id(a)]))
                else:
                    ins.append(("const", a))
            specs_by_actor.setdefault(ak, []).append(
                {"method": m._method, "key": id(m), "ins": ins,
                 "out": chan_of[id(m)]})

        self._loop_refs = []
        for ak, ops in specs_by_actor.items():
            refs = handles[ak]._actor_method_call(
                "__adag_loop__", ({"ops": ops},), {}, {})
            self._loop_refs.append(refs[0] if isinstance(refs, list) else refs)
        self._out_chans = [chan_of[id(o)] for o in outputs]
        self._multi = isinstance(self._root, MultiOutputNode)
        self._all_chans = [c for c in chan_of.values() if c is not None]
        if self._in_chan is not None:
            self._all_chans.append(self._in_chan)
        self._dag_actor_ids = {actor_key(m) for m in order}
        return True

    def _check_dag_actors_alive(self):
        """Raise if a participating actor died (its resident loop is
        gone, so channel reads would block forever)."""
        from ant_ray_amd._private.worker import global_worker
        from ant_ray_amd.exceptions import ActorDiedError

        cw = global_worker.core_worker
        for aid in getattr(self, "_dag_actor_ids", ()):  # pubsub-updated
            st = cw._actors.get(aid)
            if st is not None and st.state == "DEAD":
                self.teardown()
                raise ActorDiedError(
                    f"compiled DAG actor {aid.hex()[:8]} died: "
                    f"{st.death_cause}")

    # --------------------------------------------------------------- execute
    def execute(self, *args, **kwargs):
        if not self._channel_mode:
            return self._root.execute(*args, **kwargs)
        if self._torn_down:
            raise RuntimeError("compiled DAG was torn down")
        if len(args) == 1 and not kwargs:
            value = args[0]
        elif not args and not kwargs:
            value = None
        else:
            value = _DagInput(args, kwargs)
        if self._in_chan is not None:
            self._in_chan.write(value, timeout=60)
        self._seq += 1
        return CompiledDAGRef(self, self._seq)

    async def execute_async(self, *args, **kwargs):
        """Asyncio variant (parity: CompiledDAG.execute_async returning an
        awaitable future): returns an awaitable that resolves to the
        execution's result without blocking the event loop."""
        import asyncio

        ref = self.execute(*args, **kwargs)
        loop = asyncio.get_running_loop()

        async def _await_result():
            import ant_ray_amd as ray

            return await loop.run_in_executor(None, lambda: ray.get(ref))

        return asyncio.ensure_future(_await_result())

    def _result_for(self, seq: int, timeout=None):
        from ant_ray_amd.experimental.channel import _WrappedError

        with self._result_lock:
            while self._read_seq < seq:
                # read unwrapped: an error result must still advance the
                # read cursor, else the next get() desynchronizes from the
                # channel. Reads poll in 1s slices so a DEAD actor (whose
                # loop can never write) surfaces as ActorDiedError instead
                # of an indefinite hang.
                vals = []
                for c in self._out_chans:
                    deadline = (None if timeout is None
                                else _now() + timeout)
                    while True:
                        slice_t = 1.0
                        if deadline is not None:
                            slice_t = min(slice_t, max(0.0, deadline - _now()))
                        try:
                            vals.append(c.read(slice_t, unwrap=False))
                            break
                        except Exception as e:
                            if "timed out" not in str(e):
                                raise
                            self._check_dag_actors_alive()
                            if deadline is not None and _now() >= deadline:
                                raise
                self._read_seq += 1
                self._results[self._read_seq] = (
                    vals if self._multi else vals[0])
                # only the latest few results are retained
                self._results.pop(self._read_seq - 8, None)
            try:
                out = self._results[seq]
            except KeyError:
                raise RuntimeError("compiled DAG result no longer buffered")
        if isinstance(out, _WrappedError):
            raise out.exc
        if isinstance(out, list):
            for v in out:
                if isinstance(v, _WrappedError):
                    raise v.exc
        return out

    def teardown(self):
        import ant_ray_amd as ray

        if self._channel_mode and not self._torn_down:
            self._torn_down = True
            if self._in_chan is not None:
                self._in_chan.close()
            for ref in self._loop_refs:
                try:
                    ray.get(ref, timeout=10)
                except Exception:
                    pass
            for c in self._all_chans:
                c.destroy()
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
