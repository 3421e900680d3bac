"""Ray DAG tests: bind/execute for tasks + actors, InputNode, diamond
dependencies, MultiOutputNode, compiled DAG reuse."""
import time

import pytest


@pytest.fixture(scope="module")
def ray_mod():
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()  # never inherit another module's (possibly dying) session
    if not ray.is_initialized():
        ray.init(num_cpus=8)
    yield ray
    ray.shutdown()


def test_function_dag(ray_mod):
    ray = ray_mod

    @ray.remote
    def a(x):
        return x + 1

    @ray.remote
    def b(x):
        return x * 2

    @ray.remote
    def c(x, y):
        return x + y

    from ant_ray_amd.dag import InputNode

    with InputNode() as inp:
        dag = c.bind(a.bind(inp), b.bind(inp))
    assert ray.get(dag.execute(10)) == 31  # (10+1) + (10*2)
    assert ray.get(dag.execute(0)) == 1


def test_diamond_submits_shared_node_once(ray_mod):
    ray = ray_mod

    @ray.remote
    class Counter:
        def __init__(self):
            self.n = 0

        def bump(self, _x=None):
            self.n += 1
            return self.n

        def get(self):
            return self.n

    counter = Counter.remote()

    @ray.remote
    def passthrough(x):
        return x

    from ant_ray_amd.dag import InputNode, MultiOutputNode

    shared = counter.bump.bind()  # ActorHandle method .bind
    with InputNode() as inp:
        dag = MultiOutputNode([passthrough.bind(shared),
                               passthrough.bind(shared)])
    out = ray.get(dag.execute(None))
    assert out[0] == out[1]  # same upstream call, submitted once
    assert ray.get(counter.get.remote()) == 1


def test_class_node_dag(ray_mod):
    ray = ray_mod

    @ray.remote
    class Adder:
        def __init__(self, base):
            self.base = base

        def add(self, x):
            return self.base + x

    from ant_ray_amd.dag import InputNode

    adder = Adder.bind(100)
    with InputNode() as inp:
        dag = adder.add.bind(inp)
    assert ray.get(dag.execute(5)) == 105
    # actor persists across executes (state lives in one actor)
    assert ray.get(dag.execute(7)) == 107


def test_compiled_dag(ray_mod):
    ray = ray_mod

    @ray.remote
    class Stage:
        def __init__(self, mult):
            self.mult = mult

        def run(self, x):
            return x * self.mult

    from ant_ray_amd.dag import InputNode

    s1 = Stage.bind(2)
    s2 = Stage.bind(10)
    with InputNode() as inp:
        dag = s2.run.bind(s1.run.bind(inp))
    compiled = dag.experimental_compile()
    for i in range(5):
        assert ray.get(compiled.execute(i)) == i * 20
    compiled.teardown()


def test_input_attribute_node(ray_mod):
    ray = ray_mod

    @ray.remote
    def f(x):
        return x * 2

    from ant_ray_amd.dag import InputNode

    with InputNode() as inp:
        dag = f.bind(inp["k"])
    assert ray.get(dag.execute({"k": 21})) == 42


class TestCompiledChannelDAG:
    """aDAG channel mode: edges are mutable shm channels, actors run
    resident __adag_loop__ fibers (csrc ChanHeader + experimental/channel)."""

    def test_linear_chain(self, ray_mod):
        import ant_ray_amd as ray
        from ant_ray_amd.dag import InputNode

        @ray.remote
        class Adder:
            def __init__(self, k):
                self.k = k

            def add(self, x):
                return x + self.k

        a = Adder.remote(1)
        b = Adder.remote(100)
        with InputNode() as inp:
            dag = b.add.bind(a.add.bind(inp)).experimental_compile()
        assert dag._channel_mode, "single-node actor chain must use channels"
        for i in range(20):
            assert ray.get(dag.execute(i)) == i + 101
        dag.teardown()
        # existing handles survive teardown and are usable again
        assert ray.get(a.add.remote(5), timeout=30) == 6

    def test_diamond_multi_output_and_errors(self, ray_mod):
        import pytest as _pytest

        import ant_ray_amd as ray
        from ant_ray_amd.dag import InputNode, MultiOutputNode

        @ray.remote
        class W:
            def __init__(self, k):
                self.k = k

            def add(self, x):
                return x + self.k

            def mul2(self, x):
                return x * 2

            def combine(self, a, b):
                return a + b

            def boom(self, x):
                if x == 13:
                    raise ValueError("boom13")
                return x

        a = W.remote(1)
        b = W.remote(100)
        with InputNode() as inp:
            p = a.add.bind(inp["v"])
            q = a.mul2.bind(p)
            r = b.combine.bind(p, q)
            dag = MultiOutputNode([q, r]).experimental_compile()
        assert dag._channel_mode
        out = ray.get(dag.execute({"v": 5}))
        assert out == [12, 18]
        dag.teardown()

        # node exceptions surface at get() and the loop keeps running
        c = W.remote(0)
        with InputNode() as inp:
            d2 = c.boom.bind(inp).experimental_compile()
        assert d2._channel_mode
        assert ray.get(d2.execute(1)) == 1
        with _pytest.raises(ValueError, match="boom13"):
            ray.get(d2.execute(13))
        assert ray.get(d2.execute(2)) == 2
        d2.teardown()


def test_execute_async(ray_mod):
    """CompiledDAG.execute_async: awaitable results in submission order
    (parity compiled_dag_node.py execute_async / CompiledDAGFuture)."""
    import asyncio

    import ant_ray_amd as ray
    from ant_ray_amd.dag import InputNode

    @ray.remote
    class Adder:
        def add(self, x):
            return x + 100

    a = Adder.remote()
    with InputNode() as inp:
        dag = a.add.bind(inp).experimental_compile()

    async def main():
        out = []
        # channels are depth-1: keep at most two executions in flight
        fut = await dag.execute_async(0)
        for i in range(1, 4):
            nxt = await dag.execute_async(i)
            out.append(await fut)
            fut = nxt
        out.append(await fut)
        return out

    out = asyncio.run(main())
    assert out == [100, 101, 102, 103]
    dag.teardown()


def test_compiled_dag_actor_death_surfaces(ray_mod):
    """A compiled-DAG actor dying mid-run raises ActorDiedError at get()
    instead of hanging on the channel forever."""
    import os

    import ant_ray_amd as ray
    from ant_ray_amd.dag import InputNode
    from ant_ray_amd.exceptions import ActorDiedError

    @ray.remote
    class Stage:
        def step(self, x):
            if x == "die":
                os._exit(1)
            return x + 1

    s = Stage.remote()
    with InputNode() as inp:
        dag = s.step.bind(inp).experimental_compile()
    assert dag._channel_mode
    assert ray.get(dag.execute(1), timeout=60) == 2
    ref = dag.execute("die")
    t0 = time.time()
    with pytest.raises(Exception) as ei:
        ray.get(ref, timeout=90)
    assert time.time() - t0 < 60, "actor death took too long to surface"
    assert isinstance(ei.value, (ActorDiedError, RuntimeError)), ei.value


def test_dag_context_and_plot(ray_mod):
    """DAGContext env-config singleton; plot emits dot text (parity:
    dag/context.py:38, vis utils)."""
    import ant_ray_amd as ray
    from ant_ray_amd.dag import DAGContext, DAGInputData, plot

    ctx = DAGContext.get_current()
    assert ctx.buffer_size_bytes >= 1
    assert DAGContext.get_current() is ctx

    d = DAGInputData(1, 2, k=3)
    assert d[0] == 1 and d["k"] == 3

    @ray.remote
    def add(a, b):
        return a + b

    node = add.bind(add.bind(1, 2), 3)
    dot = plot(node)
    assert dot.startswith("digraph") and "FunctionNode" in dot
    assert ray.get(node.execute(), timeout=30) == 6
