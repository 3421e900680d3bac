"""Ray DAG tests: bind/execute for tasks + actors, InputNode, diamond
dependencies, MultiOutputNode, compiled DAG reuse."""
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
