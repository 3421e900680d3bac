"""Streaming generator tasks: num_returns="streaming" -> ObjectRefGenerator.

Parity: reference streaming generators (python/ray/_raylet.pyx
ObjectRefGenerator; used by Serve response streaming and Data): each
yielded value becomes its own object, shipped to the caller AS PRODUCED
so consumption overlaps with the producing task."""
import time

import pytest


@pytest.fixture(scope="module")
def ray_mod():
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()
    ray.init(num_cpus=4)
    yield ray
    ray.shutdown()


def test_streaming_basic(ray_mod):
    ray = ray_mod

    @ray.remote(num_returns="streaming")
    def gen(n):
        for i in range(n):
            yield i * 10

    out = [ray.get(ref) for ref in gen.remote(5)]
    assert out == [0, 10, 20, 30, 40]


def test_streaming_overlap(ray_mod):
    """The first item must be consumable long before the generator task
    finishes (items stream as produced, not at task completion)."""
    ray = ray_mod

    @ray.remote(num_returns="streaming")
    def slow_gen():
        for i in range(4):
            yield i
            time.sleep(0.5)

    g = slow_gen.remote()
    t0 = time.time()
    first = ray.get(next(g))
    first_latency = time.time() - t0
    assert first == 0
    rest = [ray.get(r) for r in g]
    assert rest == [1, 2, 3]
    assert first_latency < 1.2, f"first item took {first_latency:.2f}s (no overlap)"


def test_streaming_large_items(ray_mod):
    """Yields above the inline threshold travel through the shm store."""
    import numpy as np

    ray = ray_mod

    @ray.remote(num_returns="streaming")
    def big_gen():
        for i in range(3):
            yield np.full(300_000, i, dtype=np.float64)  # 2.4 MB each

    sums = [float(ray.get(r).sum()) for r in big_gen.remote()]
    assert sums == [0.0, 300_000.0, 600_000.0]


def test_streaming_mid_generator_error(ray_mod):
    ray = ray_mod

    @ray.remote(num_returns="streaming")
    def flaky():
        yield 1
        yield 2
        raise ValueError("stream broke")

    g = flaky.remote()
    assert ray.get(next(g)) == 1
    assert ray.get(next(g)) == 2
    with pytest.raises(Exception) as ei:
        for _ in range(3):
            next(g)
    assert "stream broke" in str(ei.value)


def test_streaming_async_iteration(ray_mod):
    import asyncio

    ray = ray_mod

    @ray.remote(num_returns="streaming")
    def gen():
        yield "a"
        yield "b"

    async def consume():
        out = []
        async for ref in gen.remote():
            out.append(ray.get(ref))
        return out

    assert asyncio.run(consume()) == ["a", "b"]


def test_streaming_actor_method(ray_mod):
    """@ray.method(num_returns="streaming") on an actor (the substrate the
    reference's Serve response streaming rides on)."""
    ray = ray_mod

    @ray.remote
    class Gen:
        @ray.method(num_returns="streaming")
        def chunks(self, n):
            for i in range(n):
                yield f"chunk-{i}"

    g = Gen.remote()
    out = [ray.get(r) for r in g.chunks.remote(4)]
    assert out == [f"chunk-{i}" for i in range(4)]
    # a second call streams independently
    assert [ray.get(r) for r in g.chunks.remote(2)] == ["chunk-0", "chunk-1"]


def test_zz_streaming_local_mode():
    # runs LAST: local_mode re-inits the global session, which would
    # kill the module fixture's cluster for later tests
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()
    ray.init(local_mode=True)

    @ray.remote(num_returns="streaming")
    def gen(n):
        for i in range(n):
            yield i + 100

    assert [ray.get(r) for r in gen.remote(3)] == [100, 101, 102]
    ray.shutdown()
