"""Core API semantics, local mode + distributed single node.

Modeled on the reference's python/ray/tests/test_basic*.py coverage.
"""
import time

import numpy as np
import pytest


class TestLocalMode:
    def test_task_roundtrip(self, ray_local):
        ray = ray_local

        @ray.remote
        def f(x):
            return x + 1

        assert ray.get(f.remote(1)) == 2

    def test_actor_roundtrip(self, ray_local):
        ray = ray_local

        @ray.remote
        class C:
            def __init__(self, v):
                self.v = v

            def add(self, k):
                self.v += k
                return self.v

        c = C.remote(5)
        assert ray.get(c.add.remote(3)) == 8

    def test_error(self, ray_local):
        ray = ray_local

        @ray.remote
        def boom():
            raise KeyError("nope")

        with pytest.raises(KeyError):
            ray.get(boom.remote())

    def test_multi_returns(self, ray_local):
        ray = ray_local

        @ray.remote(num_returns=2)
        def two():
            return 1, 2

        a, b = two.remote()
        assert ray.get([a, b]) == [1, 2]

    def test_nested_refs(self, ray_local):
        ray = ray_local
        r = ray.put(41)

        @ray.remote
        def f(x):
            return x + 1

        assert ray.get(f.remote(r)) == 42


@pytest.mark.usefixtures("ray_start_regular_module")
class TestDistributed:
    def test_tasks(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        def sq(x):
            return x * x

        assert ray.get([sq.remote(i) for i in range(10)]) == [i * i for i in range(10)]

    def test_put_get_large(self, ray_start_regular_module):
        ray = ray_start_regular_module
        arr = np.random.rand(500_000)
        ref = ray.put(arr)
        out = ray.get(ref)
        assert np.array_equal(out, arr)

    def test_task_large_arg_and_return(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        def double(a):
            return a * 2

        arr = np.ones(400_000)
        out = ray.get(double.remote(ray.put(arr)))
        assert out.sum() == 800_000

    def test_actor_state_and_order(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        class Counter:
            def __init__(self):
                self.n = 0

            def incr(self):
                self.n += 1
                return self.n

        c = Counter.remote()
        results = ray.get([c.incr.remote() for _ in range(20)])
        assert results == list(range(1, 21))

    def test_actor_error_propagation(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        class Bad:
            def fail(self):
                raise ValueError("actor boom")

        b = Bad.remote()
        with pytest.raises(ValueError):
            ray.get(b.fail.remote())

    def test_actor_init_failure(self, ray_start_regular_module):
        ray = ray_start_regular_module
        from ant_ray_amd.exceptions import RayActorError

        @ray.remote
        class BadInit:
            def __init__(self):
                raise RuntimeError("init boom")

            def m(self):
                return 1

        b = BadInit.remote()
        with pytest.raises(RayActorError):
            ray.get(b.m.remote(), timeout=30)

    def test_wait(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        def fast():
            return 1

        @ray.remote
        def slow():
            time.sleep(20)
            return 2

        ray.get(fast.remote())  # warm the lease so timing is deterministic
        a, b = fast.remote(), slow.remote()
        ready, not_ready = ray.wait([a, b], num_returns=1, timeout=10)
        assert ready == [a] and not_ready == [b]

    def test_get_timeout(self, ray_start_regular_module):
        ray = ray_start_regular_module
        from ant_ray_amd.exceptions import GetTimeoutError

        @ray.remote
        def slow():
            time.sleep(10)

        with pytest.raises(GetTimeoutError):
            ray.get(slow.remote(), timeout=0.5)

    def test_named_actor(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        class Reg:
            def who(self):
                return "reg"

        keep = Reg.options(name="registry").remote()  # keep a handle: a
        # non-detached actor whose handles all drop is terminated (the
        # out-of-scope kill now also covers still-scheduling actors)
        h = ray.get_actor("registry")
        assert ray.get(h.who.remote()) == "reg"

    def test_async_actor(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        class Async:
            async def work(self, x):
                import asyncio

                await asyncio.sleep(0.01)
                return x * 10

        a = Async.remote()
        assert ray.get([a.work.remote(i) for i in range(5)]) == [0, 10, 20, 30, 40]

    def test_threaded_actor(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote(max_concurrency=4)
        class T:
            def slow(self):
                time.sleep(0.3)
                return 1

        t = T.remote()
        ray.get(t.slow.remote())  # warm: actor worker spawn happens here
        t0 = time.time()
        assert sum(ray.get([t.slow.remote() for _ in range(4)])) == 4
        assert time.time() - t0 < 1.1  # concurrent, not 1.2s serial

    def test_nested_tasks(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        def inner(x):
            return x + 1

        @ray.remote
        def outer(x):
            import ant_ray_amd as ray2

            return ray2.get(inner.remote(x)) + 100

        assert ray.get(outer.remote(1)) == 102

    def test_actor_handle_passing(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        class Holder:
            def __init__(self):
                self.v = 7

            def get(self):
                return self.v

        @ray.remote
        def reader(h):
            import ant_ray_amd as ray2

            return ray2.get(h.get.remote())

        h = Holder.remote()
        assert ray.get(reader.remote(h)) == 7

    def test_cluster_resources(self, ray_start_regular_module):
        ray = ray_start_regular_module
        assert ray.cluster_resources().get("CPU", 0) >= 4

    def test_runtime_context(self, ray_start_regular_module):
        ray = ray_start_regular_module

        @ray.remote
        def ctx():
            import ant_ray_amd as ray2

            c = ray2.get_runtime_context()
            return (c.get_worker_id(), c.get_task_id())

        wid, tid = ray.get(ctx.remote())
        assert wid and tid


class TestObjectStorePressure:
    def test_spill_and_restore(self):
        """Primary copies are pinned while referenced; under memory pressure
        the owner SPILLS its oldest objects to disk (LocalObjectManager
        parity) and gets restore them transparently."""
        import numpy as np

        import ant_ray_amd as ray

        if ray.is_initialized():
            ray.shutdown()
        ray.init(num_cpus=2, object_store_memory=200 * 1024 * 1024)
        try:
            refs = [ray.put(np.full(10 * 1024 * 1024, i % 251, dtype=np.uint8))
                    for i in range(30)]  # 300 MB into a 200 MB store
            for i in (0, 10, 29):
                v = ray.get(refs[i], timeout=30)
                assert v[0] == i % 251 and v.nbytes == 10 * 1024 * 1024
            from ant_ray_amd._private.worker import global_worker

            assert global_worker.core_worker._spilled, "spill should have run"
        finally:
            ray.shutdown()

    def test_dropped_refs_free_store(self):
        """Objects whose refs are dropped are freed, not spilled."""
        import numpy as np

        import ant_ray_amd as ray

        if ray.is_initialized():
            ray.shutdown()
        ray.init(num_cpus=2, object_store_memory=200 * 1024 * 1024)
        try:
            for _ in range(30):
                ref = ray.put(np.zeros(10 * 1024 * 1024, dtype=np.uint8))
                del ref  # ref dropped -> owner unpins + frees
            from ant_ray_amd._private.worker import global_worker

            assert not global_worker.core_worker._spilled
        finally:
            ray.shutdown()


def test_kill_actor_while_scheduling(ray_start_regular):
    """ray.kill (or the last handle dropping) on an actor that is still
    WAITING for resources must terminate the scheduling loop — not leave
    a zombie that later grabs the lease and holds its resources forever
    (the shared-GPU-session wedge: GC'd fractional-GPU actors kept
    starving the next test's lease)."""
    import time

    import ant_ray_amd as ray
    from ant_ray_amd.util import state

    @ray.remote(resources={"no_such_resource": 1})
    class Stuck:
        def ping(self):
            return 1

    a = Stuck.remote()
    aid = a._ray_actor_id
    time.sleep(0.3)  # let it enter the scheduling loop
    ray.kill(a)

    deadline = time.time() + 15
    dead = False
    while time.time() < deadline:
        rows = [r for r in state.list_actors() if r["actor_id"] == aid.hex()]
        if rows and rows[0]["state"] == "DEAD":
            dead = True
            break
        time.sleep(0.2)
    assert dead, "killed-while-scheduling actor should transition to DEAD"


def test_actor_task_ids_unique_across_callers(ray_start_regular):
    """Two different callers of one actor must not collide return-object
    ids: big (shm-resident) results from distinct callers stay distinct
    even at identical per-caller call counts."""
    import numpy as np

    ray = ray_start_regular

    @ray.remote
    class Producer:
        def big(self, fill):
            return np.full(200_000, float(fill))  # 1.6 MB -> shm

    @ray.remote
    class CallerA:
        def run(self, p):
            import numpy as _np

            out = ray.get(p.big.remote(1.0))
            return float(out.sum())

    @ray.remote
    class CallerB:
        def run(self, p):
            import numpy as _np

            out = ray.get(p.big.remote(2.0))
            return float(out.sum())

    p = Producer.remote()
    a, b = CallerA.remote(), CallerB.remote()
    # same call-count (1) from both callers, nearly simultaneous
    ra = a.run.remote(p)
    rb = b.run.remote(p)
    sa, sb = ray.get([ra, rb], timeout=120)
    assert sa == 200_000 * 1.0, sa
    assert sb == 200_000 * 2.0, sb


def test_out_of_scope_actor_drains_submitted_tasks(ray_start_regular):
    """Dropping the last handle to an actor with submitted tasks must NOT
    kill it mid-call: the out-of-scope teardown waits for every submitted
    task to complete (reference queues __ray_terminate__ behind them).
    Regression for the r01 hipIpc mixed-payload failure — the same
    chained-temporaries pattern, CPU-only."""
    ray = ray_start_regular

    @ray.remote
    class P:
        def make(self):
            time.sleep(0.2)  # widen the window the old code lost the race in
            return {"w": [1.0] * 256, "meta": "hello", "n": 5}

    @ray.remote
    class C:
        def read(self, d):
            time.sleep(0.2)
            return (sum(d["w"]), d["meta"], d["n"])

    # both handles are temporaries: refcount drops to zero right after
    # submission, long before the tasks run
    out = ray.get(C.remote().read.remote(P.remote().make.remote()), timeout=60)
    assert out == (256.0, "hello", 5)


def test_out_of_scope_actor_eventually_dies(ray_start_regular):
    """After the drained teardown, the actor must actually terminate (no
    leak): its state reaches DEAD within a bounded wait."""
    ray = ray_start_regular
    from ant_ray_amd.util import state as ray_state

    @ray.remote
    class Once:
        def ping(self):
            return "pong"

    ref = Once.remote().ping.remote()  # handle is a temporary
    assert ray.get(ref, timeout=60) == "pong"
    deadline = time.time() + 30
    dead = False
    while time.time() < deadline:
        actors = ray_state.list_actors()
        states = {a.get("actor_id"): a.get("state") for a in actors}
        alive = [s for s in states.values() if s not in ("DEAD",)]
        if not alive:
            dead = True
            break
        time.sleep(0.3)
    assert dead, f"out-of-scope actor leaked: {states}"
