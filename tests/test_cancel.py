"""ray.cancel parity: queued tasks dropped, running tasks interrupted
(KeyboardInterrupt -> TaskCancelledError), force=True kills the worker."""
import time

import pytest


@pytest.fixture(scope="module")
def ray_mod():
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()
    ray.init(num_cpus=2)
    yield ray
    ray.shutdown()


def test_cancel_running_task(ray_mod):
    ray = ray_mod
    from ant_ray_amd.exceptions import TaskCancelledError

    @ray.remote(num_cpus=1)
    def sleeper():
        # short sleeps: the soft cancel lands at the next Python bytecode
        # boundary (a single 300s C sleep would need force=True)
        for _ in range(3000):
            time.sleep(0.1)
        return "never"

    ref = sleeper.remote()
    time.sleep(1.0)  # let it start
    t0 = time.time()
    ray.cancel(ref)
    with pytest.raises(TaskCancelledError):
        ray.get(ref, timeout=60)
    assert time.time() - t0 < 30


def test_cancel_queued_task(ray_mod):
    ray = ray_mod
    from ant_ray_amd.exceptions import TaskCancelledError

    @ray.remote(num_cpus=2)
    def hog():
        time.sleep(4)
        return "hog done"

    @ray.remote(num_cpus=2)
    def queued():
        return "queued ran"

    h = hog.remote()
    time.sleep(0.5)
    q = queued.remote()  # can't start: hog holds both CPUs
    time.sleep(0.3)
    ray.cancel(q)
    with pytest.raises(TaskCancelledError):
        ray.get(q, timeout=60)
    assert ray.get(h, timeout=60) == "hog done"


def test_cancel_force_kills_worker(ray_mod):
    ray = ray_mod
    from ant_ray_amd.exceptions import TaskCancelledError

    @ray.remote(num_cpus=1, max_retries=3)
    def stubborn():
        while True:  # ignores KeyboardInterrupt-level cancels promptly?
            try:
                time.sleep(300)
            except KeyboardInterrupt:
                time.sleep(300)  # swallow the soft cancel

    ref = stubborn.remote()
    time.sleep(1.0)
    ray.cancel(ref, force=True)
    with pytest.raises(TaskCancelledError):
        ray.get(ref, timeout=60)


def test_exit_actor(ray_mod):
    """ray.actor.exit_actor(): the call completes, the actor then dies;
    later calls raise the actor-died error (no restart: graceful exits
    don't count as failures)."""
    import ant_ray_amd as ray_

    ray = ray_mod

    @ray.remote(max_restarts=3)
    class Quitter:
        def ping(self):
            return "pong"

        def quit(self):
            ray_.actor.exit_actor()

    q = Quitter.remote()
    assert ray.get(q.ping.remote(), timeout=60) == "pong"
    ray.get(q.quit.remote(), timeout=60)  # reply arrives, then actor exits
    time.sleep(1.0)
    with pytest.raises(Exception):
        ray.get(q.ping.remote(), timeout=30)


def test_request_resources_sdk(ray_mod):
    """autoscaler.sdk.request_resources: standing demand visible to the
    autoscaler beyond free capacity."""
    from ant_ray_amd.autoscaler import NodeTypeConfig, StandardAutoscaler
    from ant_ray_amd.autoscaler.sdk import (
        get_requested_resources,
        request_resources,
    )

    class NullProvider:
        def non_terminated_nodes(self):
            return {}

        def create_node(self, cfg):
            pass

    request_resources(bundles=[{"CPU": 64}])  # far beyond the 2-CPU session
    assert get_requested_resources() == [{"CPU": 64}]
    scaler = StandardAutoscaler(
        {"big": NodeTypeConfig("big", {"CPU": 64}, max_workers=2)},
        NullProvider())
    demands = scaler.pending_demands()
    assert {"CPU": 64} in demands
    request_resources()  # clear
    assert get_requested_resources() == []


def test_concurrency_groups(ray_mod):
    """Named concurrency groups: a saturated 'compute' group must not
    block 'io' methods, and each group honors its own max_concurrency
    (parity: reference concurrency_group_manager)."""
    ray = ray_mod

    @ray.remote(concurrency_groups={"io": 2, "compute": 1})
    class Worker:
        def __init__(self):
            self.log = []

        @ray.method(concurrency_group="compute")
        def crunch(self):
            time.sleep(2.0)
            return "crunched"

        @ray.method(concurrency_group="io")
        def fetch(self, i):
            return f"fetched-{i}"

    w = Worker.remote()
    slow = w.crunch.remote()
    time.sleep(0.3)  # compute group now saturated
    t0 = time.time()
    fast = ray.get([w.fetch.remote(i) for i in range(4)], timeout=60)
    io_latency = time.time() - t0
    assert fast == [f"fetched-{i}" for i in range(4)]
    assert io_latency < 1.5, f"io group blocked behind compute ({io_latency:.2f}s)"
    assert ray.get(slow, timeout=60) == "crunched"


def test_cancel_async_actor_task(ray_mod):
    """ray.cancel on a RUNNING async actor method cancels its coroutine;
    queued calls behind it cancel before starting."""
    ray = ray_mod
    from ant_ray_amd.exceptions import TaskCancelledError

    @ray.remote
    class AsyncWorker:
        async def long(self):
            import asyncio

            await asyncio.sleep(300)
            return "never"

        async def quick(self):
            return "quick"

    a = AsyncWorker.remote()
    ref = a.long.remote()
    time.sleep(1.0)
    ray.cancel(ref)
    t0 = time.time()
    with pytest.raises(TaskCancelledError):
        ray.get(ref, timeout=60)
    assert time.time() - t0 < 30
    # the actor stays healthy for later calls
    assert ray.get(a.quick.remote(), timeout=60) == "quick"


def test_max_calls_recycles_worker(ray_mod):
    """@ray.remote(max_calls=N): the worker process exits after N calls
    (reference worker recycling for leaky native libs) and later tasks
    run in a fresh process."""
    ray = ray_mod

    @ray.remote(num_cpus=0.1, max_calls=2)
    def pid():
        import os

        return os.getpid()

    pids = [ray.get(pid.remote(), timeout=60) for _ in range(6)]
    # 6 calls at max_calls=2 -> at least 3 distinct processes
    assert len(set(pids)) >= 3, pids


def test_retry_exceptions(ray_mod):
    """retry_exceptions: application errors matching the policy consume
    max_retries and resubmit; non-matching errors surface immediately."""
    import os

    ray = ray_mod
    marker = "/tmp/antray_retry_exc"
    for f in (marker, marker + "2"):
        try:
            os.unlink(f)
        except FileNotFoundError:
            pass

    @ray.remote(max_retries=3, retry_exceptions=[ValueError])
    def flaky():
        if not os.path.exists(marker):
            open(marker, "w").close()
            raise ValueError("transient")
        return "recovered"

    assert ray.get(flaky.remote(), timeout=60) == "recovered"

    @ray.remote(max_retries=3, retry_exceptions=[KeyError])
    def wrong_type():
        raise ValueError("not retryable for this policy")

    import pytest as _pt

    with _pt.raises(Exception) as ei:
        ray.get(wrong_type.remote(), timeout=60)
    assert "not retryable" in str(ei.value)

    @ray.remote(max_retries=2, retry_exceptions=True)
    def always_fails():
        with open(marker + "2", "a") as f:
            f.write("x")
        raise RuntimeError("permanent")

    with _pt.raises(Exception):
        ray.get(always_fails.remote(), timeout=60)
    with open(marker + "2") as f:
        assert len(f.read()) == 3  # initial try + 2 retries


def test_actor_method_retry_exceptions(ray_mod):
    """@ray.method(retry_exceptions=[...], max_task_retries=N): a failing
    actor METHOD retries on the live actor without restarting it."""
    ray = ray_mod

    @ray.remote
    class Flaky:
        def __init__(self):
            self.calls = 0

        @ray.method(retry_exceptions=[ValueError], max_task_retries=3)
        def work(self):
            self.calls += 1
            if self.calls < 3:
                raise ValueError("transient")
            return f"ok after {self.calls}"

    f = Flaky.remote()
    assert ray.get(f.work.remote(), timeout=60) == "ok after 3"


def test_max_pending_calls(ray_mod):
    """max_pending_calls: client-side backpressure raises
    PendingCallsLimitExceeded instead of queueing without bound."""
    ray = ray_mod
    from ant_ray_amd.exceptions import PendingCallsLimitExceeded

    @ray.remote
    class Slow:
        def work(self):
            time.sleep(1.0)
            return 1

    a = Slow.options(max_pending_calls=3).remote()
    refs = [a.work.remote() for _ in range(3)]
    with pytest.raises(PendingCallsLimitExceeded):
        for _ in range(20):  # the 4th+ submission must trip the limit
            refs.append(a.work.remote())
    assert ray.get(refs[:3], timeout=120) == [1, 1, 1]
