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
