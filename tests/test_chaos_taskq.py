"""Chaos injection, NodeKiller recovery, serve task queue."""
import os
import subprocess
import sys
import textwrap
import time

import pytest


def test_chaos_delay_injection():
    """RAY_testing_asio_delay_us slows targeted RPC handlers (parity
    asio_chaos.cc)."""
    script = textwrap.dedent("""
        import time
        import ant_ray_amd as ray
        ray.init(num_cpus=2)

        @ray.remote
        def f():
            return 1

        ray.get(f.remote())
        t0 = time.time()
        for _ in range(20):
            ray.cluster_resources()  # one GCS RPC per call
        print("ELAPSED", time.time() - t0)
        ray.shutdown()
    """)
    env = dict(os.environ)
    out1 = subprocess.run([sys.executable, "-c", script], env=env,
                          capture_output=True, text=True, timeout=180)
    base = float(out1.stdout.split("ELAPSED")[1].split()[0])
    env["RAY_testing_asio_delay_us"] = "cluster_resources=30000:30001"
    out2 = subprocess.run([sys.executable, "-c", script], env=env,
                          capture_output=True, text=True, timeout=300)
    delayed = float(out2.stdout.split("ELAPSED")[1].split()[0])
    assert delayed > base + 0.3, (base, delayed)  # 20 x 30ms injected


def test_node_killer_recovery():
    """Actors killed with their node get restarted on surviving nodes."""
    import ant_ray_amd as ray
    from ant_ray_amd._private.test_utils import NodeKiller, wait_for_condition
    from ant_ray_amd.cluster_utils import Cluster

    if ray.is_initialized():
        ray.shutdown()
    c = Cluster(initialize_head=True, head_node_args={"num_cpus": 2})
    try:
        c.connect()
        c.add_node(num_cpus=2)
        c.add_node(num_cpus=2)

        @ray.remote(num_cpus=1, max_restarts=5, max_task_retries=5)
        class Counter:
            def __init__(self):
                self.n = 0

            def bump(self):
                self.n += 1
                return self.n

        counters = [Counter.remote() for _ in range(4)]
        assert ray.get([a.bump.remote() for a in counters], timeout=60)

        killer = NodeKiller(c, interval_s=0.5, max_kills=1, seed=1).run()
        time.sleep(1.5)
        killer.stop()
        assert killer.killed, "should have killed one node"
        # every actor still answers (restarted elsewhere if its node died)
        out = ray.get([a.bump.remote() for a in counters], timeout=120)
        assert len(out) == 4
    finally:
        c.shutdown()


def test_serve_task_queue():
    import ant_ray_amd as ray
    from ant_ray_amd.serve.task_processor import QueueTaskProcessorAdapter

    if ray.is_initialized():
        ray.shutdown()
    ray.init(num_cpus=4)
    adapter = QueueTaskProcessorAdapter("t1")
    adapter.register("double", lambda x: x * 2)
    adapter.start_consumer()
    tid = adapter.enqueue("double", 21)
    st = adapter.wait(tid, timeout=30)
    assert st["status"] == "SUCCEEDED" and st["result"] == 42
    bad = adapter.enqueue("missing", 1)
    assert adapter.wait(bad, timeout=30)["status"] == "FAILED"
    adapter.stop_consumer()
    ray.shutdown()
