"""Ray-Client mode: a driver without local shm access (remote host
semantics) proxies big objects through the raylet data plane."""
import subprocess
import sys
import textwrap

import pytest


def test_client_mode_driver():
    import ant_ray_amd as ray
    from ant_ray_amd.cluster_utils import Cluster

    if ray.is_initialized():
        ray.shutdown()
    c = Cluster(initialize_head=True, head_node_args={"num_cpus": 4})
    try:
        script = textwrap.dedent(f"""
            import os
            os.environ["ANTRAY_FORCE_CLIENT"] = "1"
            import numpy as np
            import ant_ray_amd as ray
            ray.init(address="ray://{c.address}")
            from ant_ray_amd._private.worker import global_worker
            assert global_worker.core_worker.client_mode

            # big put/get round trip through the raylet store
            arr = np.arange(500_000, dtype=np.int64)
            ref = ray.put(arr)
            back = ray.get(ref, timeout=60)
            assert (back == arr).all()

            # tasks with big results
            @ray.remote
            def make(n):
                return np.ones(n, dtype=np.float32)

            out = ray.get(make.remote(400_000), timeout=60)
            assert out.sum() == 400_000

            # actors
            @ray.remote
            class A:
                def echo(self, x):
                    return x * 2

            a = A.remote()
            assert ray.get(a.echo.remote(21), timeout=60) == 42
            print("CLIENT_OK")
        """)
        out = subprocess.run([sys.executable, "-c", script],
                             capture_output=True, text=True, timeout=180)
        assert "CLIENT_OK" in out.stdout, out.stdout + out.stderr
    finally:
        c.shutdown()
