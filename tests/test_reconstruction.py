"""Lineage reconstruction: a task output lost with its node is recomputed
by resubmitting the producing task (parity: reference
object_recovery_manager.h + test_reconstruction*.py)."""
import time

import pytest


@pytest.fixture()
def cluster():
    from ant_ray_amd.cluster_utils import Cluster

    c = Cluster(initialize_head=True, head_node_args={"num_cpus": 2})
    yield c
    c.shutdown()


def test_lost_object_reconstructed(cluster):
    import numpy as np

    import ant_ray_amd as ray

    cluster.connect()
    n1 = cluster.add_node(num_cpus=2, resources={"rack": 1})
    deadline = time.time() + 30
    while time.time() < deadline and not ray.cluster_resources().get("rack"):
        time.sleep(0.2)

    @ray.remote(num_cpus=1, resources={"rack": 0.1}, max_retries=3)
    def produce():
        import os

        return np.full(1024 * 1024, 7.0), os.getpid()  # ~8 MB, stays remote

    ref = produce.remote()
    # wait for completion WITHOUT pulling the bytes locally
    deadline = time.time() + 60
    while time.time() < deadline:
        from ant_ray_amd._private.worker import global_worker

        cw = global_worker.core_worker
        if cw._object_locations.get(ref.binary()):
            break
        time.sleep(0.2)
    assert cw._object_locations.get(ref.binary()), "task should have finished"

    # second rack node BEFORE killing the first (reconstruction target)
    cluster.add_node(num_cpus=2, resources={"rack": 1})
    time.sleep(1.0)
    cluster.remove_node(n1)  # holder dies WITH the data
    time.sleep(1.0)

    arr, pid = ray.get(ref, timeout=120)
    assert float(arr.sum()) == 1024 * 1024 * 7.0
