"""Fault-injection test utilities.

Role parity: reference python/ray/_private/test_utils.py
(ResourceKillerActor :1372, NodeKillerBase :1458, kill_raylet :1908) and
the chaos suite. Used with cluster_utils.Cluster to exercise
failure-detection/recovery paths.
"""
from __future__ import annotations

import random
import threading
import time
from typing import List, Optional


class NodeKiller:
    """Kills random worker nodes of a cluster_utils.Cluster on an interval
    (parity NodeKillerBase: the reference runs it as an actor against cloud
    nodes; locally we SIGKILL raylet subprocesses)."""

    def __init__(self, cluster, interval_s: float = 5.0,
                 max_kills: int = 3, seed: Optional[int] = None):
        self.cluster = cluster
        self.interval_s = interval_s
        self.max_kills = max_kills
        self.rng = random.Random(seed)
        self.killed: List[str] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def _loop(self):
        while not self._stop.is_set() and len(self.killed) < self.max_kills:
            self._stop.wait(self.interval_s)
            if self._stop.is_set():
                return
            nodes = list(self.cluster.worker_nodes)
            if not nodes:
                continue
            victim = self.rng.choice(nodes)
            self.killed.append(f"pid:{victim.proc.pid}")
            self.cluster.remove_node(victim)

    def run(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)


def kill_raylet(node):
    """SIGKILL one ClusterNode's raylet (parity kill_raylet :1908)."""
    node.kill()


def wait_for_condition(fn, timeout: float = 30, retry_interval_ms: int = 100,
                       **kwargs) -> bool:
    """Parity: test_utils.wait_for_condition."""
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            if fn(**kwargs):
                return True
        except Exception:
            pass
        time.sleep(retry_interval_ms / 1000)
    raise RuntimeError(f"condition {fn} not met within {timeout}s")
