"""Fake multi-node cluster in one host — the distributed-test backbone.

Role parity: reference python/ray/cluster_utils.py:135 (class Cluster,
add_node :202, remove_node :286): extra raylets with their own ports and
object stores join the head's GCS, so multi-node scheduling/failure paths
run on one machine.
"""
from __future__ import annotations

import json
import os
import subprocess
import sys
import time
from typing import Dict, List, Optional


class ClusterNode:
    def __init__(self, proc: subprocess.Popen, port: int, store_path: str):
        self.proc = proc
        self.port = port
        self.store_path = store_path

    @property
    def node_id(self) -> Optional[str]:
        return getattr(self, "_node_id", None)

    def kill(self):
        try:
            self.proc.kill()
            self.proc.wait(timeout=10)
        except Exception:
            pass


class Cluster:
    def __init__(self, initialize_head: bool = True,
                 connect: bool = False,
                 head_node_args: Optional[dict] = None):
        self.head = None
        self.worker_nodes: List[ClusterNode] = []
        self.gcs_address: Optional[str] = None
        if initialize_head:
            from ant_ray_amd._private.node import start_head

            args = dict(head_node_args or {})
            self.head = start_head(
                num_cpus=args.get("num_cpus"),
                num_gpus=args.get("num_gpus", 0),
                resources=args.get("resources"),
            )
            self.gcs_address = self.head.info["gcs_addr"]
            if connect:
                import ant_ray_amd as ray

                ray.init(address=self.gcs_address)

    @property
    def address(self):
        return self.gcs_address

    def add_node(self, num_cpus: int = 1, num_gpus: int = 0,
                 resources: Optional[Dict[str, float]] = None,
                 object_store_memory: int = 512 * 1024 * 1024,
                 wait: bool = True, labels: Optional[Dict[str, str]] = None,
                 **_) -> ClusterNode:
        assert self.gcs_address, "head must be started first"
        session_dir = self.head.info["session_dir"]
        store_path = os.path.join(
            "/dev/shm", f"antray_node_{os.getpid()}_{len(self.worker_nodes)}"
            f"_{int(time.time()*1000)}")
        r_fd, w_fd = os.pipe()
        cmd = [
            sys.executable, "-m", "ant_ray_amd._private.raylet",
            "--gcs", self.gcs_address,
            "--num-cpus", str(num_cpus),
            "--num-gpus", str(num_gpus),
            "--store-path", store_path,
            "--store-capacity", str(object_store_memory),
            "--session-dir", session_dir,
            "--announce-fd", str(w_fd),
        ]
        if resources:
            cmd += ["--resources", json.dumps(resources)]
        if labels:
            cmd += ["--labels", json.dumps(labels)]
        proc = subprocess.Popen(cmd, pass_fds=(w_fd,),
                                stderr=open(os.path.join(
                                    session_dir, "logs",
                                    f"raylet_{len(self.worker_nodes)}.err"),
                                    "wb"))
        os.close(w_fd)
        with os.fdopen(r_fd) as f:
            parts = f.readline().split()
            port = int(parts[0])
        node = ClusterNode(proc, port, store_path)
        if len(parts) > 1:
            node._node_id = parts[1]
        self.worker_nodes.append(node)
        if wait:
            self.wait_for_nodes()
        return node

    def remove_node(self, node: ClusterNode, allow_graceful: bool = False):
        """allow_graceful=True drains the raylet first (running leases
        finish, queued ones spill elsewhere) — the autoscaler's idle
        termination path; default stays the hard kill used by failure
        tests."""
        if allow_graceful:
            try:
                import asyncio

                from ant_ray_amd._private import protocol

                async def _drain():
                    conn = await protocol.connect(("127.0.0.1", node.port),
                                                  None, name="drain")
                    await conn.call("drain", {"timeout_s": 30.0}, timeout=5)

                asyncio.run(_drain())
                node.proc.wait(timeout=45)
            except Exception:
                node.kill()
        else:
            node.kill()
        if node in self.worker_nodes:
            self.worker_nodes.remove(node)

    def wait_for_nodes(self, timeout: float = 30):
        """Block until every live raylet is registered and alive in the GCS."""
        import ant_ray_amd as ray

        expect = 1 + len(self.worker_nodes)
        deadline = time.monotonic() + timeout
        if not ray.is_initialized():
            ray.init(address=self.gcs_address)

        while time.monotonic() < deadline:
            alive = [n for n in ray.nodes() if n.get("alive", n.get("Alive"))]
            if len(alive) >= expect:
                return
            time.sleep(0.2)
        raise TimeoutError(f"cluster did not reach {expect} nodes")

    def connect(self):
        import ant_ray_amd as ray

        if ray.is_initialized():
            ray.shutdown()  # a stale session from another test/cluster
        return ray.init(address=self.gcs_address)

    def shutdown(self):
        import ant_ray_amd as ray

        try:
            ray.shutdown()
        except Exception:
            pass
        for n in list(self.worker_nodes):
            self.remove_node(n)
        if self.head is not None:
            self.head.terminate()
            self.head = None
