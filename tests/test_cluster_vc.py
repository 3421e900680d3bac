"""Multi-node tests: cluster_utils.Cluster (fake multi-raylet cluster),
node death handling, and virtual clusters (ant-fork feature parity)."""
import time

import pytest


@pytest.fixture()
def cluster():
    from ant_ray_amd.cluster_utils import Cluster

    c = Cluster(initialize_head=True, head_node_args={"num_cpus": 2})
    yield c
    c.shutdown()


def test_add_remove_node_and_spread(cluster):
    import ant_ray_amd as ray

    cluster.connect()
    n1 = cluster.add_node(num_cpus=2, resources={"tag_a": 1})
    cluster.add_node(num_cpus=2, resources={"tag_b": 1})
    assert len(ray.nodes()) == 3
    total = ray.cluster_resources()
    assert total["CPU"] == 6.0
    assert total.get("tag_a") == 1.0

    # actors land on the custom-resource nodes
    @ray.remote(num_cpus=1, resources={"tag_a": 0.1})
    class A:
        def where(self):
            import os

            return os.getpid()

    a = A.remote()
    assert ray.get(a.where.remote(), timeout=60) > 0

    # node removal -> resources drop, node marked dead
    cluster.remove_node(n1)
    deadline = time.time() + 30
    while time.time() < deadline:
        alive = [n for n in ray.nodes() if n["Alive"]]
        if len(alive) == 2:
            break
        time.sleep(0.3)
    assert len([n for n in ray.nodes() if n["Alive"]]) == 2


def test_virtual_cluster_scheduling(cluster):
    import ant_ray_amd as ray
    from ant_ray_amd.util import virtual_cluster as vc

    cluster.connect()
    cluster.add_node(num_cpus=2)
    cluster.add_node(num_cpus=2)

    nodes = ray.nodes()
    head_id = min(n["NodeID"] for n in nodes)  # arbitrary but stable pick
    view = vc.create_or_update_virtual_cluster("vc1", node_ids=[head_id])
    assert view["node_ids"] == [head_id]
    assert any(v["virtual_cluster_id"] == "vc1"
               for v in vc.list_virtual_clusters())

    # a second vc cannot steal vc1's node
    with pytest.raises(RuntimeError):
        vc.create_or_update_virtual_cluster("vc2", node_ids=[head_id])

    # count-based vc takes nodes from the unassigned pool
    view2 = vc.create_or_update_virtual_cluster("vc2", node_count=1)
    assert len(view2["node_ids"]) == 1 and view2["node_ids"] != [head_id]

    # actors of a vc1-pinned driver land only on vc1's node
    import subprocess
    import sys
    import textwrap

    script = textwrap.dedent(f"""
        import os
        import ant_ray_amd as ray
        ray.init(address="{cluster.address}", _virtual_cluster_id="vc1")

        @ray.remote(num_cpus=1)
        class W:
            def node(self):
                return ray.get_runtime_context().get_node_id()

        ws = [W.remote() for _ in range(2)]
        nodes = set(ray.get([w.node.remote() for w in ws], timeout=60))
        assert nodes == {{"{head_id}"}}, nodes
        print("VC_OK")
    """)
    out = subprocess.run([sys.executable, "-c", script], capture_output=True,
                         text=True, timeout=120)
    assert "VC_OK" in out.stdout, out.stdout + out.stderr

    assert vc.remove_virtual_cluster("vc1")
    assert vc.get_virtual_cluster("vc1") is None


def test_unknown_virtual_cluster_rejected(cluster):
    import subprocess
    import sys
    import textwrap

    script = textwrap.dedent(f"""
        import ant_ray_amd as ray
        try:
            ray.init(address="{cluster.address}", _virtual_cluster_id="nope")
            print("CONNECTED")
        except ConnectionError as e:
            print("REJECTED", e)
    """)
    out = subprocess.run([sys.executable, "-c", script], capture_output=True,
                         text=True, timeout=120)
    assert "REJECTED" in out.stdout, out.stdout + out.stderr


def test_cross_node_chunked_pull(cluster):
    """A >5MiB object created on one node is pulled by a worker on another
    node through the chunked data plane (PULL_CHUNK_BYTES slices pipelined
    on one connection)."""
    import numpy as np

    import ant_ray_amd as ray

    cluster.connect()
    cluster.add_node(num_cpus=2, resources={"far": 1})
    deadline = time.time() + 30
    while time.time() < deadline and not ray.cluster_resources().get("far"):
        time.sleep(0.2)

    arr = np.arange(3 * 1024 * 1024, dtype=np.float64)  # 24 MB -> 5 chunks
    ref = ray.put(arr)

    @ray.remote(num_cpus=1, resources={"far": 0.1})
    def consume(x):
        return float(x.sum()), x.shape[0]

    total, n = ray.get(consume.remote(ref), timeout=120)
    assert n == arr.shape[0]
    assert total == pytest.approx(float(arr.sum()))

    # the pulled copy must land in the consumer node's shm store (the
    # shm-destination write path), so a second task on that node reads
    # it locally without a cross-node pull
    @ray.remote(num_cpus=1, resources={"far": 0.1})
    def consume_nested(refs):
        x = ray.get(refs[0])
        from ant_ray_amd._private.worker import global_worker

        shm = global_worker.core_worker.store.shm
        return float(x.sum()), shm.contains(refs[0].binary())

    total2, in_shm = ray.get(consume_nested.remote([ref]), timeout=120)
    assert total2 == pytest.approx(float(arr.sum()))
    assert in_shm, "pulled big object should be cached in local shm"


def test_wait_many_refs_drain(cluster):
    """ray.wait drains 300 refs one at a time (the ray_perf wait-1k shape)."""
    import ant_ray_amd as ray

    cluster.connect()

    @ray.remote
    def unit():
        return 1

    not_ready = [unit.remote() for _ in range(300)]
    got = 0
    deadline = time.time() + 90
    while not_ready and time.time() < deadline:
        ready, not_ready = ray.wait(not_ready)
        got += len(ready)
    assert got == 300 and not not_ready


def test_node_label_scheduling(cluster):
    """NodeLabelSchedulingStrategy: hard label constraints route actors and
    tasks to matching nodes (including lease spillback for tasks)."""
    import ant_ray_amd as ray
    from ant_ray_amd.util.scheduling_strategies import (
        NodeLabelSchedulingStrategy,
    )

    cluster.connect()
    cluster.add_node(num_cpus=2, labels={"accel": "mi355x", "zone": "a"})
    deadline = time.time() + 30
    while time.time() < deadline and len(
            [n for n in ray.nodes() if n["Alive"]]) < 2:
        time.sleep(0.2)

    @ray.remote(num_cpus=1,
                scheduling_strategy=NodeLabelSchedulingStrategy(
                    hard={"accel": "mi355x"}))
    class Pinned:
        def node(self):
            return ray.get_runtime_context().get_node_id()

    a = Pinned.remote()
    nid = ray.get(a.node.remote(), timeout=60)
    labeled = [n for n in ray.nodes()
               if n["Resources"].get("CPU") and not n.get("is_head")]
    assert nid is not None

    @ray.remote(num_cpus=1,
                scheduling_strategy=NodeLabelSchedulingStrategy(
                    hard={"accel": ["mi355x", "mi300x"]}))
    def where():
        return ray.get_runtime_context().get_node_id()

    tid = None
    for _ in range(3):
        try:
            tid = ray.get(where.remote(), timeout=60)
            break
        except Exception:
            time.sleep(1.0)  # worker churn from the previous cluster teardown
    assert tid == nid, "task must spill back to the labeled node"


def test_locality_aware_leasing(cluster):
    """A task whose (big) argument lives on another node leases THERE
    (reference locality-aware LeasePolicy) instead of pulling the bytes
    to a random node."""
    import numpy as np

    import ant_ray_amd as ray

    cluster.connect()
    cluster.add_node(num_cpus=2, resources={"datanode": 1})
    deadline = time.time() + 30
    while time.time() < deadline and not ray.cluster_resources().get("datanode"):
        time.sleep(0.2)

    @ray.remote(num_cpus=1, resources={"datanode": 0.1})
    def produce():
        import ant_ray_amd as ray2

        data = np.ones(2 * 1024 * 1024, dtype=np.float64)  # 16 MB
        # inner put: the bytes stay OWNED by this datanode worker; only
        # the tiny ref wrapper travels to the driver
        return [ray2.put(data)]

    inner = ray.get(produce.remote(), timeout=60)[0]

    @ray.remote(num_cpus=1)
    def consume_where(x):
        from ant_ray_amd._private.worker import global_worker

        cw = global_worker.core_worker
        return float(x.sum()), cw.node_id.hex()

    # no resource pin: the locality hint must route the lease to the
    # node owning the 16 MB argument
    total, consumer_node = ray.get(consume_where.remote(inner), timeout=120)
    assert total == 2 * 1024 * 1024
    rows = {n["NodeID"]: n["Resources"] for n in ray.nodes()}
    assert "datanode" in rows.get(consumer_node, {}), (
        "consumer should be scheduled on the node holding the argument")


def test_wait_remote_held_object(cluster):
    """ray.wait must report a remotely-held (un-fetched) object as ready:
    readiness means 'exists', not 'local' (reference semantics)."""
    import numpy as np

    import ant_ray_amd as ray

    cluster.connect()
    cluster.add_node(num_cpus=2, resources={"remote": 1})
    deadline = time.time() + 30
    while time.time() < deadline and not ray.cluster_resources().get("remote"):
        time.sleep(0.2)

    @ray.remote(num_cpus=1, resources={"remote": 0.1})
    def make_big():
        return np.zeros(1024 * 1024)  # 8 MB: stays on the remote node

    ref = make_big.remote()
    ready, not_ready = ray.wait([ref], timeout=60)
    assert ready == [ref] and not_ready == []
    assert float(ray.get(ref, timeout=60).sum()) == 0.0


def test_spread_scheduling_strategy(cluster):
    """scheduling_strategy="SPREAD": actors fan out across feasible nodes
    instead of packing the first one."""
    import ant_ray_amd as ray

    cluster.connect()
    cluster.add_node(num_cpus=4)
    cluster.add_node(num_cpus=4)
    deadline = time.time() + 30
    while time.time() < deadline and len(
            [n for n in ray.nodes() if n["Alive"]]) < 3:
        time.sleep(0.2)

    @ray.remote(num_cpus=0.1, scheduling_strategy="SPREAD")
    class S:
        def node(self):
            return ray.get_runtime_context().get_node_id()

    actors = [S.remote() for _ in range(9)]
    nodes = set(ray.get([a.node.remote() for a in actors], timeout=120))
    assert len(nodes) >= 2, f"SPREAD placed everything on {nodes}"


def test_graceful_drain(cluster):
    """remove_node(allow_graceful=True): the raylet finishes its running
    lease before exiting (no task failure, no retry)."""
    import os

    import ant_ray_amd as ray

    cluster.connect()
    n = cluster.add_node(num_cpus=2, resources={"drainme": 1})
    deadline = time.time() + 30
    while time.time() < deadline and not ray.cluster_resources().get("drainme"):
        time.sleep(0.2)

    attempts = "/tmp/antray_drain_attempts"
    open(attempts, "w").close()

    @ray.remote(num_cpus=1, resources={"drainme": 0.1}, max_retries=0)
    def slowish():
        import os as _os
        import time as _t

        with open(attempts, "a") as f:
            f.write(f"{_os.getpid()}\n")
        _t.sleep(2.0)
        return "finished"

    ref = slowish.remote()
    time.sleep(0.8)  # task is running
    cluster.remove_node(n, allow_graceful=True)  # drain waits for it
    assert ray.get(ref, timeout=60) == "finished"
    with open(attempts) as f:
        assert len(f.read().splitlines()) == 1  # ran exactly once


def test_wait_fetch_local_prefetches(cluster):
    """ray.wait(fetch_local=True) pulls ready remote objects into local
    shm in the background (reference wait semantics)."""
    import numpy as np

    import ant_ray_amd as ray
    from ant_ray_amd._private.worker import global_worker

    cluster.connect()
    cluster.add_node(num_cpus=2, resources={"src": 1})
    deadline = time.time() + 30
    while time.time() < deadline and not ray.cluster_resources().get("src"):
        time.sleep(0.2)

    @ray.remote(num_cpus=1, resources={"src": 0.1})
    def make():
        return np.ones(1024 * 1024)  # 8 MB task output, stays remote

    inner = make.remote()
    ready, _ = ray.wait([inner], timeout=60, fetch_local=True)
    assert ready == [inner]
    cw = global_worker.core_worker
    deadline = time.time() + 60
    while time.time() < deadline:
        if cw.store.shm.contains(inner.binary()):
            break
        time.sleep(0.2)
    assert cw.store.shm.contains(inner.binary()), "prefetch never landed"
    assert float(ray.get(inner, timeout=30).sum()) == 1024 * 1024


def test_wait_on_borrowed_ref(cluster):
    """ray.wait works on BORROWED refs (owned by a remote worker): the
    waiter probes the owner for existence."""
    import numpy as np

    import ant_ray_amd as ray

    cluster.connect()
    cluster.add_node(num_cpus=2, resources={"own": 1})
    deadline = time.time() + 30
    while time.time() < deadline and not ray.cluster_resources().get("own"):
        time.sleep(0.2)

    @ray.remote(num_cpus=1, resources={"own": 0.1})
    def owner_task():
        import time as _t

        import ant_ray_amd as ray2

        slow = ray2.put(np.ones(700_000))  # 5.6 MB owned by this worker
        return [slow]

    inner = ray.get(owner_task.remote(), timeout=60)[0]
    ready, not_ready = ray.wait([inner], timeout=60)
    assert ready == [inner], (ready, not_ready)
    assert float(ray.get(inner, timeout=60).sum()) == 700_000


def test_serve_replicas_on_worker_node(cluster):
    """Serve replicas placed on a second node serve traffic through the
    head-node proxy (cross-node actor data plane)."""
    import urllib.request

    import ant_ray_amd as ray
    from ant_ray_amd import serve

    cluster.connect()
    cluster.add_node(num_cpus=4, resources={"edge": 1})
    deadline = time.time() + 30
    while time.time() < deadline and not ray.cluster_resources().get("edge"):
        time.sleep(0.2)

    @serve.deployment(num_replicas=2,
                      ray_actor_options={"num_cpus": 1,
                                         "resources": {"edge": 0.1}})
    class EdgeApp:
        def __call__(self, request):
            import ant_ray_amd as ray2

            return {"node": ray2.get_runtime_context().get_node_id()}

    serve.run(EdgeApp.bind(), name="edgeapp", route_prefix="/edge")
    try:
        port = ray.get(ray.get_actor("SERVE_PROXY_ACTOR").ready.remote())
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/edge", timeout=30).read().decode()
        import json as _json

        node = _json.loads(body)["node"]
        edge_nodes = {n["NodeID"] for n in ray.nodes()
                      if n["Resources"].get("edge")}
        assert node in edge_nodes, (node, edge_nodes)
    finally:
        serve.shutdown()


def test_train_across_nodes(cluster):
    """TorchTrainer with one worker pinned to each of two nodes: the gloo
    process group forms across raylets and gradients sync."""
    import ant_ray_amd as ray
    from ant_ray_amd.train import RunConfig, ScalingConfig
    from ant_ray_amd.train.torch import TorchTrainer

    cluster.connect()
    cluster.add_node(num_cpus=2, resources={"trainnode": 1})
    cluster.add_node(num_cpus=2, resources={"trainnode": 1})
    deadline = time.time() + 30
    while time.time() < deadline and ray.cluster_resources().get(
            "trainnode", 0) < 2:
        time.sleep(0.2)

    def train_fn(config):
        import torch
        import torch.distributed as dist

        from ant_ray_amd import train

        assert dist.get_world_size() == 2
        t = torch.ones(4) * (dist.get_rank() + 1)
        dist.all_reduce(t)
        train.report({
            "sum": float(t[0]),
            "node": train.get_context().get_node_id()
            if hasattr(train.get_context(), "get_node_id") else "",
        })

    import tempfile

    res = TorchTrainer(
        train_fn,
        scaling_config=ScalingConfig(
            num_workers=2,
            # CPU:2 per worker on 2-CPU nodes -> exactly one worker per
            # node: the group really spans raylets
            resources_per_worker={"CPU": 2, "trainnode": 0.1}),
        run_config=RunConfig(name="xnode",
                             storage_path=tempfile.mkdtemp()),
    ).fit()
    assert res.error is None, res.error
    assert res.metrics["sum"] == 3.0  # 1+2 all-reduced across nodes
