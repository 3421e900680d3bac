"""GCS — the cluster-global control plane.

Role parity: reference gcs_server (src/ray/gcs/gcs_server.cc:270 DoStart wires
NodeManager/ResourceManager/HealthCheck/JobManager/PlacementGroupManager/
ActorManager/WorkerManager + KV + pubsub). Same responsibilities, one asyncio
process, msgpack-RPC transport (see protocol.py):

  * node table + health (heartbeats; dead after miss window — parity with
    gcs_health_check_manager.h:45, 3s period / 5 misses),
  * actor table with scheduling (lease a worker from a raylet, push the
    creation task, restart per max_restarts — parity with
    gcs_actor_manager.h:94 + gcs_actor_scheduler.h:108),
  * named actors, job ids, cluster KV (function table lives here —
    gcs_kv_manager.cc / gcs_function_manager.h),
  * placement groups (reserve bundles on raylets, 2PC-lite — parity with
    gcs_placement_group_manager.h:55),
  * pubsub channels pushed over registered connections (src/ray/pubsub/).
"""
from __future__ import annotations

import argparse
import asyncio
import logging
import os
import time
from typing import Any, Dict, List, Optional, Tuple

from ant_ray_amd._private import protocol
from ant_ray_amd._private.protocol import Connection

logger = logging.getLogger("antray.gcs")

# Actor states (parity with reference rpc::ActorTableData states)
DEPENDENCIES_UNREADY = "DEPENDENCIES_UNREADY"
PENDING_CREATION = "PENDING_CREATION"
ALIVE = "ALIVE"
RESTARTING = "RESTARTING"
DEAD = "DEAD"

HEARTBEAT_PERIOD_S = 1.0
HEARTBEAT_MISS_LIMIT = 5


class NodeInfo:
    def __init__(self, node_id, addr, resources, store_path, object_store_bytes=0):
        self.node_id: bytes = node_id
        self.addr: Tuple[str, int] = tuple(addr)
        self.resources_total: Dict[str, float] = dict(resources)
        self.resources_available: Dict[str, float] = dict(resources)
        self.store_path: str = store_path
        self.object_store_bytes = object_store_bytes
        self.labels: dict = {}
        self.alive = True
        self.last_heartbeat = time.monotonic()
        self.conn: Optional[Connection] = None

    def view(self):
        return {
            "node_id": self.node_id,
            "addr": list(self.addr),
            "resources_total": self.resources_total,
            "resources_available": self.resources_available,
            "store_path": self.store_path,
            "alive": self.alive,
            "labels": self.labels,
        }


class ActorInfo:
    def __init__(self, actor_id, owner, name, namespace, opts, create_payload):
        self.actor_id: bytes = actor_id
        self.owner: Optional[bytes] = owner
        self.name: str = name or ""
        self.namespace: str = namespace or ""
        self.opts: Dict[str, Any] = opts or {}
        self.create_payload = create_payload  # msgpack-able dict to push_task
        self.state = PENDING_CREATION
        self.addr: Optional[Tuple[str, int]] = None
        self.worker_id: Optional[bytes] = None
        self.node_id: Optional[bytes] = None
        self.num_restarts = 0
        self.death_cause = ""
        self.killed = False  # ray.kill / out-of-scope while still scheduling
        self.pending_waiters: List[asyncio.Future] = []

    def view(self):
        return {
            "actor_id": self.actor_id,
            "state": self.state,
            "addr": list(self.addr) if self.addr else None,
            "worker_id": self.worker_id,
            "node_id": self.node_id,
            "name": self.name,
            "namespace": self.namespace,
            "num_restarts": self.num_restarts,
            "death_cause": self.death_cause,
            "max_restarts": int(self.opts.get("max_restarts", 0)),
            "max_task_retries": int(self.opts.get("max_task_retries", 0)),
            "required_resources": {
                "CPU": float(self.opts.get("num_cpus", 1)),
                **({"GPU": float(self.opts["num_gpus"])}
                   if self.opts.get("num_gpus") else {}),
                **(self.opts.get("resources") or {}),
            },
        }


class PlacementGroupInfo:
    def __init__(self, pg_id, bundles, strategy, name=""):
        self.pg_id: bytes = pg_id
        self.bundles: List[Dict[str, float]] = bundles
        self.strategy = strategy
        self.name = name
        self.state = "PENDING"
        self.bundle_nodes: List[Optional[bytes]] = [None] * len(bundles)

    def view(self):
        return {
            "pg_id": self.pg_id,
            "state": self.state,
            "strategy": self.strategy,
            "bundles": self.bundles,
            "bundle_nodes": self.bundle_nodes,
            "name": self.name,
        }


class GcsServer:
    def __init__(self):
        self.nodes: Dict[bytes, NodeInfo] = {}
        self.actors: Dict[bytes, ActorInfo] = {}
        self.named_actors: Dict[Tuple[str, str], bytes] = {}
        self.pgs: Dict[bytes, PlacementGroupInfo] = {}
        self.kv: Dict[str, Dict[bytes, bytes]] = {}
        self.workers: Dict[bytes, Dict[str, Any]] = {}  # worker_id -> info
        self.jobs: Dict[int, Dict[str, Any]] = {}
        self._job_counter = 0
        self._subscribers: Dict[str, List[Connection]] = {}
        self._worker_conns: Dict[Tuple[str, int], Connection] = {}
        self._server = None
        self.port = None
        self._shutdown = asyncio.Event()
        # virtual clusters (ant fork feature — reference
        # src/ray/gcs/gcs_virtual_cluster.h:29-610): named partitions of the
        # node set; jobs pinned to a cluster schedule only on its nodes, and
        # node-count-based clusters replenish from the unassigned pool on
        # node death (gcs_virtual_cluster_manager.cc:730)
        self.virtual_clusters: Dict[str, Dict[str, Any]] = {}
        self.worker_vc: Dict[bytes, str] = {}  # worker_id -> vc id
        # fault tolerance: durable tables are snapshotted to this file and
        # restored by a restarted GCS (role parity with the reference's
        # Redis-backed gcs_table_storage / redis_store_client.cc — this
        # air-gapped image has no Redis, so the store client is a file)
        self.persist_path: Optional[str] = None
        self._persist_task: Optional[asyncio.Task] = None
        self._persist_dirty = False

    # ------------------------------------------------------------------ serve
    async def start(self, host="127.0.0.1", port=0, persist_path=None):
        self.persist_path = persist_path or os.environ.get("ANTRAY_GCS_PERSIST")
        if self.persist_path and os.path.exists(self.persist_path):
            try:
                self._restore_tables()
                logger.warning("GCS restored %d actors, %d kv namespaces, "
                               "%d jobs from %s", len(self.actors),
                               len(self.kv), len(self.jobs), self.persist_path)
            except Exception:
                logger.exception("GCS table restore failed; starting fresh")
        self._server, self.port = await protocol.serve(self._handle, host, port)
        asyncio.get_running_loop().create_task(self._health_loop())
        # export-event pipeline (file sink / HTTP aggregator), env-gated
        from ant_ray_amd._private import event_export

        self._event_exporter = event_export.maybe_start(self)
        if self.actors:
            asyncio.get_running_loop().create_task(
                self._verify_restored_actors())
        logger.info("GCS listening on %s:%s", host, self.port)
        return self.port

    async def _verify_restored_actors(self):
        """A worker that died DURING a GCS outage was reported to the
        dead GCS: restored ALIVE actors must be liveness-checked or they
        stay reachable-on-paper forever. Ping each restored actor's
        worker; unreachable -> normal worker-death handling (restart per
        max_restarts or DEAD)."""
        await asyncio.sleep(3.0)  # give raylets/workers time to reconnect
        for info in list(self.actors.values()):
            if info.state != ALIVE or not info.addr:
                continue
            alive = False
            for _ in range(3):
                try:
                    conn = await self._get_worker_conn(tuple(info.addr))
                    r = await conn.call("ping", {}, timeout=5)
                    alive = bool(r.get("ok"))
                    break
                except Exception:
                    await asyncio.sleep(1.0)
            if not alive:
                logger.warning("restored actor %s unreachable at %s; "
                               "declaring its worker dead",
                               info.actor_id.hex()[:8], info.addr)
                await self._on_actor_worker_died(
                    info, "worker died while the GCS was down")

    # ------------------------------------------------------- fault tolerance
    def _persist_soon(self):
        """Debounced snapshot: coalesce bursts of mutations into one write.

        A mutation landing while a persist is already in flight sets the
        dirty flag; _persist_later loops until it drains, so every
        acknowledged mutation reaches disk even if it arrived during the
        pack/write phase of the previous snapshot."""
        if not self.persist_path:
            return
        self._persist_dirty = True
        if self._persist_task and not self._persist_task.done():
            return
        try:
            loop = asyncio.get_running_loop()
        except RuntimeError:
            return
        self._persist_task = loop.create_task(self._persist_later())

    async def _persist_later(self):
        await asyncio.sleep(0.1)
        while getattr(self, "_persist_dirty", False):
            self._persist_dirty = False
            try:
                # pack the snapshot ON the loop (msgpack of plain dicts —
                # no concurrent-mutation hazard, nested mutables included),
                # then only the file write goes OFF the loop so tens-of-MB
                # disk IO can't stall heartbeats
                import msgpack

                packed = msgpack.packb(self._snapshot(), use_bin_type=True)
                await asyncio.get_running_loop().run_in_executor(
                    None, self._write_snapshot_bytes, packed)
            except Exception:
                logger.exception("GCS table persist failed")

    def _persist_now(self):
        import msgpack

        self._persist_dirty = False
        self._write_snapshot_bytes(
            msgpack.packb(self._snapshot(), use_bin_type=True))

    def _snapshot(self):
        snap = {
            "kv": {ns: dict(d) for ns, d in self.kv.items()},
            "named_actors": [[list(k), v] for k, v in self.named_actors.items()],
            "jobs": self.jobs,
            "job_counter": self._job_counter,
            "virtual_clusters": {
                k: {**v, "nodes": list(v["nodes"])}
                for k, v in self.virtual_clusters.items()
            },
            "actors": [
                {
                    "actor_id": a.actor_id, "owner": a.owner, "name": a.name,
                    "namespace": a.namespace, "opts": a.opts,
                    "create_payload": a.create_payload, "state": a.state,
                    "addr": list(a.addr) if a.addr else None,
                    "worker_id": a.worker_id, "node_id": a.node_id,
                    "num_restarts": a.num_restarts,
                    "death_cause": a.death_cause,
                }
                for a in self.actors.values()
            ],
            "pgs": [
                {
                    "pg_id": g.pg_id, "bundles": g.bundles,
                    "strategy": g.strategy, "name": g.name, "state": g.state,
                    "bundle_nodes": g.bundle_nodes,
                }
                for g in self.pgs.values()
            ],
        }
        return snap

    def _write_snapshot_bytes(self, packed: bytes):
        tmp = self.persist_path + ".tmp"
        with open(tmp, "wb") as f:
            f.write(packed)
        os.replace(tmp, self.persist_path)

    def _restore_tables(self):
        import msgpack

        with open(self.persist_path, "rb") as f:
            snap = msgpack.unpackb(f.read(), raw=False, strict_map_key=False)
        self.kv = {ns: dict(d) for ns, d in snap.get("kv", {}).items()}
        self.named_actors = {tuple(k): v
                             for k, v in snap.get("named_actors", [])}
        self.jobs = {int(k): v for k, v in snap.get("jobs", {}).items()}
        self._job_counter = snap.get("job_counter", 0)
        self.virtual_clusters = {
            k: {**v, "nodes": set(v.get("nodes") or [])}
            for k, v in snap.get("virtual_clusters", {}).items()
        }
        for a in snap.get("actors", []):
            info = ActorInfo(a["actor_id"], a["owner"], a["name"],
                             a["namespace"], a["opts"], a["create_payload"])
            info.state = a["state"]
            info.addr = tuple(a["addr"]) if a["addr"] else None
            info.worker_id = a["worker_id"]
            info.node_id = a["node_id"]
            info.num_restarts = a["num_restarts"]
            info.death_cause = a["death_cause"]
            # in-flight creations did not survive the GCS death; their
            # owners resubmit (non-detached) or the actor is simply gone
            if info.state in (PENDING_CREATION, RESTARTING):
                info.state = DEAD
                info.death_cause = "GCS restarted during actor creation"
            self.actors[a["actor_id"]] = info
        for g in snap.get("pgs", []):
            pg = PlacementGroupInfo(g["pg_id"], g["bundles"], g["strategy"],
                                    g.get("name", ""))
            pg.state = g["state"]
            pg.bundle_nodes = g["bundle_nodes"]
            self.pgs[g["pg_id"]] = pg

    async def _health_loop(self):
        while not self._shutdown.is_set():
            await asyncio.sleep(HEARTBEAT_PERIOD_S)
            now = time.monotonic()
            for node in list(self.nodes.values()):
                if node.alive and now - node.last_heartbeat > HEARTBEAT_PERIOD_S * HEARTBEAT_MISS_LIMIT:
                    logger.warning("node %s missed heartbeats; marking dead", node.node_id.hex()[:8])
                    await self._on_node_dead(node)

    # --------------------------------------------------------------- dispatch
    async def _handle(self, conn: Connection, method: str, p: Any):
        fn = getattr(self, "rpc_" + method, None)
        if fn is None:
            raise ValueError(f"unknown GCS method {method}")
        return await fn(conn, p or {})

    # ------------------------------------------------------------------ nodes
    async def rpc_register_node(self, conn, p):
        node = NodeInfo(
            p["node_id"], tuple(p["addr"]), p.get("resources", {}),
            p.get("store_path", ""), p.get("object_store_bytes", 0),
        )
        node.labels = dict(p.get("labels") or {})
        node.conn = conn
        self.nodes[node.node_id] = node
        conn.session["node_id"] = node.node_id
        conn.on_close = self._make_node_close_cb(node)
        self._record_event("NODE", "REGISTERED", node.node_id,
                           f"{node.addr[0]}:{node.addr[1]}")
        await self._publish("NODE", node.node_id, node.view())
        return {"ok": True}

    def _make_node_close_cb(self, node: NodeInfo):
        def cb(conn):
            if node.alive:
                asyncio.get_running_loop().create_task(self._on_node_dead(node))
        return cb

    async def _on_node_dead(self, node: NodeInfo):
        node.alive = False
        self._record_event("NODE", "DEAD", node.node_id)
        await self._publish("NODE", node.node_id, node.view())
        # virtual-cluster replenishment (gcs_virtual_cluster_manager.cc:730):
        # a dead member of a count-based cluster is replaced from the
        # unassigned pool when possible
        for vc in self.virtual_clusters.values():
            if node.node_id in vc["nodes"]:
                vc["nodes"].discard(node.node_id)
                free = [n.node_id for n in self.nodes.values()
                        if n.alive and
                        n.node_id not in self._vc_assigned_nodes()]
                if free and len(vc["nodes"]) < vc["node_count"]:
                    vc["nodes"].add(free[0])
                vc["revision"] += 1
        # fail actors on that node
        for actor in list(self.actors.values()):
            if actor.node_id == node.node_id and actor.state in (ALIVE, PENDING_CREATION, RESTARTING):
                await self._on_actor_worker_died(actor, f"node {node.node_id.hex()[:8]} died")

    async def rpc_heartbeat(self, conn, p):
        node = self.nodes.get(p["node_id"])
        if node:
            node.last_heartbeat = time.monotonic()
            if "resources_available" in p:
                node.resources_available = p["resources_available"]
            node.pending_demands = p.get("pending_demands") or []
        return {"ok": True}

    async def rpc_pending_resource_demands(self, conn, p):
        """Queued lease demand across nodes (autoscaler input)."""
        out = []
        for node in self.nodes.values():
            if node.alive:
                out.extend(getattr(node, "pending_demands", []))
        return out

    async def rpc_node_table(self, conn, p):
        return [n.view() for n in self.nodes.values()]

    async def rpc_get_local_node(self, conn, p):
        """Pick the node whose store_path the caller should attach to."""
        ip = p.get("ip")
        for n in self.nodes.values():
            if n.alive and (ip is None or n.addr[0] == ip):
                return n.view()
        return None

    # ---------------------------------------------------------------- workers
    async def rpc_register_worker(self, conn, p):
        wid = p["worker_id"]
        is_driver = p.get("is_driver", False)
        job_id = p.get("job_id")
        if is_driver and job_id is None:
            self._job_counter += 1
            job_id = self._job_counter
            self.jobs[job_id] = {
                "job_id": job_id,
                "start_time": time.time(),
                "driver": wid,
                "state": "RUNNING",
            }
        self.workers[wid] = {
            "worker_id": wid,
            "addr": p.get("addr"),
            "node_id": p.get("node_id"),
            "is_driver": is_driver,
            "job_id": job_id,
            "pid": p.get("pid"),
        }
        vc = p.get("virtual_cluster_id")
        if vc:
            if vc not in self.virtual_clusters:
                return {"error": f"virtual cluster {vc!r} does not exist"}
            self.worker_vc[wid] = vc
            if is_driver and job_id in self.jobs:
                self.jobs[job_id]["virtual_cluster_id"] = vc
        conn.session["worker_id"] = wid
        self.workers[wid]["conn"] = conn
        prev = conn.on_close

        def _closed(c, wid=wid, is_driver=is_driver, job_id=job_id, prev=prev):
            if prev:
                prev(c)
            asyncio.get_running_loop().create_task(
                self._on_worker_conn_closed(wid, is_driver, job_id, c))

        conn.on_close = _closed
        return {"job_id": job_id}

    async def _on_worker_conn_closed(self, wid, is_driver, job_id, conn):
        """A registered worker/driver's GCS connection dropped. Grace
        period covers live reconnects (transient TCP drops); if the
        process is really gone and it was a DRIVER, its non-detached
        actors die with it (owner fate-sharing, reference job cleanup)
        and the job completes."""
        await asyncio.sleep(10.0)
        cur = self.workers.get(wid)
        if cur is None or cur.get("conn") is not conn:
            return  # re-registered (or already cleaned up)
        self.workers.pop(wid, None)
        if is_driver:
            logger.warning("driver %s disconnected; cleaning up its actors",
                           wid.hex()[:8] if isinstance(wid, bytes) else wid)
            await self._kill_actors_owned_by(wid, "driver exited")
            job = self.jobs.get(job_id)
            if job is not None and job.get("state") == "RUNNING":
                job["state"] = "FINISHED"
                job["end_time"] = time.time()
                self._persist_soon()

    async def rpc_pick_raylet(self, conn, p):
        """Name a feasible raylet for a task lease that is infeasible on
        the requester's node (spillback target), or — when
        preferred_worker names a worker holding the task's biggest arg —
        the raylet co-located with that data (locality-aware lease
        policy, parity: core_worker lease_policy.cc LocalityAwareLeasePolicy)."""
        res = dict(p.get("resources") or {})
        if p.get("spread"):
            node = self._pick_node(res, label_selector=p.get("_label_selector"),
                                   spread=True)
            if node is None:
                return {"addr": None}
            return {"addr": list(node.addr), "node_id": node.node_id,
                    "locality": True}
        pref = p.get("preferred_worker")
        if pref is not None:
            pref = tuple(pref)
            for w in self.workers.values():
                if w.get("addr") and tuple(w["addr"]) == pref:
                    node = self.nodes.get(w.get("node_id"))
                    if (node is not None and node.alive
                            and node.conn is not None and not node.conn.closed
                            and all(node.resources_total.get(k, 0) >= v
                                    for k, v in res.items())):
                        return {"addr": list(node.addr),
                                "node_id": node.node_id, "locality": True}
                    break
        node = self._pick_node(res, label_selector=p.get("_label_selector"))
        if node is None:
            return {"addr": None}
        return {"addr": list(node.addr), "node_id": node.node_id}

    async def rpc_report_worker_failure(self, conn, p):
        wid = p["worker_id"]
        self.workers.pop(wid, None)
        for actor in list(self.actors.values()):
            if actor.worker_id == wid and actor.state in (ALIVE, PENDING_CREATION, RESTARTING):
                await self._on_actor_worker_died(actor, p.get("reason", "worker died"))
        await self._kill_actors_owned_by(wid, p.get("reason", "owner died"))
        return {"ok": True}

    async def _kill_actors_owned_by(self, owner_wid: bytes, reason: str):
        """Owner fate-sharing (parity: reference non-detached actors die
        with their owner): when the creating worker/driver dies, its
        non-detached actors are torn down — e.g. a crashed Serve
        controller's replicas, so its restart rebuilds a clean set."""
        for actor in list(self.actors.values()):
            if (actor.owner == owner_wid
                    and actor.state in (ALIVE, PENDING_CREATION, RESTARTING)
                    and actor.opts.get("lifetime") != "detached"):
                actor.opts["max_restarts"] = 0
                actor.killed = True
                if actor.addr:
                    try:
                        wconn = await self._get_worker_conn(tuple(actor.addr))
                        await wconn.notify("exit_worker", {
                            "reason": f"owner died: {reason}"})
                    except Exception:
                        pass


    # -------------------------------------------------- virtual clusters
    # parity: ant fork GcsVirtualClusterManager (gcs_virtual_cluster.h,
    # doc/source/virtual-cluster/design-overview.rst)

    def _vc_assigned_nodes(self):
        out = set()
        for vc in self.virtual_clusters.values():
            out |= vc["nodes"]
        return out

    def _vc_allowed(self, vc_id):
        """Node-id set a vc may schedule on, or None = unrestricted."""
        if not vc_id:
            return None
        vc = self.virtual_clusters.get(vc_id)
        return vc["nodes"] if vc is not None else set()

    async def rpc_create_or_update_virtual_cluster(self, conn, p):
        vc_id = p["virtual_cluster_id"]
        vc = self.virtual_clusters.get(vc_id)
        if vc is None:
            vc = {"id": vc_id, "divisible": bool(p.get("divisible", False)),
                  "nodes": set(), "node_count": 0, "revision": 0}
            self.virtual_clusters[vc_id] = vc
            self._persist_soon()
        if "node_ids" in p and p["node_ids"] is not None:
            wanted = {bytes.fromhex(n) if isinstance(n, str) else n
                      for n in p["node_ids"]}
            taken = self._vc_assigned_nodes() - vc["nodes"]
            conflict = wanted & taken
            if conflict:
                return {"ok": False, "error": "nodes already assigned: "
                        + ",".join(n.hex()[:8] for n in conflict)}
            vc["nodes"] = wanted
            vc["node_count"] = len(wanted)
        else:
            want = int(p.get("node_count", 0))
            vc["node_count"] = want
            free = [n.node_id for n in self.nodes.values()
                    if n.alive and n.node_id not in self._vc_assigned_nodes()]
            while len(vc["nodes"]) < want and free:
                vc["nodes"].add(free.pop(0))
            while len(vc["nodes"]) > want:
                vc["nodes"].pop()
            if len(vc["nodes"]) < want:
                vc["revision"] += 1
                return {"ok": False,
                        "error": f"only {len(vc['nodes'])} of {want} nodes "
                                 "available", "view": self._vc_view(vc)}
        vc["revision"] += 1
        return {"ok": True, "view": self._vc_view(vc)}

    def _vc_view(self, vc):
        return {"virtual_cluster_id": vc["id"], "divisible": vc["divisible"],
                "node_ids": [n.hex() for n in vc["nodes"]],
                "node_count": vc["node_count"], "revision": vc["revision"]}

    async def rpc_remove_virtual_cluster(self, conn, p):
        vc = self.virtual_clusters.pop(p["virtual_cluster_id"], None)
        if vc is not None:
            self._persist_soon()
        return {"ok": vc is not None}

    async def rpc_list_virtual_clusters(self, conn, p):
        return [self._vc_view(vc) for vc in self.virtual_clusters.values()]

    async def rpc_get_virtual_cluster(self, conn, p):
        vc = self.virtual_clusters.get(p["virtual_cluster_id"])
        return self._vc_view(vc) if vc else None

    # --------------------------------------------------------------------- kv
    async def rpc_kv_put(self, conn, p):
        ns = self.kv.setdefault(p.get("ns", ""), {})
        key = p["key"]
        if not p.get("overwrite", True) and key in ns:
            return {"added": False}
        seq = p.get("seq")
        if seq is not None:
            # last-writer-wins by CLIENT sequence: concurrent handler
            # tasks (or chaos-delayed ones) must not let a stale write
            # overwrite a newer value (metrics publishes rely on this).
            # Scoped per client (seq_id): each publisher's counter starts
            # at 0, so comparing across clients would silently reject a
            # restarted worker's writes until it caught up
            if not hasattr(self, "_kv_seq"):
                self._kv_seq = {}
            k = (p.get("seq_id", b""), p.get("ns", ""), key)
            if seq < self._kv_seq.get(k, -1):
                return {"added": False, "stale": True}
            self._kv_seq[k] = seq
        ns[key] = p["value"]
        self._persist_soon()
        return {"added": True}

    async def rpc_kv_get(self, conn, p):
        ns = self.kv.get(p.get("ns", ""), {})
        return {"value": ns.get(p["key"])}

    async def rpc_kv_del(self, conn, p):
        ns = self.kv.get(p.get("ns", ""), {})
        existed = ns.pop(p["key"], None) is not None
        if existed:
            if hasattr(self, "_kv_seq"):
                # drop seq state for the deleted key (any client): a fresh
                # publisher of a re-created key starts clean
                nsk = p.get("ns", "")
                for k in [k for k in self._kv_seq
                          if k[1] == nsk and k[2] == p["key"]]:
                    del self._kv_seq[k]
            self._persist_soon()
        return {"deleted": existed}

    async def rpc_kv_keys(self, conn, p):
        ns = self.kv.get(p.get("ns", ""), {})
        prefix = p.get("prefix", b"")
        return {"keys": [k for k in ns.keys() if k.startswith(prefix)]}

    async def rpc_next_job_id(self, conn, p):
        self._job_counter += 1
        self._persist_soon()
        return {"job_id": self._job_counter}

    # ----------------------------------------------------------------- pubsub
    async def rpc_subscribe(self, conn, p):
        for ch in p["channels"]:
            subs = self._subscribers.setdefault(ch, [])
            if conn not in subs:
                subs.append(conn)
        return {"ok": True}

    def _record_event(self, source: str, event: str, entity_id, message: str = ""):
        """Structured cluster lifecycle events (parity: observability/
        ray_event_recorder.cc definition+lifecycle events; surfaced by
        `ray list cluster-events` and state.list_cluster_events)."""
        import collections

        if not hasattr(self, "cluster_events"):
            self.cluster_events = collections.deque(maxlen=10000)
        self._event_seq = getattr(self, "_event_seq", 0) + 1
        self.cluster_events.append({
            "seq": self._event_seq,    # export-pipeline high-water mark
            "timestamp": time.time(),
            "source": source,          # NODE | ACTOR | JOB | WORKER
            "event": event,            # REGISTERED | ALIVE | DEAD | ...
            "entity_id": entity_id.hex() if isinstance(entity_id, bytes)
            else str(entity_id),
            "message": message,
        })

    async def rpc_list_cluster_events(self, conn, p):
        limit = p.get("limit", 1000)
        evs = list(getattr(self, "cluster_events", []))
        if p.get("source"):
            evs = [e for e in evs if e["source"] == p["source"]]
        return evs[-limit:]

    async def _publish(self, channel: str, key: bytes, data: Any):
        if channel == "ACTOR":
            self._persist_soon()
        subs = self._subscribers.get(channel, [])
        dead = []
        for c in subs:
            if c.closed:
                dead.append(c)
                continue
            try:
                await c.notify("pub", {"channel": channel, "key": key, "data": data})
            except Exception:
                dead.append(c)
        for c in dead:
            try:
                subs.remove(c)
            except ValueError:
                pass

    # ----------------------------------------------------------------- actors
    async def rpc_create_actor(self, conn, p):
        actor_id = p["actor_id"]
        name = p.get("name") or ""
        namespace = p.get("namespace") or ""
        if name:
            key = (namespace, name)
            existing_id = self.named_actors.get(key)
            if existing_id is not None:
                existing = self.actors.get(existing_id)
                if existing is not None and existing.state != DEAD:
                    if p.get("get_if_exists"):
                        return {"existing": True, "actor_id": existing_id}
                    raise ValueError(f"actor name '{name}' already taken")
            self.named_actors[key] = actor_id
        info = ActorInfo(
            actor_id, p.get("owner"), name, namespace, p.get("opts", {}), p["create_payload"],
        )
        self.actors[actor_id] = info
        asyncio.get_running_loop().create_task(self._schedule_actor(info))
        return {"existing": False, "actor_id": actor_id}

    def _pick_node(self, resources: Dict[str, float], pg: Optional[dict] = None,
                   node_affinity: Optional[bytes] = None,
                   allowed: Optional[set] = None,
                   label_selector: Optional[dict] = None,
                   spread: bool = False) -> Optional[NodeInfo]:
        if pg:
            pg_info = self.pgs.get(pg["pg_id"])
            if pg_info and pg_info.state == "CREATED":
                idx = pg.get("bundle_index", 0)
                if idx < 0:
                    idx = 0
                nid = pg_info.bundle_nodes[idx]
                node = self.nodes.get(nid)
                if node and node.alive:
                    return node
            return None
        if node_affinity is not None:
            node = self.nodes.get(node_affinity)
            return node if node and node.alive else None
        # feasibility + best-fit by available CPU fraction
        best, best_score = None, -1.0
        for node in self.nodes.values():
            if not node.alive:
                continue
            if allowed is not None and node.node_id not in allowed:
                continue
            if label_selector:
                from ant_ray_amd._private.raylet import _labels_match

                if not _labels_match(label_selector.get("hard") or {},
                                     node.labels):
                    continue
            feasible = all(
                node.resources_total.get(k, 0) >= v for k, v in resources.items()
            )
            if not feasible:
                continue
            available = all(
                node.resources_available.get(k, 0) >= v for k, v in resources.items()
            )
            score = 1.0 if available else 0.0
            score += node.resources_available.get("CPU", 0) / max(
                node.resources_total.get("CPU", 1), 1
            )
            if label_selector and label_selector.get("soft"):
                from ant_ray_amd._private.raylet import _labels_match

                if _labels_match(label_selector["soft"], node.labels):
                    score += 10.0  # prefer soft-matching nodes
            if spread:
                # SPREAD strategy: round-robin across feasible nodes
                # (reference spread_scheduling_policy) — rotate by a
                # global counter so consecutive picks alternate nodes
                self._spread_seq = getattr(self, "_spread_seq", 0) + 1
                import hashlib as _h

                rot = int(_h.blake2b(
                    node.node_id + self._spread_seq.to_bytes(4, "little"),
                    digest_size=2).hexdigest(), 16)
                score += rot / 65536.0
            if score > best_score:
                best, best_score = node, score
        return best

    async def _schedule_actor(self, info: ActorInfo, delay: float = 0.0):
        if delay:
            await asyncio.sleep(delay)
        opts = info.opts
        resources = dict(opts.get("resources") or {})
        resources.setdefault("CPU", float(opts.get("num_cpus", 1)))
        if opts.get("num_gpus"):
            resources["GPU"] = float(opts["num_gpus"])
        pg = opts.get("placement_group")
        deadline = time.monotonic() + float(opts.get("_scheduling_timeout", 3600.0))
        vc_id = opts.get("virtual_cluster_id") or self.worker_vc.get(info.owner)
        while True:
            if info.killed:
                info.state = DEAD
                info.death_cause = "actor killed before creation completed"
                self._record_event("ACTOR", "DEAD", info.actor_id, info.death_cause)
                await self._publish("ACTOR", info.actor_id, info.view())
                self._wake_waiters(info)
                return
            node = self._pick_node(resources, pg, opts.get("_node_affinity"),
                                   self._vc_allowed(vc_id),
                                   opts.get("_label_selector"),
                                   spread=bool(opts.get("_spread")))
            if node is not None and node.conn is not None and not node.conn.closed:
                # optimistic accounting: concurrent schedulings must not all
                # pile onto the same node while its heartbeat is stale (the
                # reference's ClusterResourceManager debits its view the
                # same way); the next heartbeat trues it up
                if not pg:
                    for k, v in resources.items():
                        node.resources_available[k] = (
                            node.resources_available.get(k, 0) - v)
                try:
                    lease = await node.conn.call(
                        "lease_worker",
                        {
                            "resources": resources,
                            "actor_id": info.actor_id,
                            "runtime_env": opts.get("runtime_env"),
                            "pg": pg,
                            "detached": bool(opts.get("lifetime") == "detached"),
                            # deny-don't-queue: a full node must not hold
                            # this scheduling loop while other nodes are
                            # free (it retries every 0.2s)
                            "no_wait": True,
                        },
                        timeout=120,
                    )
                except Exception as e:
                    logger.warning("actor lease on node failed: %s", e)
                    lease = None
                if lease and lease.get("granted"):
                    await self._push_actor_creation(info, node, lease)
                    return
                if not pg:
                    # denied/failed: credit the optimistic debit back now
                    # instead of waiting ~1s for the next heartbeat to
                    # true it up (the retry loop would otherwise see a
                    # phantom-full node)
                    for k, v in resources.items():
                        node.resources_available[k] = (
                            node.resources_available.get(k, 0) + v)
            if time.monotonic() > deadline or self._shutdown.is_set():
                info.state = DEAD
                self._record_event("ACTOR", "DEAD", info.actor_id, info.death_cause)
                info.death_cause = "scheduling timed out (insufficient resources)"
                await self._publish("ACTOR", info.actor_id, info.view())
                self._wake_waiters(info)
                return
            await asyncio.sleep(0.2)

    async def _push_actor_creation(self, info: ActorInfo, node: NodeInfo, lease):
        addr = tuple(lease["addr"])
        if info.killed:
            # killed between lease grant and creation push: hand the
            # leased worker back (its death returns the resources)
            try:
                wconn = await self._get_worker_conn(addr)
                await wconn.notify("exit_worker", {"reason": "ray.kill"})
            except Exception:
                pass
            info.state = DEAD
            info.death_cause = "actor killed before creation completed"
            self._record_event("ACTOR", "DEAD", info.actor_id, info.death_cause)
            await self._publish("ACTOR", info.actor_id, info.view())
            self._wake_waiters(info)
            return
        info.worker_id = lease["worker_id"]
        info.node_id = node.node_id
        try:
            wconn = await self._get_worker_conn(addr)
            payload = dict(info.create_payload)
            payload["actor_id"] = info.actor_id
            payload["num_restarts"] = info.num_restarts
            reply = await wconn.call("push_task", payload, timeout=None)
        except Exception as e:
            logger.warning("actor creation push failed: %s", e)
            await self._on_actor_worker_died(info, f"creation push failed: {e}")
            return
        if reply.get("status") == "ok":
            info.state = ALIVE
            self._record_event("ACTOR", "ALIVE", info.actor_id, info.name)
            info.addr = addr
            await self._publish("ACTOR", info.actor_id, info.view())
            self._wake_waiters(info)
            if info.killed:  # kill raced the creation push: finish it now
                try:
                    wconn = await self._get_worker_conn(addr)
                    await wconn.notify("exit_worker", {"reason": "ray.kill"})
                except Exception:
                    pass
        else:
            info.state = DEAD
            self._record_event("ACTOR", "DEAD", info.actor_id, info.death_cause)
            info.death_cause = reply.get("error", "actor __init__ failed")
            info.creation_error = reply.get("error_payload")
            await self._publish("ACTOR", info.actor_id, info.view())
            self._wake_waiters(info)

    def _wake_waiters(self, info: ActorInfo):
        for fut in info.pending_waiters:
            if not fut.done():
                fut.set_result(info.view())
        info.pending_waiters.clear()

    async def _get_worker_conn(self, addr: Tuple[str, int]) -> Connection:
        conn = self._worker_conns.get(addr)
        if conn is None or conn.closed:
            conn = await protocol.connect(addr, self._handle, name=f"gcs->worker{addr}")
            self._worker_conns[addr] = conn
        return conn

    async def _on_actor_worker_died(self, info: ActorInfo, reason: str):
        if info.state == DEAD:
            return
        max_restarts = int(info.opts.get("max_restarts", 0))
        if max_restarts == -1 or info.num_restarts < max_restarts:
            info.num_restarts += 1
            info.state = RESTARTING
            self._record_event("ACTOR", "RESTARTING", info.actor_id)
            info.addr = None
            info.worker_id = None
            await self._publish("ACTOR", info.actor_id, info.view())
            asyncio.get_running_loop().create_task(self._schedule_actor(info, delay=0.1))
        else:
            info.state = DEAD
            self._record_event("ACTOR", "DEAD", info.actor_id, info.death_cause)
            info.death_cause = reason
            info.addr = None
            await self._publish("ACTOR", info.actor_id, info.view())
            self._wake_waiters(info)

    async def rpc_get_actor(self, conn, p):
        info = self.actors.get(p["actor_id"])
        return info.view() if info else None

    async def rpc_list_named_actors(self, conn, p):
        # parity: GCS ListNamedActors (ray.util.list_named_actors)
        out = []
        for (namespace, name), aid in self.named_actors.items():
            info = self.actors.get(aid)
            if info is None or info.state == DEAD:
                continue
            if p.get("all_namespaces") or namespace == (p.get("namespace") or ""):
                out.append({"namespace": namespace, "name": name})
        return {"actors": out}

    async def rpc_get_actor_by_name(self, conn, p):
        aid = self.named_actors.get((p.get("namespace") or "", p["name"]))
        if aid is None:
            return None
        info = self.actors.get(aid)
        if info is None or info.state == DEAD:
            # a dead actor's name is free (reference GCS removes the name
            # on death); callers re-creating under the same name must not
            # receive a handle to the corpse
            return None
        return info.view()

    async def rpc_wait_actor_ready(self, conn, p):
        info = self.actors.get(p["actor_id"])
        if info is None:
            return None
        if info.state in (ALIVE, DEAD):
            view = info.view()
            if info.state == DEAD:
                view["creation_error"] = getattr(info, "creation_error", None)
            return view
        fut = asyncio.get_running_loop().create_future()
        info.pending_waiters.append(fut)
        view = await fut
        if view.get("state") == DEAD:
            view["creation_error"] = getattr(info, "creation_error", None)
        return view

    async def rpc_kill_actor(self, conn, p):
        info = self.actors.get(p["actor_id"])
        if info is None:
            return {"ok": False}
        no_restart = p.get("no_restart", True)
        if no_restart:
            info.opts["max_restarts"] = 0
            # an actor killed while STILL SCHEDULING must not become a
            # zombie: the _schedule_actor loop checks this flag before
            # leasing and before pushing the creation task (otherwise a
            # handle GC'd during creation leaves an unkillable actor
            # holding its GPU/CPU fraction and starving later leases)
            info.killed = True
        if info.addr:
            try:
                wconn = await self._get_worker_conn(tuple(info.addr))
                await wconn.notify("exit_worker", {"reason": "ray.kill"})
            except Exception:
                pass
        # actual state transition happens when the raylet reports worker death
        return {"ok": True}

    async def rpc_actor_out_of_scope(self, conn, p):
        """All handles to a non-detached actor dropped -> tear it down."""
        return await self.rpc_kill_actor(conn, p)

    async def rpc_list_actors(self, conn, p):
        return [a.view() for a in self.actors.values()]

    async def rpc_store_stats(self, conn, p):
        """Fan out per-node object-store stats (for `ray memory`)."""
        out = []
        for node in list(self.nodes.values()):
            if node.conn is None or node.conn.closed:
                continue
            try:
                out.append(await node.conn.call("store_stats", {}, timeout=5))
            except Exception:
                pass
        return out

    # --------------------------------------------------------- task events
    # parity: GcsTaskManager (src/ray/gcs/gcs_task_manager.h) fed by the
    # CoreWorker TaskEventBuffer; ring buffer, newest wins
    async def rpc_task_events(self, conn, p):
        import collections

        if not hasattr(self, "task_events"):
            self.task_events = collections.deque(maxlen=20000)
        for ev in p["events"]:
            self._event_seq = getattr(self, "_event_seq", 0) + 1
            ev["seq"] = self._event_seq
            self.task_events.append(ev)
        return {"ok": True}

    async def rpc_list_task_events(self, conn, p):
        limit = p.get("limit", 1000)
        evs = list(getattr(self, "task_events", []))
        return evs[-limit:]

    async def rpc_list_workers(self, conn, p):
        return [{k: v for k, v in w.items() if k != "conn"}
                for w in self.workers.values()]

    async def rpc_list_jobs(self, conn, p):
        return list(self.jobs.values())

    # ------------------------------------------------------------ placement
    async def rpc_create_placement_group(self, conn, p):
        pg = PlacementGroupInfo(p["pg_id"], p["bundles"], p.get("strategy", "PACK"), p.get("name", ""))
        pg.vc_id = self.worker_vc.get(conn.session.get("worker_id"))
        self.pgs[pg.pg_id] = pg
        asyncio.get_running_loop().create_task(self._schedule_pg(pg))
        return {"ok": True}

    async def _schedule_pg(self, pg: PlacementGroupInfo, timeout_s: float = 3600.0):
        deadline = time.monotonic() + timeout_s
        while time.monotonic() < deadline and not self._shutdown.is_set():
            placement = self._plan_pg(pg)
            if placement is not None:
                # 2PC-lite: reserve each bundle on its raylet; rollback on fail
                reserved = []
                ok = True
                for idx, node in placement:
                    try:
                        r = await node.conn.call(
                            "reserve_bundle",
                            {"pg_id": pg.pg_id, "bundle_index": idx, "resources": pg.bundles[idx]},
                            timeout=30,
                        )
                    except Exception:
                        r = None
                    if not r or not r.get("ok"):
                        ok = False
                        break
                    reserved.append((idx, node))
                if ok:
                    for idx, node in placement:
                        pg.bundle_nodes[idx] = node.node_id
                    pg.state = "CREATED"
                    await self._publish("PG", pg.pg_id, pg.view())
                    return
                for idx, node in reserved:
                    try:
                        await node.conn.call("return_bundle", {"pg_id": pg.pg_id, "bundle_index": idx})
                    except Exception:
                        pass
            await asyncio.sleep(0.2)
        pg.state = "FAILED"
        await self._publish("PG", pg.pg_id, pg.view())

    def _plan_pg(self, pg: PlacementGroupInfo):
        """Return [(bundle_index, NodeInfo)] or None if not placeable now."""
        alive = [n for n in self.nodes.values() if n.alive]
        allowed = self._vc_allowed(getattr(pg, "vc_id", None))
        if allowed is not None:
            alive = [n for n in alive if n.node_id in allowed]
        if not alive:
            return None
        avail = {n.node_id: dict(n.resources_available) for n in alive}

        def fits(nid, res):
            return all(avail[nid].get(k, 0) >= v for k, v in res.items())

        def take(nid, res):
            for k, v in res.items():
                avail[nid][k] = avail[nid].get(k, 0) - v

        placement = []
        strategy = pg.strategy
        if strategy in ("STRICT_PACK", "PACK"):
            # try to fit all on one node first
            for n in alive:
                trial = dict(avail[n.node_id])
                ok = True
                for b in pg.bundles:
                    if all(trial.get(k, 0) >= v for k, v in b.items()):
                        for k, v in b.items():
                            trial[k] = trial.get(k, 0) - v
                    else:
                        ok = False
                        break
                if ok:
                    return [(i, n) for i in range(len(pg.bundles))]
            if strategy == "STRICT_PACK":
                return None
        if strategy == "STRICT_SPREAD" and len(pg.bundles) > len(alive):
            return None
        # spread/fallback: round-robin over nodes with capacity
        used_nodes = set()
        for i, b in enumerate(pg.bundles):
            candidates = sorted(
                alive,
                key=lambda n: (n.node_id in used_nodes, -avail[n.node_id].get("CPU", 0)),
            )
            placed = False
            for n in candidates:
                if strategy == "STRICT_SPREAD" and n.node_id in used_nodes:
                    continue
                if fits(n.node_id, b):
                    take(n.node_id, b)
                    placement.append((i, n))
                    used_nodes.add(n.node_id)
                    placed = True
                    break
            if not placed:
                return None
        return placement

    async def rpc_get_placement_group(self, conn, p):
        pg = self.pgs.get(p["pg_id"])
        return pg.view() if pg else None

    async def rpc_remove_placement_group(self, conn, p):
        pg = self.pgs.pop(p["pg_id"], None)
        if pg and pg.state == "CREATED":
            for idx, nid in enumerate(pg.bundle_nodes):
                node = self.nodes.get(nid)
                if node and node.conn and not node.conn.closed:
                    try:
                        await node.conn.call("return_bundle", {"pg_id": pg.pg_id, "bundle_index": idx})
                    except Exception:
                        pass
        return {"ok": True}

    async def rpc_list_placement_groups(self, conn, p):
        return [pg.view() for pg in self.pgs.values()]

    # ----------------------------------------------------------------- state
    async def rpc_cluster_resources(self, conn, p):
        total: Dict[str, float] = {}
        avail: Dict[str, float] = {}
        for n in self.nodes.values():
            if not n.alive:
                continue
            for k, v in n.resources_total.items():
                total[k] = total.get(k, 0) + v
            for k, v in n.resources_available.items():
                avail[k] = avail.get(k, 0) + v
        return {"total": total, "available": avail}

    async def rpc_shutdown(self, conn, p):
        self._shutdown.set()
        for node in self.nodes.values():
            if node.conn and not node.conn.closed:
                try:
                    await node.conn.notify("shutdown", {})
                except Exception:
                    pass
        asyncio.get_running_loop().call_later(0.2, lambda: os._exit(0))
        return {"ok": True}


async def run_gcs(host="127.0.0.1", port=0, announce_fd: int = None,
                  persist_path: str = None):
    gcs = GcsServer()
    bound = await gcs.start(host, port, persist_path=persist_path)
    if announce_fd is not None:
        os.write(announce_fd, (str(bound) + "\n").encode())
        os.close(announce_fd)
    await gcs._shutdown.wait()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=0)
    ap.add_argument("--persist", default=None,
                    help="snapshot/restore durable tables at this path")
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)
    asyncio.run(run_gcs(args.host, args.port, persist_path=args.persist))


if __name__ == "__main__":
    main()
