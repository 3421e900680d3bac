"""Raylet — the per-node daemon: worker pool + lease-based local scheduler +
shm object store host.

Role parity: reference raylet (src/ray/raylet/node_manager.h:134 NodeManager;
lease protocol HandleRequestWorkerLease node_manager.cc:1794 ->
ClusterLeaseManager -> LocalLeaseManager; WorkerPool worker_pool.h:285 forks
language workers; placement-group bundle reservation
placement_group_resource_manager.cc). Differences by design: the shm store is
direct-mapped (no store server thread), and scheduling is lease-queue +
local-first with spillback hints from the GCS node table rather than a full
cluster resource view (the GCS owns the global view).

GPU accounting: the raylet owns the node's HIP device id pool; a lease that
asks for GPUs is granted specific ids which the worker exports as
HIP_VISIBLE_DEVICES before user code initializes the runtime (parity with
python/ray/_private/accelerators/amd_gpu.py:10).
"""
from __future__ import annotations

import argparse
import asyncio
import json
import logging
import os
import signal
import subprocess
import sys
import time
from typing import Any, Dict, List, Optional, Tuple

from ant_ray_amd._private import protocol
from ant_ray_amd._private.ids import NodeID, WorkerID
from ant_ray_amd._private.protocol import Connection

logger = logging.getLogger("antray.raylet")

IDLE_WORKER_KILL_S = 60.0


def detect_num_gpus() -> int:
    """Count AMD GPUs without importing torch (KFD topology)."""
    if "HIP_VISIBLE_DEVICES" in os.environ:
        v = os.environ["HIP_VISIBLE_DEVICES"]
        return 0 if v == "" else len(v.split(","))
    base = "/sys/class/kfd/kfd/topology/nodes"
    count = 0
    try:
        for d in os.listdir(base):
            p = os.path.join(base, d, "gpu_id")
            try:
                with open(p) as f:
                    if int(f.read().strip()) != 0:
                        count += 1
            except Exception:
                pass
    except FileNotFoundError:
        pass
    return count


class WorkerProc:
    def __init__(self, worker_id: bytes, proc: subprocess.Popen):
        self.worker_id = worker_id
        self.proc = proc
        self.conn: Optional[Connection] = None
        self.addr: Optional[Tuple[str, int]] = None
        self.pid = proc.pid
        self.registered = asyncio.Event()
        self.leased = False
        self.lease_owner: Optional[Connection] = None
        self.lease_resources: Dict[str, float] = {}
        self.gpu_ids: List[int] = []
        self.is_actor = False
        self.detached_actor = False
        self.idle_since = time.monotonic()


def _labels_match(hard: dict, labels: dict) -> bool:
    """True when every hard constraint matches this node's labels; a list
    value means "in" (reference label-selector semantics, common.proto
    LabelMatchExpressions reduced to equality/in)."""
    for k, v in (hard or {}).items():
        have = labels.get(k)
        if isinstance(v, (list, tuple, set)):
            if have not in v:
                return False
        elif have != v:
            return False
    return True


class Raylet:
    def __init__(self, node_ip: str, gcs_addr: Tuple[str, int], resources: Dict[str, float],
                 store_path: str, store_capacity: int, session_dir: str):
        self.node_id = NodeID.from_random().binary()
        self.node_ip = node_ip
        self.gcs_addr = gcs_addr
        self.resources_total = dict(resources)
        self.resources_available = dict(resources)
        self.store_path = store_path
        self.store_capacity = store_capacity
        self.session_dir = session_dir
        # node labels (NodeLabelSchedulingStrategy parity); auto node-id label
        self.labels: dict = {"ray.io/node_id": self.node_id.hex()}
        self.workers: Dict[bytes, WorkerProc] = {}
        self.idle_workers: List[WorkerProc] = []
        self.pending_leases: List[Tuple[dict, asyncio.Future]] = []
        self.gpu_pool: Dict[int, float] = {}  # gpu id -> available fraction
        ngpus = int(self.resources_total.get("GPU", 0))
        for i in range(ngpus):
            self.gpu_pool[i] = 1.0
        # placement group bundle reservations: (pg_id, idx) -> available dict
        self.bundles: Dict[Tuple[bytes, int], Dict[str, float]] = {}
        self.gcs_conn: Optional[Connection] = None
        self.port = None
        self._server = None
        self._store = None
        self._shutdown = asyncio.Event()

    # ----------------------------------------------------------------- start
    async def start(self, port=0):
        # host the node's shm store
        from ant_ray_amd._shm_store import ShmStore

        if not os.path.exists(self.store_path):
            self._store = ShmStore.create(self.store_path, self.store_capacity)
        else:
            self._store = ShmStore.open(self.store_path)
        self._server, self.port = await protocol.serve(self._handle, self.node_ip, port)
        self.gcs_conn = await protocol.connect(self.gcs_addr, self._handle, name="raylet->gcs")
        self.gcs_conn.on_close = self._on_gcs_conn_closed
        await self.gcs_conn.call("register_node", self._register_payload())
        asyncio.get_running_loop().create_task(self._heartbeat_loop())
        asyncio.get_running_loop().create_task(self._reap_loop())
        asyncio.get_running_loop().create_task(self._memory_monitor_loop())
        # per-node physical/GPU stats reporter (parity: reference
        # dashboard/modules/reporter/reporter_agent.py per-node agent)
        from ant_ray_amd.dashboard.reporter import reporter_loop

        asyncio.get_running_loop().create_task(reporter_loop(self))
        logger.info(
            "raylet %s on %s:%s (%s)", self.node_id.hex()[:8], self.node_ip, self.port,
            {k: v for k, v in self.resources_total.items()},
        )
        return self.port

    def _register_payload(self):
        return {
            "node_id": self.node_id,
            "addr": [self.node_ip, self.port],
            "resources": self.resources_total,
            "store_path": self.store_path,
            "object_store_bytes": self.store_capacity,
            "labels": self.labels,
        }

    def _on_gcs_conn_closed(self, conn):
        """GCS fault tolerance (parity: reference raylets survive a GCS
        restart and re-register via NotifyGCSRestart, node_manager.proto
        :446): reconnect with backoff instead of fate-sharing; exit only
        after the reconnect window expires."""
        if self._shutdown.is_set():
            return
        try:
            asyncio.get_running_loop().create_task(self._reconnect_gcs())
        except RuntimeError:
            os._exit(1)  # event loop gone: process is tearing down anyway

    async def _reconnect_gcs(self):
        window = float(os.environ.get("ANTRAY_GCS_RECONNECT_TIMEOUT_S", "60"))
        deadline = time.monotonic() + window
        logger.warning("GCS connection lost; reconnecting for up to %.0fs",
                       window)
        while time.monotonic() < deadline and not self._shutdown.is_set():
            try:
                conn = await protocol.connect(self.gcs_addr, self._handle,
                                              name="raylet->gcs")
                conn.on_close = self._on_gcs_conn_closed
                await conn.call("register_node", self._register_payload(),
                                timeout=10)
                self.gcs_conn = conn
                logger.warning("re-registered with restarted GCS")
                return
            except Exception:
                await asyncio.sleep(1.0)
        if self._shutdown.is_set():
            return
        logger.error("GCS unreachable after %.0fs; raylet exiting", window)
        os._exit(1)

    async def _heartbeat_loop(self):
        while not self._shutdown.is_set():
            try:
                await self.gcs_conn.call(
                    "heartbeat",
                    {"node_id": self.node_id,
                     "resources_available": self.resources_available,
                     # queued lease demand (parity: reference resource_load
                     # in the syncer messages — what the autoscaler scales
                     # for beyond pending actors/PGs)
                     "pending_demands": [
                         dict(p.get("resources") or {})
                         for (p, _conn), _fut in self.pending_leases[:64]
                     ] + [r for t, r in getattr(self, "_infeasible", [])
                          if time.monotonic() - t < 10.0]},
                    timeout=5,
                )
            except Exception:
                pass
            await asyncio.sleep(1.0)

    async def _reap_loop(self):
        """Detect dead worker processes + kill over-provisioned idle workers."""
        while not self._shutdown.is_set():
            await asyncio.sleep(0.5)
            for w in list(self.workers.values()):
                if w.proc.poll() is not None:
                    await self._on_worker_dead(w, f"exit code {w.proc.returncode}")
            now = time.monotonic()
            max_idle = int(self.resources_total.get("CPU", 1))
            if len(self.idle_workers) > max_idle:
                for w in [w for w in self.idle_workers if now - w.idle_since > IDLE_WORKER_KILL_S][: len(self.idle_workers) - max_idle]:
                    self._kill_worker(w)

    # ------------------------------------------------------- memory monitor
    # Parity: src/ray/common/memory_monitor.h (RAY_memory_usage_threshold
    # 0.95, refresh-interval loop) + raylet worker_killing_policy.h: when
    # node memory crosses the threshold, kill the largest leased worker,
    # preferring retriable normal-task workers over actors.
    @staticmethod
    def _node_memory_fraction() -> float:
        total = avail = None
        with open("/proc/meminfo") as f:
            for line in f:
                if line.startswith("MemTotal:"):
                    total = float(line.split()[1])
                elif line.startswith("MemAvailable:"):
                    avail = float(line.split()[1])
                if total is not None and avail is not None:
                    break
        if not total:
            return 0.0
        return 1.0 - (avail or 0.0) / total

    @staticmethod
    def _rss_bytes(pid: int) -> int:
        try:
            with open(f"/proc/{pid}/statm") as f:
                return int(f.read().split()[1]) * os.sysconf("SC_PAGE_SIZE")
        except Exception:
            return 0

    def _pick_oom_victim(self) -> Optional[WorkerProc]:
        leased = [w for w in self.workers.values()
                  if w.leased and w.proc.poll() is None]
        if not leased:
            return None
        # retriable task workers first (group), then actors; largest RSS
        tasks = [w for w in leased if not w.is_actor]
        pool = tasks or [w for w in leased if not w.detached_actor] or leased
        return max(pool, key=lambda w: self._rss_bytes(w.pid))

    async def _memory_monitor_loop(self):
        threshold = float(os.environ.get("RAY_memory_usage_threshold", "0.95"))
        period = float(os.environ.get(
            "RAY_memory_monitor_refresh_ms", "1000")) / 1000.0
        if period <= 0:
            return  # monitor disabled (parity: refresh_ms = 0)
        while not self._shutdown.is_set():
            await asyncio.sleep(period)
            try:
                frac = self._node_memory_fraction()
            except Exception:
                continue
            if frac <= threshold:
                continue
            victim = self._pick_oom_victim()
            if victim is None:
                continue
            logger.warning(
                "memory monitor: node memory %.1f%% > %.1f%% — killing "
                "worker %s (rss %.1f MB)", frac * 100, threshold * 100,
                victim.worker_id.hex()[:8],
                self._rss_bytes(victim.pid) / 1e6)
            victim.oom_killed = True
            if victim.lease_owner is not None and not victim.lease_owner.closed:
                try:
                    await asyncio.wait_for(victim.lease_owner.call(
                        "worker_killed_notice",
                        {"worker_id": victim.worker_id,
                         "reason": f"worker killed by the node memory "
                                   f"monitor: node memory usage {frac:.2f} > "
                                   f"threshold {threshold:.2f} "
                                   f"(OutOfMemoryError)"}, timeout=2), 3)
                except Exception:
                    pass
            self._kill_worker(victim)
            await self._on_worker_dead(
                victim,
                f"killed by the memory monitor: node memory usage {frac:.2f} "
                f"exceeds threshold {threshold:.2f} (OutOfMemoryError)")

    async def _on_worker_dead(self, w: WorkerProc, reason: str):
        self.workers.pop(w.worker_id, None)
        if w in self.idle_workers:
            self.idle_workers.remove(w)
        if w.leased:
            self._release_resources(w)
        # reclaim leases this worker held as LESSEE (e.g. an actor that was
        # submitting tasks when it died): without this its leased workers
        # stay marked busy forever and the node starves
        if w.conn is not None:
            self._reclaim_leases_of(w.conn)
        try:
            await self.gcs_conn.call(
                "report_worker_failure", {"worker_id": w.worker_id, "reason": reason}, timeout=5
            )
        except Exception:
            pass
        self._pump_leases()

    # -------------------------------------------------------------- dispatch
    async def _handle(self, conn: Connection, method: str, p: Any):
        fn = getattr(self, "rpc_" + method, None)
        if fn is None:
            raise ValueError(f"unknown raylet method {method}")
        return await fn(conn, p or {})

    # --------------------------------------------------------------- workers
    def _spawn_worker(self, runtime_env: Optional[dict] = None) -> WorkerProc:
        worker_id = WorkerID.from_random().binary()
        log_dir = os.path.join(self.session_dir, "logs")
        os.makedirs(log_dir, exist_ok=True)
        out = open(os.path.join(log_dir, f"worker-{worker_id.hex()[:8]}.log"), "wb")
        env = dict(os.environ)
        env["ANTRAY_WORKER_ID"] = worker_id.hex()
        env["ANTRAY_GCS"] = f"{self.gcs_addr[0]}:{self.gcs_addr[1]}"
        env["ANTRAY_RAYLET"] = f"{self.node_ip}:{self.port}"
        env["ANTRAY_NODE_ID"] = self.node_id.hex()
        env["ANTRAY_STORE"] = self.store_path
        env["ANTRAY_SESSION_DIR"] = self.session_dir
        cmd = [sys.executable, "-m",
               "ant_ray_amd._private.workers.default_worker"]
        cwd = None
        if runtime_env:
            # runtime-env plugins applied at spawn (parity with the
            # runtime-env agent's worker-command mutation,
            # _private/runtime_env/: env_vars, working_dir, py_modules,
            # rocprof_sys/nsight profiler prefixes)
            from ant_ray_amd._private.runtime_env import build_worker_spawn

            cmd, env, cwd = build_worker_spawn(cmd, env, runtime_env)
        proc = subprocess.Popen(
            cmd,
            env=env,
            cwd=cwd,
            stdout=out,
            stderr=subprocess.STDOUT,
            start_new_session=True,
        )
        out.close()
        w = WorkerProc(worker_id, proc)
        w.dedicated = bool(runtime_env)
        self.workers[worker_id] = w
        return w

    async def rpc_register_worker(self, conn, p):
        w = self.workers.get(p["worker_id"])
        if w is None:
            raise ValueError("unknown worker")
        w.conn = conn
        w.addr = tuple(p["addr"])
        conn.session["worker"] = w
        conn.on_close = lambda c: asyncio.get_running_loop().create_task(
            self._on_worker_conn_closed(w)
        )
        w.registered.set()
        return {"ok": True}

    async def _on_worker_conn_closed(self, w: WorkerProc):
        if w.proc.poll() is None:
            # connection lost but process alive; give it a moment then kill
            await asyncio.sleep(1.0)
            if w.proc.poll() is None:
                self._kill_worker(w)
        await self._on_worker_dead(w, "connection closed")

    def _kill_worker(self, w: WorkerProc):
        if w in self.idle_workers:
            self.idle_workers.remove(w)
        try:
            os.killpg(w.proc.pid, signal.SIGKILL)
        except Exception:
            try:
                w.proc.kill()
            except Exception:
                pass

    # ---------------------------------------------------------------- leases
    def _fits(self, avail: Dict[str, float], res: Dict[str, float]) -> bool:
        return all(avail.get(k, 0.0) + 1e-9 >= v for k, v in res.items())

    def _feasible(self, res: Dict[str, float]) -> bool:
        return all(self.resources_total.get(k, 0.0) + 1e-9 >= v for k, v in res.items())

    def _take(self, res: Dict[str, float], pg: Optional[dict]):
        pool = (
            self.bundles.get((pg["pg_id"], pg.get("bundle_index", 0)))
            if pg
            else self.resources_available
        )
        for k, v in res.items():
            pool[k] = pool.get(k, 0.0) - v
        gpu_ids: List[int] = []
        need = res.get("GPU", 0.0)
        if need > 0 and not pg:
            remaining = need
            for gid in sorted(self.gpu_pool, key=lambda g: -self.gpu_pool[g]):
                if remaining <= 1e-9:
                    break
                grab = min(1.0, remaining, self.gpu_pool[gid])
                if grab > 1e-9 and self.gpu_pool[gid] >= min(1.0, remaining) - 1e-9:
                    self.gpu_pool[gid] -= grab
                    remaining -= grab
                    gpu_ids.append(gid)
        elif need > 0 and pg:
            # GPU ids for PG bundles come from the general pool too
            remaining = need
            for gid in sorted(self.gpu_pool, key=lambda g: -self.gpu_pool[g]):
                if remaining <= 1e-9:
                    break
                grab = min(1.0, remaining, self.gpu_pool[gid])
                if grab > 1e-9:
                    self.gpu_pool[gid] -= grab
                    remaining -= grab
                    gpu_ids.append(gid)
        return gpu_ids

    def _give_back(self, res: Dict[str, float], pg: Optional[dict], gpu_ids: List[int]):
        pool = (
            self.bundles.get((pg["pg_id"], pg.get("bundle_index", 0)))
            if pg
            else self.resources_available
        )
        if pool is None:
            # bundle already returned (PG removed while the lease was live):
            # return_bundle only gave back what the bundle had free at that
            # moment, so the lease's share goes straight to the node pool
            pool = self.resources_available
        for k, v in res.items():
            pool[k] = pool.get(k, 0.0) + v
        need = res.get("GPU", 0.0)
        if need > 0:
            remaining = need
            for gid in gpu_ids:
                back = min(1.0, remaining)
                self.gpu_pool[gid] = min(1.0, self.gpu_pool[gid] + back)
                remaining -= back

    def _reclaim_leases_of(self, owner_conn):
        if owner_conn is self.gcs_conn:
            # the GCS connection dropping is a GCS restart, not a lessee
            # death: actor workers it leased (incl. detached actors) keep
            # running; the restarted GCS reconciles from its persisted
            # actor table
            return
        for w2 in list(self.workers.values()):
            if w2.leased and w2.lease_owner is owner_conn:
                logger.info("reclaiming lease of dead lessee: worker %s",
                            w2.worker_id.hex()[:8])
                self._release_resources(w2)
                if w2.is_actor or w2.gpu_ids or getattr(w2, "dedicated", False):
                    self._kill_worker(w2)
                elif w2 not in self.idle_workers:
                    w2.idle_since = time.monotonic()
                    self.idle_workers.append(w2)
        self._pump_leases()

    def _release_resources(self, w: WorkerProc):
        if not w.leased:
            return
        self._give_back(w.lease_resources, getattr(w, "lease_pg", None), w.gpu_ids)
        w.leased = False
        w.lease_resources = {}
        w.gpu_ids = []
        w.lease_pg = None

    async def rpc_lease_worker(self, conn, p):
        """Grant a worker lease; queue if resources busy; spillback hint if
        infeasible on this node."""
        logger.info("lease request: %s", p.get("resources"))
        if getattr(self, "_draining", False):
            return {"granted": False, "reason": "node draining"}
        res = dict(p.get("resources") or {})
        pg = p.get("pg")
        selector = p.get("_label_selector")
        if selector and not _labels_match(selector.get("hard") or {}, self.labels):
            reply = {"granted": False, "infeasible": True}
            try:
                r = await self.gcs_conn.call(
                    "pick_raylet",
                    {"resources": res, "_label_selector": selector}, timeout=5)
                if r.get("addr"):
                    reply["spillback"] = r["addr"]
            except Exception:
                pass
            return reply
        if not pg and not self._feasible(res):
            # remember the demand so the autoscaler sees it (reference
            # reports infeasible resource load through the syncer)
            now = time.monotonic()
            self._infeasible = [(t, r) for t, r in
                                getattr(self, "_infeasible", [])
                                if now - t < 10.0][-63:] + [(now, dict(res))]
            reply = {"granted": False, "infeasible": True}
            try:
                # ask the GCS for a node whose TOTAL resources fit -> the
                # client re-leases there (spillback parity)
                r = await self.gcs_conn.call(
                    "pick_raylet",
                    {"resources": res,
                     "_label_selector": p.get("_label_selector")}, timeout=5)
                if r.get("addr"):
                    reply["spillback"] = r["addr"]
            except Exception:
                pass
            return reply
        # make sure a dying lessee's leases come back even when the lessee
        # is a remote driver (spillback) that never registered as a worker
        if not getattr(conn, "_lease_reclaim_hooked", False):
            conn._lease_reclaim_hooked = True
            prev = conn.on_close

            def _reclaim(c, prev=prev):
                if prev:
                    prev(c)
                self._reclaim_leases_of(c)

            conn.on_close = _reclaim
        fut = asyncio.get_running_loop().create_future()
        self.pending_leases.append(((p, conn), fut))
        self._pump_leases()
        if p.get("no_wait") and any(x[1] is fut for x in self.pending_leases):
            # GCS actor scheduling: the pump left us queued (resources
            # unavailable NOW) — deny instead of holding the RPC so the
            # scheduler can try another (possibly empty) node right away;
            # blocking here serializes actor placement behind whichever
            # node was picked first (parity: reference raylets reply with
            # spillback rather than holding the lease). A request the pump
            # already picked up (async worker spawn) is NOT denied.
            self.pending_leases = [x for x in self.pending_leases
                                   if x[1] is not fut]
            fut.cancel()
            return {"granted": False, "reason": "resources unavailable now"}
        return await fut

    def _pump_leases(self):
        if not self.pending_leases:
            return
        granted_any = True
        while granted_any and self.pending_leases:
            granted_any = False
            for item in list(self.pending_leases):
                (p, requester), fut = item
                if fut.done():
                    self.pending_leases.remove(item)
                    continue
                res = dict(p.get("resources") or {})
                pg = p.get("pg")
                pool = (
                    self.bundles.get((pg["pg_id"], pg.get("bundle_index", 0)))
                    if pg
                    else self.resources_available
                )
                if pool is None:
                    continue  # bundle not reserved yet
                if not self._fits(pool, {k: v for k, v in res.items() if k != "GPU"} if pg else res):
                    continue
                if res.get("GPU", 0) > 0:
                    free_gpu = sum(self.gpu_pool.values())
                    if free_gpu + 1e-9 < res["GPU"]:
                        continue
                t = asyncio.get_running_loop().create_task(self._grant_lease(item))
                t.add_done_callback(
                    lambda t: t.exception() and logger.error(
                        "grant_lease failed: %r", t.exception()))
                self.pending_leases.remove(item)
                granted_any = True

    async def _grant_lease(self, item):
        (p, requester), fut = item
        res = dict(p.get("resources") or {})
        pg = p.get("pg")
        gpu_ids = self._take(res, pg)
        renv = p.get("runtime_env") or {}
        needs_dedicated = any(k != "env_vars" for k in renv)
        w = None
        while w is None:
            if self.idle_workers and not needs_dedicated:
                w = self.idle_workers.pop()
                if w.proc.poll() is not None:
                    self.workers.pop(w.worker_id, None)
                    w = None
                    continue
            else:
                # Popen + log-file open cost ~5-15ms of syscalls: off the
                # event loop, or a 200-wide actor burst serializes its
                # spawns behind the loop and starves the registration
                # RPCs those same spawns are waiting on
                w = await asyncio.get_running_loop().run_in_executor(
                    None, self._spawn_worker,
                    renv if needs_dedicated else None)
            try:
                await asyncio.wait_for(w.registered.wait(), timeout=60)
            except asyncio.TimeoutError:
                self._kill_worker(w)
                self.workers.pop(w.worker_id, None)
                w = None
        w.leased = True
        w.lease_owner = requester
        w.lease_resources = res
        w.lease_pg = pg
        w.gpu_ids = gpu_ids
        w.is_actor = p.get("actor_id") is not None
        w.detached_actor = bool(p.get("detached"))
        # tell the worker its lease context (GPU visibility, job env)
        try:
            await w.conn.call(
                "set_lease",
                {
                    "gpu_ids": gpu_ids,
                    "resources": res,
                    "actor_id": p.get("actor_id"),
                    "runtime_env": p.get("runtime_env"),
                },
                timeout=30,
            )
        except Exception as e:
            logger.warning("set_lease failed: %s", e)
            self._release_resources(w)
            self._kill_worker(w)
            if not fut.done():
                fut.set_result({"granted": False, "error": str(e)})
            return
        if not fut.done():
            fut.set_result(
                {
                    "granted": True,
                    "worker_id": w.worker_id,
                    "addr": list(w.addr),
                    "gpu_ids": gpu_ids,
                    "node_id": self.node_id,
                }
            )

    async def rpc_return_worker(self, conn, p):
        w = self.workers.get(p["worker_id"])
        if w is None:
            return {"ok": False}
        self._release_resources(w)
        kill = (p.get("kill", False) or w.is_actor or w.gpu_ids
                or getattr(w, "dedicated", False))
        if kill or w.proc.poll() is not None:
            self._kill_worker(w)
        else:
            w.idle_since = time.monotonic()
            w.leased = False
            self.idle_workers.append(w)
        self._pump_leases()
        return {"ok": True}

    async def rpc_prestart_workers(self, conn, p):
        n = int(p.get("n", 1))
        for _ in range(n):
            if len(self.workers) < int(self.resources_total.get("CPU", 1)) + 2:
                w = self._spawn_worker()

                async def _pool(w=w):
                    try:
                        await asyncio.wait_for(w.registered.wait(), timeout=60)
                        if not w.leased:
                            self.idle_workers.append(w)
                            self._pump_leases()
                    except asyncio.TimeoutError:
                        pass

                asyncio.get_running_loop().create_task(_pool())
        return {"ok": True}

    # ---------------------------------------------------------------- bundles
    async def rpc_reserve_bundle(self, conn, p):
        res = dict(p["resources"])
        # CPU/mem come out of the node pool now; GPU ids are taken at lease time
        non_gpu = {k: v for k, v in res.items() if k != "GPU"}
        if not self._fits(self.resources_available, non_gpu):
            return {"ok": False}
        gpu_need = res.get("GPU", 0.0)
        if gpu_need > 0 and sum(self.gpu_pool.values()) + 1e-9 < gpu_need:
            return {"ok": False}
        for k, v in non_gpu.items():
            self.resources_available[k] = self.resources_available.get(k, 0.0) - v
        self.bundles[(p["pg_id"], p["bundle_index"])] = dict(res)
        self._pump_leases()
        return {"ok": True}

    async def rpc_return_bundle(self, conn, p):
        b = self.bundles.pop((p["pg_id"], p["bundle_index"]), None)
        if b is not None:
            for k, v in b.items():
                if k != "GPU":
                    self.resources_available[k] = self.resources_available.get(k, 0.0) + v
        return {"ok": True}

    # ------------------------------------------------- client-mode data plane
    # parity: the Ray Client server's object plane (util/client/server) —
    # remote drivers have no shm mapping, so the raylet stores/serves bytes
    async def rpc_store_put(self, conn, p):
        oid, data, meta = p["oid"], p["data"], p.get("meta", b"py")
        try:
            off = self._store.create_object(oid, len(data), meta)
            mv = self._store.view_at(off, len(data), True)
            mv[:] = data
            del mv
            self._store.seal(oid)
            self._store.release(oid)
        except ValueError:
            pass  # already exists
        return {"ok": True}

    async def rpc_pull_object(self, conn, p):
        buf, meta = self._store.get_buffer(p["oid"], 0.0)
        if buf is None:
            return {"data": None}
        from ant_ray_amd._private.object_store import PULL_CHUNK_BYTES

        mv = memoryview(buf)
        n = len(mv)
        off = p.get("offset")
        if off is not None:
            end = min(n, off + p.get("length", PULL_CHUNK_BYTES))
            return {"data": bytes(mv[off:end])}
        if n <= PULL_CHUNK_BYTES:
            return {"data": bytes(mv), "meta": bytes(meta)}
        return {"data": bytes(mv[:PULL_CHUNK_BYTES]),
                "meta": bytes(meta), "size": n}

    # ------------------------------------------------------------------ misc
    async def rpc_node_info(self, conn, p):
        return {
            "node_id": self.node_id,
            "addr": [self.node_ip, self.port],
            "resources_total": self.resources_total,
            "resources_available": self.resources_available,
            "store_path": self.store_path,
            "num_workers": len(self.workers),
        }

    async def rpc_store_stats(self, conn, p):
        """Object-store usage for `ray memory` / state API (parity:
        reference raylet GetNodeStats -> memory summary)."""
        try:
            stats = dict(self._store.stats())
        except Exception:
            stats = {}
        stats["node_id"] = self.node_id
        stats["spilled_files"] = len(getattr(self, "_spilled", {}) or {})
        return stats

    async def rpc_drain(self, conn, p):
        """Graceful drain (parity: reference DrainNode / autoscaler
        graceful termination): stop accepting leases, deny the queue,
        wait for leased workers to finish (bounded), then exit."""
        self._draining = True
        deadline = time.monotonic() + float(p.get("timeout_s", 60.0))

        async def _drain():
            # deny queued leases so owners re-lease elsewhere
            for (pl, _c), fut in list(self.pending_leases):
                if not fut.done():
                    fut.set_result({"granted": False,
                                    "reason": "node draining"})
            self.pending_leases.clear()
            while time.monotonic() < deadline:
                busy = [w for w in self.workers.values()
                        if w.leased and w.proc.poll() is None]
                if not busy:
                    break
                await asyncio.sleep(0.5)
            await self._do_shutdown()

        asyncio.get_running_loop().create_task(_drain())
        return {"ok": True, "draining": True}

    async def rpc_shutdown(self, conn, p):
        await self._do_shutdown()
        return {"ok": True}

    async def _do_shutdown(self):
        self._shutdown.set()
        for w in list(self.workers.values()):
            self._kill_worker(w)
        try:
            if os.path.exists(self.store_path):
                os.unlink(self.store_path)
        except Exception:
            pass
        asyncio.get_running_loop().call_later(0.1, lambda: os._exit(0))


async def run_raylet(args):
    resources = json.loads(args.resources) if args.resources else {}
    ncpu = args.num_cpus if args.num_cpus is not None else os.cpu_count()
    ngpu = args.num_gpus if args.num_gpus is not None else detect_num_gpus()
    resources.setdefault("CPU", float(ncpu))
    resources.setdefault("GPU", float(ngpu))
    resources.setdefault("memory", float(os.sysconf("SC_PHYS_PAGES") * os.sysconf("SC_PAGE_SIZE")))
    resources.setdefault("object_store_memory", float(args.store_capacity))
    resources.setdefault(f"node:{args.node_ip}", 1.0)
    host, port = args.gcs.split(":")
    raylet = Raylet(
        args.node_ip, (host, int(port)), resources, args.store_path,
        args.store_capacity, args.session_dir,
    )
    for src in (os.environ.get("RAY_NODE_LABELS", ""), args.labels):
        if src:
            raylet.labels.update(json.loads(src))
    await raylet.start(args.port)
    if args.announce_fd:
        os.write(args.announce_fd,
                 f"{raylet.port} {raylet.node_id.hex()}\n".encode())
        os.close(args.announce_fd)
    await raylet._shutdown.wait()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gcs", required=True)
    ap.add_argument("--node-ip", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=0)
    ap.add_argument("--num-cpus", type=int, default=None)
    ap.add_argument("--num-gpus", type=int, default=None)
    ap.add_argument("--resources", default="")
    ap.add_argument("--labels", default="")
    ap.add_argument("--store-path", required=True)
    ap.add_argument("--store-capacity", type=int, default=2 * 1024**3)
    ap.add_argument("--session-dir", required=True)
    ap.add_argument("--announce-fd", type=int, default=0)
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)
    asyncio.run(run_raylet(args))


if __name__ == "__main__":
    main()
