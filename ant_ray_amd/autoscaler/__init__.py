"""Autoscaler (v2-style core): demand-driven node scaling.

Role parity: reference python/ray/autoscaler/v2/ (scheduler.py bin-packing
over instance types, instance_manager/) and _private/monitor.py (the
control loop). Cloud node providers are out of scope offline; the
LocalNodeProvider launches extra raylets on this host (the reference's
fake_multi_node provider plays the same role in its tests).

Components:
  * ResourceDemandScheduler.get_nodes_to_launch: bin-packs pending demands
    (from GCS: queued actors + unplaceable PGs) onto candidate node types.
  * StandardAutoscaler.update(): one reconcile tick — compute demand,
    launch/terminate via the provider, respect min/max workers.
"""
from __future__ import annotations

import logging
import math
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

logger = logging.getLogger("antray.autoscaler")


@dataclass
class NodeTypeConfig:
    name: str
    resources: Dict[str, float]
    min_workers: int = 0
    max_workers: int = 10


class ResourceDemandScheduler:
    """Bin-pack resource demands onto node types (parity:
    autoscaler/_private/resource_demand_scheduler.py — first-fit-
    decreasing with the reference's utilization scoring and GPU
    avoidance; strict-spread bundle sets force distinct nodes)."""

    def __init__(self, node_types: Dict[str, NodeTypeConfig],
                 max_workers: Optional[int] = None):
        self.node_types = node_types
        self.max_workers = max_workers

    @staticmethod
    def _fits(res: Dict[str, float], demand: Dict[str, float]) -> bool:
        return all(res.get(k, 0) >= v for k, v in demand.items())

    @staticmethod
    def _utilization_score(cfg_res: Dict[str, float],
                           demand: Dict[str, float]):
        """Reference _utilization_score: prefer the node type the demand
        uses most completely; GPU nodes are avoided for GPU-free demands
        (score rank: (not wasted_gpu, min_util, mean_util))."""
        if not ResourceDemandScheduler._fits(cfg_res, demand):
            return None
        utils = [v / cfg_res[k] for k, v in demand.items()
                 if cfg_res.get(k, 0) > 0]
        if not utils:
            return None
        wasted_gpu = cfg_res.get("GPU", 0) > 0 and demand.get("GPU", 0) == 0
        return (0 if wasted_gpu else 1, min(utils),
                sum(utils) / len(utils))

    def _pick_type(self, demand: Dict[str, float],
                   existing: Dict[str, int],
                   to_launch: Dict[str, int]) -> Optional[str]:
        best = None
        best_score = None
        total = sum(existing.values()) + sum(to_launch.values())
        for tname, cfg in self.node_types.items():
            cur = existing.get(tname, 0) + to_launch.get(tname, 0)
            if cur >= cfg.max_workers:
                continue
            if self.max_workers is not None and total >= self.max_workers:
                continue
            score = self._utilization_score(cfg.resources, demand)
            if score is None:
                continue
            if best_score is None or score > best_score:
                best, best_score = tname, score
        return best

    def get_nodes_to_launch(self, demands: List[Dict[str, float]],
                            existing: Dict[str, int],
                            strict_spread: Optional[List[List[Dict[str, float]]]] = None,
                            ) -> Dict[str, int]:
        """First-fit-decreasing over a copy of current free capacity;
        strict_spread bundle groups each get pairwise-distinct nodes."""
        to_launch: Dict[str, int] = {}
        free: List[Dict[str, float]] = []
        for tname, count in existing.items():
            cfg = self.node_types.get(tname)
            if cfg:
                free.extend(dict(cfg.resources) for _ in range(count))

        def place(demand, excluded: set) -> Optional[int]:
            for i, slot in enumerate(free):
                if i in excluded:
                    continue
                if self._fits(slot, demand):
                    for k, v in demand.items():
                        slot[k] -= v
                    return i
            t = self._pick_type(demand, existing, to_launch)
            if t is None:
                return None
            to_launch[t] = to_launch.get(t, 0) + 1
            slot = dict(self.node_types[t].resources)
            for k, v in demand.items():
                slot[k] -= v
            free.append(slot)
            return len(free) - 1

        # strict-spread groups first (hardest constraints)
        for group in strict_spread or []:
            used: set = set()
            for bundle in group:
                i = place(bundle, used)
                if i is not None:
                    used.add(i)
        pending = sorted(demands, key=lambda d: -sum(d.values()))
        for demand in pending:
            place(demand, set())
        return to_launch


# --------------------------------------------------------------- v2 IM
# Parity: reference autoscaler/v2/instance_manager/ — instances move
# QUEUED -> REQUESTED -> ALLOCATED -> RAY_RUNNING -> (RAY_STOPPING) ->
# TERMINATED; reconcile retries stuck requests and garbage-collects
# failures.
IM_QUEUED = "QUEUED"
IM_REQUESTED = "REQUESTED"
IM_ALLOCATED = "ALLOCATED"
IM_RAY_RUNNING = "RAY_RUNNING"
IM_TERMINATED = "TERMINATED"


@dataclass
class Instance:
    instance_id: str
    node_type: str
    status: str = IM_QUEUED
    created_at: float = field(default_factory=time.time)
    updated_at: float = field(default_factory=time.time)
    node_handle: Any = None


class InstanceManager:
    """Tracks instance lifecycles over a node provider (parity:
    autoscaler/v2/instance_manager/instance_manager.py, reduced)."""

    def __init__(self, provider, node_types: Dict[str, NodeTypeConfig],
                 request_timeout_s: float = 120.0):
        self.provider = provider
        self.node_types = node_types
        self.request_timeout_s = request_timeout_s
        self.instances: Dict[str, Instance] = {}
        self._counter = 0

    def queue(self, node_type: str) -> Instance:
        self._counter += 1
        inst = Instance(f"inst-{self._counter}", node_type)
        self.instances[inst.instance_id] = inst
        return inst

    def _set(self, inst: Instance, status: str):
        inst.status = status
        inst.updated_at = time.time()

    def reconcile(self):
        """One pass: launch queued instances, time out stuck requests."""
        for inst in list(self.instances.values()):
            if inst.status == IM_QUEUED:
                self._set(inst, IM_REQUESTED)
                try:
                    inst.node_handle = self.provider.create_node(
                        self.node_types[inst.node_type])
                    self._set(inst, IM_ALLOCATED)
                    # the local provider's nodes join immediately
                    self._set(inst, IM_RAY_RUNNING)
                except Exception:
                    logger.exception("instance %s launch failed",
                                     inst.instance_id)
                    self._set(inst, IM_TERMINATED)
            elif inst.status == IM_REQUESTED and (
                    time.time() - inst.updated_at > self.request_timeout_s):
                self._set(inst, IM_TERMINATED)

    def running(self) -> Dict[str, int]:
        out: Dict[str, int] = {}
        for inst in self.instances.values():
            if inst.status == IM_RAY_RUNNING:
                out[inst.node_type] = out.get(inst.node_type, 0) + 1
        return out

    def terminate_one(self, node_type: str):
        for inst in reversed(list(self.instances.values())):
            if inst.node_type == node_type and inst.status == IM_RAY_RUNNING:
                try:
                    self.provider.terminate_node(node_type)
                except Exception:
                    pass
                self._set(inst, IM_TERMINATED)
                return True
        return False


class LocalNodeProvider:
    """Launches/terminates extra raylets on this host (parity:
    fake_multi_node/node_provider.py)."""

    def __init__(self, cluster):
        self.cluster = cluster  # ant_ray_amd.cluster_utils.Cluster
        self.nodes: Dict[str, List[Any]] = {}

    def create_node(self, node_type: NodeTypeConfig):
        n = self.cluster.add_node(
            num_cpus=int(node_type.resources.get("CPU", 1)),
            num_gpus=int(node_type.resources.get("GPU", 0)),
            resources={k: v for k, v in node_type.resources.items()
                       if k not in ("CPU", "GPU", "memory")},
        )
        self.nodes.setdefault(node_type.name, []).append(n)
        return n

    def terminate_node(self, node_type: str):
        lst = self.nodes.get(node_type) or []
        if lst:
            self.cluster.remove_node(lst.pop(), allow_graceful=True)

    def non_terminated_nodes(self) -> Dict[str, int]:
        return {t: len(v) for t, v in self.nodes.items()}


class StandardAutoscaler:
    """One-tick reconciler (parity: autoscaler/_private/autoscaler.py +
    v2 instance manager, driven by monitor.py's loop)."""

    def __init__(self, node_types: Dict[str, NodeTypeConfig], provider,
                 idle_timeout_s: float = 60.0):
        self.node_types = node_types
        self.provider = provider
        self.scheduler = ResourceDemandScheduler(node_types)
        self.idle_timeout_s = idle_timeout_s
        self._idle_since: Dict[str, float] = {}

    def pending_demands(self) -> List[Dict[str, float]]:
        """Unschedulable actor/PG demands from the GCS."""
        from ant_ray_amd.util.state import _gcs_call

        demands: List[Dict[str, float]] = []
        for a in _gcs_call("list_actors"):
            if a.get("state") in ("PENDING_CREATION", "DEPENDENCIES_UNREADY"):
                demands.append(a.get("required_resources") or {"CPU": 1})
        self._strict_spread: List[List[Dict[str, float]]] = []
        for pg in _gcs_call("list_placement_groups"):
            if pg.get("state") == "PENDING":
                bundles = pg.get("bundles") or []
                if pg.get("strategy") in ("STRICT_SPREAD", "SPREAD"):
                    # bundle group needing pairwise-distinct nodes
                    self._strict_spread.append([dict(b) for b in bundles])
                else:
                    demands.extend(bundles)
        # queued TASK leases reported by raylets via heartbeat
        try:
            demands.extend(_gcs_call("pending_resource_demands"))
        except Exception:
            pass
        # standing requests from autoscaler.sdk.request_resources: only
        # the part NOT already satisfiable by free resources adds demand
        try:
            from ant_ray_amd.autoscaler.sdk import get_requested_resources

            import ant_ray_amd as ray

            avail = dict(ray.available_resources())
            for b in get_requested_resources():
                if all(avail.get(k, 0) >= v for k, v in b.items()):
                    for k, v in b.items():
                        avail[k] = avail.get(k, 0) - v
                else:
                    demands.append(dict(b))
        except Exception:
            pass
        return demands

    def update(self):
        demands = self.pending_demands()
        existing = self.provider.non_terminated_nodes()
        # honor min_workers
        for t, cfg in self.node_types.items():
            while existing.get(t, 0) < cfg.min_workers:
                self.provider.create_node(cfg)
                existing[t] = existing.get(t, 0) + 1
        to_launch = self.scheduler.get_nodes_to_launch(
            demands, existing,
            strict_spread=getattr(self, "_strict_spread", None))
        for t, n in to_launch.items():
            for _ in range(n):
                logger.info("autoscaler: launching node type %s", t)
                self.provider.create_node(self.node_types[t])
        # idle downscale: after idle_timeout_s with no pending demand AND
        # fully-unused worker capacity, terminate one launched node per
        # tick (LIFO via the provider) until back at min_workers
        now = time.time()
        if demands or to_launch:
            self._no_demand_since = None
        else:
            if getattr(self, "_no_demand_since", None) is None:
                self._no_demand_since = now
            idle_for = now - self._no_demand_since
            if idle_for >= self.idle_timeout_s:
                import ant_ray_amd as ray

                idle_nodes = 0
                for node in ray.nodes():
                    if not node.get("Alive"):
                        continue
                    used = {
                        k: node["Resources"].get(k, 0)
                        - node["Available"].get(k, 0)
                        for k in node["Resources"]
                    }
                    busy = any(v > 1e-9 for k, v in used.items()
                               if k not in ("memory", "object_store_memory")
                               and not k.startswith("node:"))
                    if not busy:
                        idle_nodes += 1
                if idle_nodes:
                    for t, cfg in self.node_types.items():
                        if existing.get(t, 0) > cfg.min_workers:
                            logger.info(
                                "autoscaler: terminating one idle %s node", t)
                            self.provider.terminate_node(t)
                            break
        return to_launch
