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
    autoscaler/_private/resource_demand_scheduler.py)."""

    def __init__(self, node_types: Dict[str, NodeTypeConfig]):
        self.node_types = node_types

    def get_nodes_to_launch(self, demands: List[Dict[str, float]],
                            existing: Dict[str, int]) -> Dict[str, int]:
        """First-fit-decreasing over a copy of current free capacity."""
        to_launch: Dict[str, int] = {}
        free: List[Dict[str, float]] = []
        for tname, count in existing.items():
            cfg = self.node_types.get(tname)
            if cfg:
                free.extend(dict(cfg.resources) for _ in range(count))
        pending = sorted(demands, key=lambda d: -sum(d.values()))
        for demand in pending:
            placed = False
            for slot in free:
                if all(slot.get(k, 0) >= v for k, v in demand.items()):
                    for k, v in demand.items():
                        slot[k] -= v
                    placed = True
                    break
            if placed:
                continue
            # pick the cheapest node type that fits the demand
            for tname in sorted(self.node_types,
                                key=lambda t: sum(self.node_types[t].resources.values())):
                cfg = self.node_types[tname]
                if all(cfg.resources.get(k, 0) >= v for k, v in demand.items()):
                    cur = existing.get(tname, 0) + to_launch.get(tname, 0)
                    if cur >= cfg.max_workers:
                        continue
                    to_launch[tname] = to_launch.get(tname, 0) + 1
                    slot = dict(cfg.resources)
                    for k, v in demand.items():
                        slot[k] -= v
                    free.append(slot)
                    break
        return to_launch


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
        for pg in _gcs_call("list_placement_groups"):
            if pg.get("state") == "PENDING":
                demands.extend(pg.get("bundles") or [])
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
        to_launch = self.scheduler.get_nodes_to_launch(demands, existing)
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
