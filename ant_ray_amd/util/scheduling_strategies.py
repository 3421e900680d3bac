"""Scheduling strategies. Parity: python/ray/util/scheduling_strategies.py
(wire types reference src/ray/protobuf/common.proto:83-126)."""
from __future__ import annotations

from typing import Dict, Optional


class PlacementGroupSchedulingStrategy:
    def __init__(
        self,
        placement_group,
        placement_group_bundle_index: int = -1,
        placement_group_capture_child_tasks: Optional[bool] = None,
    ):
        self.placement_group = placement_group
        self.placement_group_bundle_index = placement_group_bundle_index
        self.placement_group_capture_child_tasks = placement_group_capture_child_tasks


class NodeAffinitySchedulingStrategy:
    def __init__(self, node_id: str, soft: bool = False,
                 _spill_on_unavailable: bool = False, _fail_on_unavailable: bool = False):
        self.node_id = node_id
        self.soft = soft


class NodeLabelSchedulingStrategy:
    def __init__(self, hard: Optional[Dict] = None, soft: Optional[Dict] = None):
        self.hard = hard or {}
        self.soft = soft or {}


DEFAULT_SCHEDULING_STRATEGY = "DEFAULT"
SPREAD_SCHEDULING_STRATEGY = "SPREAD"
