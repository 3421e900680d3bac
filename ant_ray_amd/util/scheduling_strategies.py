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


class _LabelMatchExpression:
    """One label predicate (parity: reference scheduling_strategies
    label-selector helpers). The GCS label selector consumes its
    dict form."""

    def __init__(self, key: str, operator: str, values=None):
        self.key = key
        self.operator = operator
        self.values = list(values or [])

    def to_dict(self):
        return {"key": self.key, "op": self.operator,
                "values": self.values}


class In(_LabelMatchExpression):
    def __init__(self, *values):
        super().__init__("", "in", values)


class NotIn(_LabelMatchExpression):
    def __init__(self, *values):
        super().__init__("", "not_in", values)


class Exists(_LabelMatchExpression):
    def __init__(self):
        super().__init__("", "exists")


class DoesNotExist(_LabelMatchExpression):
    def __init__(self):
        super().__init__("", "does_not_exist")
