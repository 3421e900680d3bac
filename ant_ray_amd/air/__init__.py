"""ant_ray_amd.air — shared AIR config/session/result types.

Role parity: reference python/ray/air/ (~16k LoC: config.py ScalingConfig
:99 / RunConfig / FailureConfig / CheckpointConfig, session.py, result.py).
The canonical definitions live in ant_ray_amd.train.config; this package
re-exports them under the reference's import paths (from ray.air import
ScalingConfig, ...).
"""
from ant_ray_amd.train._checkpoint import Checkpoint  # noqa: F401
from ant_ray_amd.train.config import (  # noqa: F401
    CheckpointConfig,
    FailureConfig,
    Result,
    RunConfig,
    ScalingConfig,
)
from ant_ray_amd.train.session import (  # noqa: F401
    get_checkpoint,
    get_context,
    get_dataset_shard,
    report,
)


# typing/execution aliases (reference air/__init__.py: DataBatchType from
# air.data_batch_type, ResourceRequest/AcquiredResources from
# air.execution.resources.request)
from typing import Any, Dict, List, Union  # noqa: E402

DataBatchType = Union[Dict[str, Any], "object"]  # dict / pandas / arrow batch


class ResourceRequest:
    """A bundle list + strategy a trial/worker group asks the cluster
    for (parity: air/execution/resources/request.py)."""

    def __init__(self, bundles: List[Dict[str, float]],
                 strategy: str = "PACK", *_, **__):
        self.bundles = [dict(b) for b in bundles]
        self.strategy = strategy

    @property
    def required_resources(self) -> Dict[str, float]:
        out: Dict[str, float] = {}
        for b in self.bundles:
            for k, v in b.items():
                out[k] = out.get(k, 0.0) + v
        return out

    def __eq__(self, other):
        return (isinstance(other, ResourceRequest)
                and self.bundles == other.bundles
                and self.strategy == other.strategy)


class AcquiredResources:
    """Resources granted against a ResourceRequest (parity:
    air/execution/resources/request.py AcquiredResources)."""

    def __init__(self, resource_request: ResourceRequest):
        self.resource_request = resource_request


class session:
    """Legacy ray.air.session facade (air/session.py)."""

    report = staticmethod(report)
    get_checkpoint = staticmethod(get_checkpoint)
    get_dataset_shard = staticmethod(get_dataset_shard)

    @staticmethod
    def get_world_rank():
        return get_context().get_world_rank()

    @staticmethod
    def get_world_size():
        return get_context().get_world_size()

    @staticmethod
    def get_local_rank():
        return get_context().get_local_rank()
