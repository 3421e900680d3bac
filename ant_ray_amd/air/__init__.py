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
