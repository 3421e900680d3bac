"""ant_ray_amd.train — Ray Train parity (v2-style architecture).

Role parity: reference python/ray/train/ (~49.5k LoC).
Public surface: ray.train.report / get_context / get_checkpoint /
get_dataset_shard, Checkpoint, ScalingConfig/RunConfig/FailureConfig/
CheckpointConfig, Result, and ray.train.torch.TorchTrainer.
"""
from ant_ray_amd.train._checkpoint import Checkpoint
from ant_ray_amd.train.base_trainer import (
    DataParallelTrainer,
    TrainingFailedError,
)
from ant_ray_amd.train.config import (
    BackendConfig,
    CheckpointConfig,
    DataConfig,
    FailureConfig,
    Result,
    RunConfig,
    ScalingConfig,
    SyncConfig,
    TorchConfig,
)

TRAIN_DATASET_KEY = "train"
from ant_ray_amd.train.session import (
    TrainContext,
    get_checkpoint,
    get_context,
    get_dataset_shard,
    report,
)

__all__ = [
    "BackendConfig", "DataConfig", "SyncConfig", "TRAIN_DATASET_KEY",
    "Checkpoint", "CheckpointConfig", "DataParallelTrainer", "FailureConfig",
    "Result", "RunConfig", "ScalingConfig", "TorchConfig", "TrainContext",
    "TrainingFailedError", "get_checkpoint", "get_context",
    "get_dataset_shard", "report", "torch",
]

from ant_ray_amd.train import torch  # noqa: E402  (submodule, like ray.train.torch)
