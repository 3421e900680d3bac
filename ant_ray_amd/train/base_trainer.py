"""DataParallelTrainer — SPMD function trainer over a worker group.

Role parity: reference python/ray/train/v2/api/data_parallel_trainer.py:67
(DataParallelTrainer.fit :155 spawns the controller and waits). Subclasses
(TorchTrainer) pick the backend config.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, Optional, Union

import ant_ray_amd as ray
from ant_ray_amd.train._checkpoint import Checkpoint
from ant_ray_amd.train.config import (
    Result,
    RunConfig,
    ScalingConfig,
    TorchConfig,
)
from ant_ray_amd.train.controller import TrainController


class DataParallelTrainer:
    def __init__(
        self,
        train_loop_per_worker: Union[Callable[[], None], Callable[[dict], None]],
        *,
        train_loop_config: Optional[Dict[str, Any]] = None,
        scaling_config: Optional[ScalingConfig] = None,
        run_config: Optional[RunConfig] = None,
        datasets: Optional[Dict[str, Any]] = None,
        dataset_config: Optional[Any] = None,
        resume_from_checkpoint: Optional[Checkpoint] = None,
        metadata: Optional[Dict[str, Any]] = None,
        backend_config: Optional[Any] = None,
    ):
        self.train_loop_per_worker = train_loop_per_worker
        self.train_loop_config = train_loop_config
        self.scaling_config = scaling_config or ScalingConfig()
        self.run_config = run_config or RunConfig()
        self.datasets = datasets
        self.dataset_config = dataset_config
        self.resume_from_checkpoint = resume_from_checkpoint
        self.metadata = metadata
        self.backend_config = backend_config

    def _torch_config(self) -> TorchConfig:
        if isinstance(self.backend_config, TorchConfig):
            return self.backend_config
        return TorchConfig()

    def fit(self) -> Result:
        if not ray.is_initialized():
            ray.init()
        controller = TrainController(
            train_fn=self.train_loop_per_worker,
            train_loop_config=self.train_loop_config,
            scaling_config=self.scaling_config,
            run_config=self.run_config,
            torch_config=self._torch_config(),
            datasets=self.datasets,
            resume_from_checkpoint=self.resume_from_checkpoint,
            dataset_config=self.dataset_config,
        )
        result = controller.run()
        if result.error is not None:
            raise TrainingFailedError(
                f"training failed after retries: {result.error}"
            ) from result.error
        return result


class TrainingFailedError(RuntimeError):
    pass
