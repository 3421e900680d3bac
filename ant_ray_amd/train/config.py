"""Train/AIR config dataclasses.

Role parity: reference python/ray/air/config.py (ScalingConfig :99,
RunConfig, FailureConfig, CheckpointConfig) and
python/ray/train/v2/api/config.py. Semantics kept: ScalingConfig drives the
worker group size/resources, FailureConfig.max_failures drives group
restarts, CheckpointConfig(num_to_keep, checkpoint_score_attribute) drives
retention in the checkpoint manager.
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Any, Dict, Optional


@dataclass
class ScalingConfig:
    num_workers: int = 1
    use_gpu: bool = False
    resources_per_worker: Optional[Dict[str, float]] = None
    trainer_resources: Optional[Dict[str, float]] = None
    placement_strategy: str = "PACK"
    accelerator_type: Optional[str] = None
    # elastic training (ant-fork parity): when set, a (re)started worker
    # group may run with fewer than num_workers — down to min_workers —
    # sized to the resources actually available, resuming from the latest
    # checkpoint; it grows back toward num_workers on later restarts.
    min_workers: Optional[int] = None

    @property
    def _resources_per_worker_not_none(self) -> Dict[str, float]:
        if self.resources_per_worker is not None:
            res = dict(self.resources_per_worker)
            if self.use_gpu and "GPU" not in res:
                res["GPU"] = 1
            return res
        return {"CPU": 1, "GPU": 1} if self.use_gpu else {"CPU": 1}

    def num_gpus_per_worker(self) -> float:
        return float(self._resources_per_worker_not_none.get("GPU", 0))


@dataclass
class FailureConfig:
    max_failures: int = 0


@dataclass
class CheckpointConfig:
    num_to_keep: Optional[int] = None
    checkpoint_score_attribute: Optional[str] = None
    checkpoint_score_order: str = "max"
    checkpoint_frequency: int = 0
    checkpoint_at_end: Optional[bool] = None


@dataclass
class RunConfig:
    name: Optional[str] = None
    storage_path: Optional[str] = None
    failure_config: FailureConfig = field(default_factory=FailureConfig)
    checkpoint_config: CheckpointConfig = field(default_factory=CheckpointConfig)
    verbose: int = 1
    log_to_file: bool = False
    callbacks: Optional[list] = None
    stop: Optional[Any] = None  # tune: dict {metric: threshold} / fn / Stopper
    progress_reporter: Optional[Any] = None  # tune: ProgressReporter

    def __post_init__(self):
        if self.storage_path is None:
            self.storage_path = os.path.expanduser(
                os.environ.get("ANTRAY_STORAGE_PATH", "~/ray_results")
            )


@dataclass
class TorchConfig:
    """Backend config for torch process-group setup.

    Parity: python/ray/train/torch/config.py:33 (TorchConfig: backend,
    init_method, timeout_s). backend=None → nccl (RCCL) when the worker has
    a GPU else gloo.
    """

    backend: Optional[str] = None
    init_method: str = "tcp"
    timeout_s: int = 1800

    def resolved_backend(self, use_gpu: bool) -> str:
        if self.backend:
            return self.backend
        return "nccl" if use_gpu else "gloo"


@dataclass
class Result:
    """Outcome of Trainer.fit(). Parity: python/ray/air/result.py:38."""

    metrics: Optional[Dict[str, Any]] = None
    checkpoint: Optional[Any] = None
    path: Optional[str] = None
    error: Optional[BaseException] = None
    metrics_dataframe: Optional[Any] = None
    best_checkpoints: Optional[list] = None

    def get_best_checkpoint(self, metric: str, mode: str = "max"):
        if not self.best_checkpoints:
            return None
        keyed = [
            (m.get(metric), c)
            for c, m in self.best_checkpoints
            if m and m.get(metric) is not None
        ]
        if not keyed:
            return None
        keyed.sort(key=lambda t: t[0], reverse=(mode == "max"))
        return keyed[0][1]


@dataclass
class SyncConfig:
    """Artifact/driver sync knobs (parity: reference train/_internal/
    syncer.py SyncConfig). Single shared-FS node class: syncing is a
    no-op, the fields exist so user configs carry through unchanged."""

    sync_period: int = 300
    sync_timeout: int = 1800
    sync_artifacts: bool = False
    sync_artifacts_on_checkpoint: bool = True


class BackendConfig:
    """Parent class for training-backend configurations (parity:
    reference train/backend.py:16). TorchConfig subclasses this; custom
    backends override backend_cls/train_func_context."""

    @property
    def backend_cls(self):
        return None

    @property
    def train_func_context(self):
        import contextlib

        return contextlib.nullcontext


class DataConfig:
    """Configures which datasets are split across train workers (parity:
    reference train/_internal/data_config.py:13). datasets_to_split="all"
    (default) shards every dataset via streaming split; a list limits
    sharding to those names — the rest are replicated whole to each
    worker."""

    def __init__(self, datasets_to_split="all", execution_options=None,
                 enable_shard_locality: bool = True):
        if not (datasets_to_split == "all"
                or isinstance(datasets_to_split, list)):
            raise TypeError(
                "`datasets_to_split` should be 'all' or a list of dataset "
                f"names, got {datasets_to_split!r}")
        self._datasets_to_split = datasets_to_split
        self._execution_options = execution_options
        self._enable_shard_locality = enable_shard_locality

    def _should_split(self, name: str) -> bool:
        return (self._datasets_to_split == "all"
                or name in self._datasets_to_split)

    @staticmethod
    def default_ingest_options():
        return None
