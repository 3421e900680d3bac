"""Per-worker training session: context, report(), get_checkpoint().

Role parity: reference python/ray/train/_internal/session.py (report path)
and python/ray/train/v2/_internal/execution/context.py (TrainContext with
rank/world info). `ray.train.report(metrics, checkpoint=)` persists the
checkpoint into run storage and enqueues the metrics for the controller.
"""
from __future__ import annotations

import os
import queue
import shutil
import threading
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from ant_ray_amd.train._checkpoint import Checkpoint

_ctx_lock = threading.Lock()
_train_context: Optional["TrainContext"] = None


@dataclass
class TrainContext:
    experiment_name: str
    experiment_path: str
    world_rank: int = 0
    local_rank: int = 0
    world_size: int = 1
    local_world_size: int = 1
    node_rank: int = 0
    restore_checkpoint: Optional[Checkpoint] = None
    dataset_shards: Dict[str, Any] = field(default_factory=dict)
    report_queue: "queue.Queue" = field(default_factory=queue.Queue)
    _report_seq: int = 0
    _latest_checkpoint: Optional[Checkpoint] = None

    # --------------------------------------------------- api (ray.train.*)

    def get_world_rank(self) -> int:
        return self.world_rank

    def get_local_rank(self) -> int:
        return self.local_rank

    def get_world_size(self) -> int:
        return self.world_size

    def get_local_world_size(self) -> int:
        return self.local_world_size

    def get_node_rank(self) -> int:
        return self.node_rank

    def get_experiment_name(self) -> str:
        return self.experiment_name

    def get_trial_name(self) -> str:
        return self.experiment_name

    def get_trial_id(self) -> str:
        return self.experiment_name

    def get_trial_dir(self) -> str:
        return self.experiment_path

    def get_metadata(self) -> Dict[str, Any]:
        return {}

    def get_storage(self):
        return self.experiment_path


def set_train_context(ctx: Optional[TrainContext]):
    global _train_context
    with _ctx_lock:
        _train_context = ctx


def get_context() -> TrainContext:
    """ray.train.get_context(). Outside a worker returns a default context
    (parity: reference returns a dummy context on the driver)."""
    with _ctx_lock:
        if _train_context is not None:
            return _train_context
    return TrainContext(experiment_name="default", experiment_path=os.getcwd())


def get_checkpoint() -> Optional[Checkpoint]:
    ctx = get_context()
    return ctx._latest_checkpoint or ctx.restore_checkpoint


def get_dataset_shard(dataset_name: str = "train"):
    ctx = get_context()
    return ctx.dataset_shards.get(dataset_name)


def report(metrics: Dict[str, Any], checkpoint: Optional[Checkpoint] = None,
           checkpoint_dir_name: Optional[str] = None) -> None:
    """Report metrics (+ optionally persist a checkpoint) to the controller.

    The checkpoint the user hands in (usually Checkpoint.from_directory of a
    local temp dir) is copied into run storage at
    `{experiment_path}/{checkpoint_dir_name or checkpoint_NNNNNN}`; all ranks
    reporting the same step merge into the same directory (single-node
    shared FS; rank-unique filenames recommended for sharded state).
    """
    ctx = get_context()
    seq = ctx._report_seq
    ctx._report_seq += 1
    persisted = None
    if checkpoint is not None:
        name = checkpoint_dir_name or f"checkpoint_{seq:06d}"
        target = os.path.join(ctx.experiment_path, name)
        os.makedirs(target, exist_ok=True)
        if os.path.abspath(checkpoint.path) != os.path.abspath(target):
            shutil.copytree(checkpoint.path, target, dirs_exist_ok=True)
        persisted = Checkpoint(target)
        ctx._latest_checkpoint = persisted
    stop_ev = getattr(ctx, "stop_requested", None)
    if stop_ev is not None and stop_ev.is_set():
        # tune cooperative stop: the Tuner decided this trial is done
        # (scheduler / stop criteria); unwind the trainable cleanly
        raise SystemExit(0)
    ctx.report_queue.put(
        {
            "seq": seq,
            "rank": ctx.world_rank,
            "metrics": dict(metrics),
            "checkpoint_path": persisted.path if persisted else None,
        }
    )
