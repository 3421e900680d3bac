"""Train controller: drives worker groups, failure policy, checkpoints.

Role parity: reference python/ray/train/v2/_internal/execution/controller/
controller.py:101 (TrainController control loop :505-527 — poll worker
group, apply failure/scaling decisions) and checkpoint/checkpoint_manager.py
(retention by CheckpointConfig). Runs in the driver process here (the
reference pins it to the driver node as an actor; same failure domain —
driver death ends the run either way on one node).
"""
from __future__ import annotations

import logging
import os
import shutil
import time
from dataclasses import replace
from typing import Any, Callable, Dict, List, Optional

from ant_ray_amd.train._checkpoint import Checkpoint
from ant_ray_amd.train.config import (
    Result,
    RunConfig,
    ScalingConfig,
    TorchConfig,
)
from ant_ray_amd.train.worker_group import WorkerGroup

logger = logging.getLogger("antray.train")

_RESIZE = object()  # poll sentinel: regrow the group (not a failure)


class CheckpointManager:
    """Tracks reported checkpoints; enforces num_to_keep retention."""

    def __init__(self, checkpoint_config):
        self.cfg = checkpoint_config
        self.checkpoints: List[tuple] = []  # (path, metrics)
        self.latest: Optional[str] = None

    def register(self, path: str, metrics: Dict[str, Any]):
        if self.latest == path:
            # same checkpoint dir reported by another rank: merge metrics
            for i, (p, m) in enumerate(self.checkpoints):
                if p == path:
                    m.update(metrics)
            return
        self.latest = path
        self.checkpoints.append((path, dict(metrics)))
        self._enforce_retention()

    def _enforce_retention(self):
        keep = self.cfg.num_to_keep
        if keep is None or len(self.checkpoints) <= keep:
            return
        attr = self.cfg.checkpoint_score_attribute
        if attr:
            order = self.cfg.checkpoint_score_order == "max"
            ranked = sorted(
                self.checkpoints,
                key=lambda t: (t[1].get(attr) is not None,
                               t[1].get(attr) if t[1].get(attr) is not None else 0),
                reverse=order,
            )
            doomed = [c for c in ranked[keep:] if c[0] != self.latest]
        else:
            doomed = [c for c in self.checkpoints[:-keep]]
        for path, _ in doomed:
            self.checkpoints.remove((path, _))
            try:
                shutil.rmtree(path, ignore_errors=True)
            except Exception:
                pass

    def best_checkpoints(self):
        return [(Checkpoint(p), m) for p, m in self.checkpoints]


class TrainController:
    def __init__(
        self,
        train_fn: Callable,
        train_loop_config: Optional[dict],
        scaling_config: ScalingConfig,
        run_config: RunConfig,
        torch_config: Optional[TorchConfig] = None,
        datasets: Optional[Dict[str, Any]] = None,
        resume_from_checkpoint: Optional[Checkpoint] = None,
        dataset_config=None,
    ):
        self.train_fn = train_fn
        self.train_loop_config = train_loop_config
        self.scaling = scaling_config
        self.run_config = run_config
        self.torch_config = torch_config or TorchConfig()
        self.datasets = datasets or {}
        self.dataset_config = dataset_config
        self.name = run_config.name or f"train_{int(time.time())}"
        self.experiment_path = os.path.join(run_config.storage_path, self.name)
        os.makedirs(self.experiment_path, exist_ok=True)
        self.ckpt_manager = CheckpointManager(run_config.checkpoint_config)
        self.resume_from_checkpoint = resume_from_checkpoint
        self.latest_metrics: Dict[int, Dict[str, Any]] = {}
        self.callbacks = list(run_config.callbacks or [])

    def _fire(self, hook: str, *a, **kw):
        """Invoke a user callback hook if defined (parity: reference
        train/v2 UserCallback — on_report / on_checkpoint /
        on_worker_group_start / on_worker_group_shutdown /
        on_failure)."""
        for cb in self.callbacks:
            fn = getattr(cb, hook, None)
            if fn is None:
                continue
            try:
                fn(*a, **kw)
            except Exception:
                logger.exception("user callback %s.%s failed",
                                 type(cb).__name__, hook)

    def _split_datasets(self, n: int):
        """Per-worker dataset shards: ant_ray_amd.data datasets stream-split;
        plain iterables are handed to every worker whole."""
        if not self.datasets:
            return None
        shards = [dict() for _ in range(n)]
        for name, ds in self.datasets.items():
            splits = None
            dc = self.dataset_config
            split_this = dc._should_split(name) if dc is not None else True
            if split_this and hasattr(ds, "streaming_split"):
                try:
                    splits = ds.streaming_split(n, equal=True)
                except Exception:
                    splits = None
            for i in range(n):
                shards[i][name] = splits[i] if splits else ds
        return shards

    def _fit_in_available(self) -> int:
        """How many additional workers fit in currently-free resources."""
        try:
            import ant_ray_amd as ray

            per = self.scaling._resources_per_worker_not_none
            avail = ray.available_resources()
            return min(int(avail.get(k, 0.0) / v)
                       for k, v in per.items() if v > 0)
        except Exception:
            return 0

    def _elastic_size(self) -> int:
        """Pick the attempt's world size: num_workers when resources allow,
        else the largest feasible size >= min_workers (elastic training —
        each attempt restarts from the latest checkpoint, so shrinking or
        regrowing between attempts is safe)."""
        sc = self.scaling
        if not sc.min_workers or sc.min_workers >= sc.num_workers:
            return sc.num_workers
        try:
            import ant_ray_amd as ray

            per = sc._resources_per_worker_not_none
            avail = ray.available_resources()
            n_fit = min(int(avail.get(k, 0.0) / v)
                        for k, v in per.items() if v > 0)
        except Exception:
            return sc.num_workers
        return max(sc.min_workers, min(sc.num_workers, n_fit))

    def run(self) -> Result:
        max_failures = self.run_config.failure_config.max_failures
        attempt = 0
        error: Optional[BaseException] = None
        while True:
            n = self._elastic_size()
            scaling = (self.scaling if n == self.scaling.num_workers
                       else replace(self.scaling, num_workers=n))
            if n != self.scaling.num_workers:
                logger.warning("elastic: running with %d/%d workers",
                               n, self.scaling.num_workers)
            group = WorkerGroup(scaling, self.torch_config, self.name,
                                self.experiment_path)
            try:
                group.start()
                self._fire("on_worker_group_start", num_workers=n)
                restore = None
                if self.ckpt_manager.latest:
                    restore = self.ckpt_manager.latest
                elif self.resume_from_checkpoint:
                    restore = self.resume_from_checkpoint.path
                # keep shard iterators referenced for the whole attempt: they
                # hold the SplitCoordinator actor handle alive
                self._dataset_shards = self._split_datasets(n)
                group.start_training(
                    self.train_fn, self.train_loop_config, restore,
                    self._dataset_shards,
                )
                failed = self._poll_until_done(group, n)
            except Exception as e:  # actor/scheduling level failure
                logger.exception("worker group failed")
                failed = e
            finally:
                group.shutdown()
                self._fire("on_worker_group_shutdown")
            if failed is None:
                error = None
                break
            if failed is _RESIZE:
                # scaling decision, not a failure (parity: reference
                # controller applies ScalingPolicy resize decisions by
                # restarting the group from the latest checkpoint,
                # train/v2/.../controller.py:180-191) — don't burn an
                # attempt
                logger.warning("elastic: resources recovered, regrowing "
                               "worker group from checkpoint")
                continue
            error = failed if isinstance(failed, BaseException) else RuntimeError(failed)
            attempt += 1
            if max_failures >= 0 and attempt > max_failures:
                break
            logger.warning("restarting worker group (attempt %d/%d): %s",
                           attempt, max_failures, error)
        latest_ckpt = (Checkpoint(self.ckpt_manager.latest)
                       if self.ckpt_manager.latest else None)
        return Result(
            metrics=self.latest_metrics.get(0),
            checkpoint=latest_ckpt,
            path=self.experiment_path,
            error=error,
            best_checkpoints=self.ckpt_manager.best_checkpoints(),
        )

    def _poll_until_done(self, group: WorkerGroup, current_n: int):
        """Returns None on clean finish, error string on worker failure, or
        _RESIZE when a shrunk group can regrow (upscale decision)."""
        last_scale_check = time.monotonic()
        while True:
            statuses = group.poll()
            for st in statuses:
                for rep in st["reports"]:
                    self.latest_metrics[rep["rank"]] = rep["metrics"]
                    self._fire("on_report", metrics=rep["metrics"],
                               rank=rep["rank"])
                    if rep["checkpoint_path"]:
                        self.ckpt_manager.register(rep["checkpoint_path"],
                                                   rep["metrics"])
                        self._fire("on_checkpoint",
                                   checkpoint_path=rep["checkpoint_path"],
                                   metrics=rep["metrics"], rank=rep["rank"])
            errs = [st["error"] for st in statuses if st["status"] == "errored"]
            if errs:
                self._fire("on_failure", error=errs[0])
                # drain the other ranks' queued reports before tearing the
                # group down: a checkpoint reported just before a peer's
                # failure must be registered, or the restart resumes from
                # an older (or no) checkpoint
                time.sleep(0.3)
                try:
                    for st in group.poll():
                        for rep in st["reports"]:
                            self.latest_metrics[rep["rank"]] = rep["metrics"]
                            if rep["checkpoint_path"]:
                                self.ckpt_manager.register(
                                    rep["checkpoint_path"], rep["metrics"])
                except Exception:
                    pass
                return errs[0]
            if all(st["status"] == "finished" for st in statuses):
                return None
            # upscale decision: a group running below num_workers regrows
            # once resources free up AND a checkpoint exists to resume from
            # (without one, restarting would lose more progress than the
            # extra workers regain)
            if (current_n < self.scaling.num_workers
                    and self.ckpt_manager.latest
                    and time.monotonic() - last_scale_check > 2.0):
                last_scale_check = time.monotonic()
                # the running group holds current_n workers' resources;
                # after teardown those return, so the regrown size is
                # current_n + whatever fits in what's free NOW
                if min(self.scaling.num_workers,
                       current_n + self._fit_in_available()) > current_n:
                    return _RESIZE
            time.sleep(0.2)
