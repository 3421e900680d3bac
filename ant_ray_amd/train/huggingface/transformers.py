"""HuggingFace Transformers integration for Ray Train.

Role parity: reference python/ray/train/huggingface/transformers/
(_transformers_utils.py RayTrainReportCallback + prepare_trainer): the
callback forwards HF Trainer logs/checkpoints into ray.train.report so
the Train controller tracks metrics and checkpoint retention; the
trainer runs unmodified inside a TorchTrainer train_fn.
"""
from __future__ import annotations

import os
import tempfile


_cls_cache = None


class RayTrainReportCallback:
    """transformers.TrainerCallback that reports metrics (+ checkpoint
    directories, when the HF Trainer saves one) to Ray Train."""

    def __new__(cls, *a, **kw):
        # subclass TrainerCallback lazily so importing this module does
        # not require transformers; returning a foreign type means
        # __init__ must run explicitly
        global _cls_cache
        if _cls_cache is None:
            from transformers.trainer_callback import TrainerCallback

            _cls_cache = type("RayTrainReportCallback",
                              (_CallbackImpl, TrainerCallback), {})
        inst = object.__new__(_cls_cache)
        inst.__init__(*a, **kw)
        return inst


class _CallbackImpl:
    def __init__(self):
        self._latest_metrics = {}

    # transformers calls on_log for every logging step
    def on_log(self, args, state, control, logs=None, **kwargs):
        if logs:
            self._latest_metrics.update(logs)

    def on_save(self, args, state, control, **kwargs):
        from ant_ray_amd import train
        from ant_ray_amd.train import Checkpoint

        ckpt_dir = os.path.join(
            args.output_dir, f"checkpoint-{state.global_step}")
        metrics = dict(self._latest_metrics)
        metrics.setdefault("step", state.global_step)
        metrics["epoch"] = state.epoch
        if os.path.isdir(ckpt_dir):
            train.report(metrics, checkpoint=Checkpoint.from_directory(ckpt_dir))
        else:
            train.report(metrics)

    def on_epoch_end(self, args, state, control, **kwargs):
        from ant_ray_amd import train

        metrics = dict(self._latest_metrics)
        metrics["epoch"] = state.epoch
        metrics.setdefault("step", state.global_step)
        train.report(metrics)


def prepare_trainer(trainer):
    """Attach the Ray report callback (idempotent) — reference
    prepare_trainer validates/patches the HF Trainer for Ray data
    integration; random-access HF datasets pass through unchanged."""
    has = any(type(cb).__name__ == "RayTrainReportCallback"
              for cb in trainer.callback_handler.callbacks)
    if not has:
        trainer.add_callback(RayTrainReportCallback())
    return trainer
