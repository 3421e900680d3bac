"""Class-API trainables.

Role parity: reference python/ray/tune/trainable/trainable.py (Trainable:
setup/step/save_checkpoint/load_checkpoint/cleanup/reset_config) and the
function-adapter that lets the Tuner's function-trainable runner execute a
Trainable subclass: instantiate, loop step() reporting each result, stop
on `done=True` or the experiment's stop criteria.
"""
from __future__ import annotations

import os
import tempfile
from typing import Any, Dict, Optional


class Trainable:
    """Subclass and override setup()/step() (+ optionally
    save_checkpoint()/load_checkpoint())."""

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        self.config = config or {}
        self.training_iteration = 0
        self.setup(self.config)

    # -- overridable API -------------------------------------------------
    def setup(self, config: Dict[str, Any]):
        pass

    def step(self) -> Dict[str, Any]:
        raise NotImplementedError

    def save_checkpoint(self, checkpoint_dir: str) -> Optional[str]:
        return None

    def load_checkpoint(self, checkpoint_dir: str):
        pass

    def cleanup(self):
        pass

    def reset_config(self, new_config: Dict[str, Any]) -> bool:
        return False

    # -- driver-side helpers (used by the adapter) -----------------------
    def train(self) -> Dict[str, Any]:
        out = self.step() or {}
        self.training_iteration += 1
        out.setdefault("training_iteration", self.training_iteration)
        return out


def _wrap_trainable_cls(cls, stop=None):
    """Make a tune function-trainable out of a Trainable subclass. The
    returned closure runs inside the trial runner actor."""

    def _fn(config):
        from ant_ray_amd.train._checkpoint import Checkpoint
        from ant_ray_amd.train.session import get_checkpoint, report

        obj = cls(config)
        try:
            ckpt = get_checkpoint()
            if ckpt is not None:
                obj.load_checkpoint(ckpt.path)
            while True:
                result = obj.train()
                done = bool(result.get("done"))
                if not done and stop:
                    if callable(stop):
                        done = bool(stop("trial", result))
                    else:
                        done = any(result.get(k) is not None
                                   and result[k] >= v
                                   for k, v in stop.items())
                if done:
                    result["done"] = True
                    d = tempfile.mkdtemp(prefix="trainable_ckpt_")
                    saved = obj.save_checkpoint(d)
                    cdir = saved if isinstance(saved, str) else d
                    if os.listdir(cdir):
                        report(result, checkpoint=Checkpoint.from_directory(cdir))
                    else:
                        report(result)
                    return
                report(result)
        finally:
            obj.cleanup()

    _fn.__name__ = getattr(cls, "__name__", "trainable")
    return _fn
