"""ExperimentAnalysis — load a finished (or interrupted) experiment dir.

Role parity: reference python/ray/tune/analysis/experiment_analysis.py:
best_config / best_checkpoint / dataframe over the trial rows persisted
by the Tuner (`tuner_state.pkl` in the experiment directory).
"""
import os
import pickle
from typing import Optional


class ExperimentAnalysis:
    def __init__(self, experiment_checkpoint_path: str,
                 default_metric: Optional[str] = None,
                 default_mode: Optional[str] = None):
        path = experiment_checkpoint_path
        if os.path.isdir(path):
            path = os.path.join(path, "tuner_state.pkl")
        with open(path, "rb") as f:
            self._state = pickle.load(f)
        self._dir = os.path.dirname(path)
        self.default_metric = default_metric
        self.default_mode = default_mode

    @property
    def trials(self):
        return self._state.get("trials", [])

    def _scored(self, metric):
        return [t for t in self.trials
                if (t.get("metrics") or {}).get(metric) is not None]

    def _best_trial(self, metric=None, mode=None):
        metric = metric or self.default_metric
        mode = mode or self.default_mode or "max"
        rows = self._scored(metric)
        if not rows:
            raise ValueError(f"no trial reported metric {metric!r}")
        return (max if mode == "max" else min)(
            rows, key=lambda t: t["metrics"][metric])

    def get_best_config(self, metric=None, mode=None):
        return self._best_trial(metric, mode)["config"]

    @property
    def best_config(self):
        return self.get_best_config()

    def get_best_checkpoint(self, trial=None, metric=None, mode=None):
        from ant_ray_amd.train._checkpoint import Checkpoint

        t = trial if isinstance(trial, dict) else self._best_trial(metric, mode)
        return Checkpoint(t["ckpt"]) if t.get("ckpt") else None

    @property
    def best_checkpoint(self):
        return self.get_best_checkpoint()

    def dataframe(self, metric=None, mode=None):
        import pandas as pd

        return pd.DataFrame([dict(t.get("metrics") or {},
                                  trial_idx=t["idx"],
                                  status=t.get("status"))
                             for t in self.trials])
