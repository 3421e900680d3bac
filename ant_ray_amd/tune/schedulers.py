"""Trial schedulers: FIFO, ASHA, median stopping.

Role parity: reference python/ray/tune/schedulers/ (async_hyperband.py
ASHAScheduler — async successive halving with rungs at
grace_period * reduction_factor^k; median_stopping_rule.py). The scheduler
sees every reported result and answers CONTINUE or STOP.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict, List, Optional

CONTINUE = "CONTINUE"
STOP = "STOP"


class FIFOScheduler:
    def on_trial_result(self, trial_id: str, result: dict) -> str:
        return CONTINUE

    def set_objective(self, metric: str, mode: str):
        self.metric, self.mode = metric, mode


class ASHAScheduler:
    """Single-bracket asynchronous successive halving."""

    def __init__(self, metric: Optional[str] = None, mode: Optional[str] = None,
                 max_t: int = 100, grace_period: int = 1,
                 reduction_factor: float = 4, time_attr: str = "training_iteration"):
        self.metric = metric
        self.mode = mode
        self.max_t = max_t
        self.rf = reduction_factor
        self.time_attr = time_attr
        self.rungs: List[int] = []
        t = grace_period
        while t < max_t:
            self.rungs.append(int(t))
            t *= reduction_factor
        # rung milestone -> list of metric values recorded there
        self.rung_results: Dict[int, List[float]] = defaultdict(list)
        self._trial_rung: Dict[str, int] = {}

    def set_objective(self, metric, mode):
        self.metric = self.metric or metric
        self.mode = self.mode or mode

    def on_trial_result(self, trial_id: str, result: dict) -> str:
        t = result.get(self.time_attr)
        v = result.get(self.metric)
        if t is None or v is None:
            return CONTINUE
        if t >= self.max_t:
            return STOP
        next_rung_idx = self._trial_rung.get(trial_id, 0)
        if next_rung_idx >= len(self.rungs) or t < self.rungs[next_rung_idx]:
            return CONTINUE
        milestone = self.rungs[next_rung_idx]
        self._trial_rung[trial_id] = next_rung_idx + 1
        recorded = self.rung_results[milestone]
        recorded.append(float(v))
        if len(recorded) < self.rf:
            return CONTINUE  # not enough peers to compare yet
        ranked = sorted(recorded, reverse=(self.mode == "max"))
        cutoff = ranked[max(0, int(len(ranked) / self.rf) - 1)]
        good = v >= cutoff if self.mode == "max" else v <= cutoff
        return CONTINUE if good else STOP


class MedianStoppingRule:
    """Stop a trial whose running best is worse than the median of other
    trials' running means at the same step (tune/schedulers/
    median_stopping_rule.py)."""

    def __init__(self, metric: Optional[str] = None, mode: Optional[str] = None,
                 grace_period: int = 3, min_samples_required: int = 3,
                 time_attr: str = "training_iteration"):
        self.metric = metric
        self.mode = mode
        self.grace = grace_period
        self.min_samples = min_samples_required
        self.time_attr = time_attr
        self._history: Dict[str, List[float]] = defaultdict(list)

    def set_objective(self, metric, mode):
        self.metric = self.metric or metric
        self.mode = self.mode or mode

    def on_trial_result(self, trial_id: str, result: dict) -> str:
        v = result.get(self.metric)
        t = result.get(self.time_attr, 0)
        if v is None:
            return CONTINUE
        self._history[trial_id].append(float(v))
        if t < self.grace or len(self._history) < self.min_samples:
            return CONTINUE
        means = [sum(h) / len(h) for tid, h in self._history.items()
                 if tid != trial_id and h]
        if len(means) < self.min_samples - 1:
            return CONTINUE
        means.sort()
        median = means[len(means) // 2]
        best = (max if self.mode == "max" else min)(self._history[trial_id])
        if self.mode == "max":
            return CONTINUE if best >= median else STOP
        return CONTINUE if best <= median else STOP
