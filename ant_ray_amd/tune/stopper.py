"""Stoppers.

Role parity: reference python/ray/tune/stopper/ (Stopper base,
MaximumIterationStopper, TrialPlateauStopper essentials). A Stopper is
callable per-result; stop_all() ends the whole experiment.
"""
from collections import defaultdict, deque
from typing import Any, Dict


class Stopper:
    def __call__(self, trial_id: str, result: Dict[str, Any]) -> bool:
        raise NotImplementedError

    def stop_all(self) -> bool:
        return False


class MaximumIterationStopper(Stopper):
    def __init__(self, max_iter: int):
        self._max_iter = max_iter

    def __call__(self, trial_id, result):
        return result.get("training_iteration", 0) >= self._max_iter


class TrialPlateauStopper(Stopper):
    """Stop a trial when `metric` stops moving more than `std` over
    `num_results` consecutive results."""

    def __init__(self, metric: str, std: float = 0.01, num_results: int = 4,
                 grace_period: int = 4, mode: str = "min"):
        self._metric = metric
        self._std = std
        self._num_results = num_results
        self._grace = grace_period
        self._window = defaultdict(lambda: deque(maxlen=num_results))
        self._count = defaultdict(int)

    def __call__(self, trial_id, result):
        v = result.get(self._metric)
        if v is None:
            return False
        self._count[trial_id] += 1
        w = self._window[trial_id]
        w.append(v)
        if self._count[trial_id] < self._grace or len(w) < self._num_results:
            return False
        mean = sum(w) / len(w)
        var = sum((x - mean) ** 2 for x in w) / len(w)
        return var ** 0.5 <= self._std


class CombinedStopper(Stopper):
    def __init__(self, *stoppers):
        self._stoppers = stoppers

    def __call__(self, trial_id, result):
        return any(s(trial_id, result) for s in self._stoppers)

    def stop_all(self):
        return any(s.stop_all() for s in self._stoppers)
