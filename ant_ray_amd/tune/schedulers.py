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
PERTURB = "PERTURB"


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


class PopulationBasedTraining:
    """PBT (reference tune/schedulers/pbt.py): at every
    perturbation_interval, trials in the bottom quantile EXPLOIT a top-
    quantile trial (clone its latest checkpoint + config) and EXPLORE by
    mutating hyperparameters (resample from the mutation space with
    probability resample_probability, else multiply by 1.2 / 0.8).

    Protocol with the Tuner: on_trial_result may return PERTURB; the
    Tuner then calls exploit(trial_id) -> (restore_path, new_config) and
    relaunches the trial. The Tuner feeds checkpoints and configs in via
    on_checkpoint()/on_trial_start().
    """

    def __init__(self, metric: Optional[str] = None, mode: Optional[str] = None,
                 perturbation_interval: int = 4,
                 hyperparam_mutations: Optional[Dict] = None,
                 quantile_fraction: float = 0.25,
                 resample_probability: float = 0.25,
                 time_attr: str = "training_iteration", seed: Optional[int] = None):
        import random

        self.metric = metric
        self.mode = mode
        self.interval = perturbation_interval
        self.mutations = hyperparam_mutations or {}
        self.quantile = quantile_fraction
        self.resample_p = resample_probability
        self.time_attr = time_attr
        self._rng = random.Random(seed)
        self._score: Dict[str, float] = {}
        self._ckpt: Dict[str, str] = {}
        self._config: Dict[str, dict] = {}
        self._last_perturb: Dict[str, int] = defaultdict(int)

    def set_objective(self, metric, mode):
        self.metric = self.metric or metric
        self.mode = self.mode or mode

    # -- Tuner feed-in hooks
    def on_trial_start(self, trial_id: str, config: dict):
        self._config[trial_id] = dict(config)

    def on_checkpoint(self, trial_id: str, path: str):
        self._ckpt[trial_id] = path

    # -- decisions
    def on_trial_result(self, trial_id: str, result: dict) -> str:
        v = result.get(self.metric)
        t = result.get(self.time_attr)
        if v is None or t is None:
            return CONTINUE
        self._score[trial_id] = float(v)
        if t - self._last_perturb[trial_id] < self.interval:
            return CONTINUE
        self._last_perturb[trial_id] = t
        lower, upper = self._quantiles()
        if trial_id in lower and any(u in self._ckpt for u in upper):
            return PERTURB
        return CONTINUE

    def _quantiles(self):
        trials = [tid for tid in self._score]
        if len(trials) < 2:
            return [], []
        trials.sort(key=lambda tid: self._score[tid],
                    reverse=(self.mode != "max"))  # worst first
        n = max(1, int(len(trials) * self.quantile))
        if n > len(trials) // 2:
            n = len(trials) // 2
        return trials[:n], trials[-n:]

    def exploit(self, trial_id: str):
        """Returns (restore_checkpoint_path, mutated_config) cloning a
        random top-quantile trial, or None if none has a checkpoint."""
        _, upper = self._quantiles()
        donors = [u for u in upper if u in self._ckpt and u != trial_id]
        if not donors:
            return None
        donor = self._rng.choice(donors)
        cfg = dict(self._config.get(donor, {}))
        for key, space in self.mutations.items():
            if self._rng.random() < self.resample_p or key not in cfg:
                cfg[key] = self._sample(space)
            else:
                cfg[key] = cfg[key] * self._rng.choice([0.8, 1.2])
        self._config[trial_id] = dict(cfg)
        return self._ckpt[donor], cfg

    def _sample(self, space):
        if callable(space):
            return space()
        if isinstance(space, (list, tuple)):
            return self._rng.choice(list(space))
        if hasattr(space, "sample"):
            return space.sample(self._rng)
        return space


class HyperBandScheduler:
    """Classic (synchronous-bracket) HyperBand (reference tune/schedulers/
    hyperband.py), adapted to the streaming on_trial_result protocol:
    trials are assigned round-robin to brackets with different
    (grace, reduction) trade-offs; inside a bracket, successive-halving
    rungs cut the worst performers at each milestone."""

    def __init__(self, metric: Optional[str] = None,
                 mode: Optional[str] = None, max_t: int = 81,
                 reduction_factor: int = 3,
                 time_attr: str = "training_iteration"):
        import math

        self.metric, self.mode = metric, mode
        self.max_t = max_t
        self.rf = reduction_factor
        self.time_attr = time_attr
        s_max = int(math.log(max_t, reduction_factor))
        # bracket k starts at grace rf^k
        self._brackets = [ASHAScheduler(metric=metric, mode=mode,
                                        max_t=max_t,
                                        grace_period=reduction_factor ** k,
                                        reduction_factor=reduction_factor,
                                        time_attr=time_attr)
                          for k in range(s_max + 1)]
        self._assign: Dict[str, int] = {}
        self._next = 0

    def set_objective(self, metric, mode):
        self.metric = self.metric or metric
        self.mode = self.mode or mode
        for b in self._brackets:
            b.set_objective(self.metric, self.mode)

    def on_trial_result(self, trial_id: str, result: dict) -> str:
        if trial_id not in self._assign:
            self._assign[trial_id] = self._next % len(self._brackets)
            self._next += 1
        return self._brackets[self._assign[trial_id]].on_trial_result(
            trial_id, result)


class HyperBandForBOHB(HyperBandScheduler):
    """BOHB's bracket scheduler (reference tune/schedulers/hb_bohb.py) —
    pair with TPESearch (the model-based half of BOHB) via
    Tuner(search_alg=TPESearch(), scheduler=HyperBandForBOHB())."""


class PB2(PopulationBasedTraining):
    """PBT with GP-guided exploration (reference tune/schedulers/pb2.py):
    instead of random multiply/resample mutations, continuous
    hyperparameters of the exploited config are chosen by a GP-UCB fit
    on (config -> latest score) observations. hyperparam_bounds maps
    name -> (low, high)."""

    def __init__(self, metric: Optional[str] = None,
                 mode: Optional[str] = None, perturbation_interval: int = 4,
                 hyperparam_bounds: Optional[Dict] = None,
                 quantile_fraction: float = 0.25,
                 time_attr: str = "training_iteration",
                 seed: Optional[int] = None):
        super().__init__(metric=metric, mode=mode,
                         perturbation_interval=perturbation_interval,
                         hyperparam_mutations={},
                         quantile_fraction=quantile_fraction,
                         time_attr=time_attr, seed=seed)
        self.bounds = hyperparam_bounds or {}
        self._obs: List = []  # (param-vector, score)

    def on_trial_result(self, trial_id: str, result: dict) -> str:
        v = result.get(self.metric)
        if v is not None and trial_id in self._config:
            cfg = self._config[trial_id]
            vec = [float(cfg.get(k, (lo + hi) / 2))
                   for k, (lo, hi) in self.bounds.items()]
            if vec:
                self._obs.append((vec, float(v)
                                  * (1.0 if self.mode != "min" else -1.0)))
        return super().on_trial_result(trial_id, result)

    def exploit(self, trial_id: str):
        base = super().exploit(trial_id)
        if base is None or not self.bounds:
            return base
        path, cfg = base
        cfg = dict(cfg)
        new_vals = self._gp_ucb_suggest()
        for k, val in new_vals.items():
            cfg[k] = val
        self._config[trial_id] = dict(cfg)
        return path, cfg

    def _gp_ucb_suggest(self) -> Dict[str, float]:
        import numpy as np

        keys = list(self.bounds)
        lo = np.array([self.bounds[k][0] for k in keys])
        hi = np.array([self.bounds[k][1] for k in keys])
        rng = np.random.RandomState(self._rng.randrange(2 ** 31))
        cand = rng.uniform(lo, hi, size=(64, len(keys)))
        if len(self._obs) >= 3:
            try:
                from sklearn.gaussian_process import GaussianProcessRegressor
                from sklearn.gaussian_process.kernels import Matern

                X = np.array([o[0] for o in self._obs])
                y = np.array([o[1] for o in self._obs])
                y = (y - y.mean()) / (y.std() + 1e-9)
                gp = GaussianProcessRegressor(
                    kernel=Matern(nu=2.5), normalize_y=False,
                    random_state=0).fit(X, y)
                mu, sd = gp.predict(cand, return_std=True)
                best = cand[int(np.argmax(mu + 1.0 * sd))]
            except Exception:
                best = cand[0]
        else:
            best = cand[0]
        return {k: float(b) for k, b in zip(keys, best)}


class ResourceChangingScheduler:
    """Wrapper delegating trial decisions to a base scheduler while
    exposing a resources_allocation_function hook (reference
    tune/schedulers/resource_changing_scheduler.py; this build's trials
    are single-process so reallocation is advisory)."""

    def __init__(self, base_scheduler=None,
                 resources_allocation_function=None):
        self.base = base_scheduler or FIFOScheduler()
        self.alloc_fn = resources_allocation_function

    def set_objective(self, metric, mode):
        self.base.set_objective(metric, mode)

    def on_trial_result(self, trial_id: str, result: dict) -> str:
        return self.base.on_trial_result(trial_id, result)

    def __getattr__(self, item):
        return getattr(self.base, item)
