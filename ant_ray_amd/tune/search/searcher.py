"""Sequential searchers: suggest/observe interface + wrappers.

Role parity: reference python/ray/tune/search/ — Searcher base
(searcher.py), ConcurrencyLimiter/Repeater (search/__init__.py,
repeater.py), and the algorithm integrations (bayesopt/hyperopt/optuna/
...). The external libraries are not in this image, so the two core
model-based algorithms are implemented NATIVELY here:

  * BayesOptSearch — Gaussian-process expected improvement on
    scikit-learn's GaussianProcessRegressor (sklearn ships in the image),
    matching the role of the reference's bayes_opt wrapper.
  * TPESearch — tree-structured Parzen estimator (hyperopt's algorithm):
    split observations at the gamma-quantile, model good/bad densities
    with KDEs, suggest the candidate maximizing the density ratio.

The reference's thin wrappers (OptunaSearch, HyperOptSearch, ...) exist
as import-gated shims in their reference module paths
(tune/search/optuna.py etc.) delegating to these native algorithms when
the wrapped library is absent.
"""
from __future__ import annotations

import math
import random
from typing import Any, Dict, List, Optional

from ant_ray_amd.tune.search import (
    Categorical,
    Domain,
    GridSearch,
    LogRandint,
    LogUniform,
    Quantized,
    Randint,
    Uniform,
)


class Searcher:
    """suggest(trial_id) -> config | None (None = no more suggestions now);
    on_trial_complete(trial_id, result, error)."""

    def __init__(self, metric: Optional[str] = None, mode: str = "max"):
        self.metric = metric
        self.mode = mode

    def set_search_properties(self, metric, mode, config, **kw) -> bool:
        if metric:
            self.metric = metric
        if mode:
            self.mode = mode
        if config:
            self._set_space(config)
        return True

    def _set_space(self, config: Dict[str, Any]):
        self.space = config

    def suggest(self, trial_id: str) -> Optional[Dict[str, Any]]:
        raise NotImplementedError

    def on_trial_result(self, trial_id: str, result: Dict):
        pass

    def on_trial_complete(self, trial_id: str, result: Optional[Dict] = None,
                          error: bool = False):
        pass


class _SpaceCodec:
    """Encode a param_space of Domains into [0,1]^d vectors and back."""

    def __init__(self, space: Dict[str, Any], seed: Optional[int] = None):
        self.rng = random.Random(seed)
        self.keys: List[str] = []
        self.domains: List[Domain] = []
        self.fixed: Dict[str, Any] = {}
        for k, v in space.items():
            if isinstance(v, GridSearch):
                # model-based searchers treat a grid as categorical
                v = Categorical(v.values)
            if isinstance(v, Domain):
                self.keys.append(k)
                self.domains.append(v)
            else:
                self.fixed[k] = v
        self.dim = len(self.keys)

    def _dom_to_unit(self, d: Domain, val) -> float:
        if isinstance(d, Quantized):
            return self._dom_to_unit(d.inner, val)
        if isinstance(d, Categorical):
            try:
                i = d.categories.index(val)
            except ValueError:
                i = 0
            return (i + 0.5) / len(d.categories)
        if isinstance(d, Uniform):
            return (val - d.low) / (d.high - d.low)
        if isinstance(d, LogUniform):
            return (math.log(val) - d.lo) / (d.hi - d.lo)
        if isinstance(d, Randint):
            return (val - d.low + 0.5) / (d.high - d.low)
        if isinstance(d, LogRandint):
            return (math.log(max(val, 1e-12)) - d.lo) / (d.hi - d.lo)
        return 0.5

    def _unit_to_dom(self, d: Domain, u: float):
        u = min(max(u, 0.0), 1.0 - 1e-9)
        if isinstance(d, Quantized):
            v = self._unit_to_dom(d.inner, u)
            v = round(v / d.q) * d.q
            if isinstance(d.inner, (Randint, LogRandint)):
                return int(v)
            return v
        if isinstance(d, Categorical):
            return d.categories[int(u * len(d.categories))]
        if isinstance(d, Uniform):
            return d.low + u * (d.high - d.low)
        if isinstance(d, LogUniform):
            return math.exp(d.lo + u * (d.hi - d.lo))
        if isinstance(d, Randint):
            return d.low + int(u * (d.high - d.low))
        if isinstance(d, LogRandint):
            return int(math.exp(d.lo + u * (d.hi - d.lo)))
        return u

    def encode(self, cfg: Dict[str, Any]) -> List[float]:
        return [self._dom_to_unit(d, cfg[k])
                for k, d in zip(self.keys, self.domains)]

    def decode(self, x: List[float]) -> Dict[str, Any]:
        cfg = dict(self.fixed)
        for k, d, u in zip(self.keys, self.domains, x):
            cfg[k] = self._unit_to_dom(d, u)
        return cfg

    def random_unit(self) -> List[float]:
        return [self.rng.random() for _ in range(self.dim)]


class BayesOptSearch(Searcher):
    """Native GP-EI Bayesian optimization (sklearn GaussianProcessRegressor).

    Parity role: reference tune/search/bayesopt/bayesopt_search.py (which
    wraps the `bayesian-optimization` package — same GP-EI method)."""

    def __init__(self, space: Optional[Dict[str, Any]] = None,
                 metric: Optional[str] = None, mode: str = "max",
                 random_search_steps: int = 8, candidates: int = 512,
                 seed: Optional[int] = None):
        super().__init__(metric, mode)
        self.random_search_steps = random_search_steps
        self.candidates = candidates
        self.seed = seed
        self._x: List[List[float]] = []
        self._y: List[float] = []
        self._live: Dict[str, List[float]] = {}
        self.codec = None
        if space:
            self._set_space(space)

    def _set_space(self, config):
        self.space = config
        self.codec = _SpaceCodec(config, seed=self.seed)

    def suggest(self, trial_id):
        assert self.codec is not None, "search space not set"
        if len(self._x) < self.random_search_steps or self.codec.dim == 0:
            x = self.codec.random_unit()
        else:
            x = self._suggest_ei()
        self._live[trial_id] = x
        return self.codec.decode(x)

    def _suggest_ei(self):
        import numpy as np
        from sklearn.gaussian_process import GaussianProcessRegressor
        from sklearn.gaussian_process.kernels import Matern

        X = np.array(self._x)
        y = np.array(self._y, dtype=float)
        if self.mode == "max":
            y = -y  # internally minimize
        # normalize targets for GP stability
        mu, sd = y.mean(), y.std() or 1.0
        yn = (y - mu) / sd
        gp = GaussianProcessRegressor(
            kernel=Matern(nu=2.5), alpha=1e-6, normalize_y=False,
            random_state=self.seed)
        gp.fit(X, yn)
        rng = np.random.RandomState(self.codec.rng.randrange(2 ** 31))
        cand = rng.rand(self.candidates, self.codec.dim)
        m, s = gp.predict(cand, return_std=True)
        best = yn.min()
        s = np.maximum(s, 1e-9)
        z = (best - m) / s
        from scipy.stats import norm

        ei = s * (z * norm.cdf(z) + norm.pdf(z))
        return [float(v) for v in cand[int(ei.argmax())]]

    def on_trial_complete(self, trial_id, result=None, error=False):
        x = self._live.pop(trial_id, None)
        if x is None or error or not result:
            return
        val = result.get(self.metric)
        if val is None:
            return
        self._x.append(x)
        self._y.append(float(val))


class TPESearch(Searcher):
    """Native tree-structured Parzen estimator (hyperopt's algorithm).

    Parity role: reference tune/search/hyperopt/hyperopt_search.py."""

    def __init__(self, space: Optional[Dict[str, Any]] = None,
                 metric: Optional[str] = None, mode: str = "max",
                 n_startup: int = 10, gamma: float = 0.25,
                 candidates: int = 64, seed: Optional[int] = None):
        super().__init__(metric, mode)
        self.n_startup = n_startup
        self.gamma = gamma
        self.candidates = candidates
        self.seed = seed
        self._x: List[List[float]] = []
        self._y: List[float] = []
        self._live: Dict[str, List[float]] = {}
        self.codec = None
        if space:
            self._set_space(space)

    def _set_space(self, config):
        self.space = config
        self.codec = _SpaceCodec(config, seed=self.seed)

    def suggest(self, trial_id):
        assert self.codec is not None, "search space not set"
        if len(self._x) < self.n_startup or self.codec.dim == 0:
            x = self.codec.random_unit()
        else:
            x = self._suggest_tpe()
        self._live[trial_id] = x
        return self.codec.decode(x)

    def _suggest_tpe(self):
        import numpy as np

        X = np.array(self._x)
        y = np.array(self._y, dtype=float)
        if self.mode == "max":
            y = -y
        order = np.argsort(y)
        n_good = max(1, int(self.gamma * len(y)))
        good = X[order[:n_good]]
        bad = X[order[n_good:]] if len(y) > n_good else X

        bw = max(0.05, 1.0 / max(len(good), 1) ** 0.5 * 0.5)
        rng = np.random.RandomState(self.codec.rng.randrange(2 ** 31))
        # sample candidates from the good KDE (pick a good point, jitter)
        centers = good[rng.randint(len(good), size=self.candidates)]
        cand = np.clip(centers + rng.randn(self.candidates,
                                           self.codec.dim) * bw, 0, 1)

        def log_kde(pts, data):
            # isotropic Gaussian KDE in the unit cube
            d2 = ((pts[:, None, :] - data[None, :, :]) ** 2).sum(-1)
            return np.log(np.exp(-d2 / (2 * bw * bw)).mean(1) + 1e-12)

        score = log_kde(cand, good) - log_kde(cand, bad)
        return [float(v) for v in cand[int(score.argmax())]]

    on_trial_complete = BayesOptSearch.on_trial_complete


class ConcurrencyLimiter(Searcher):
    """Cap in-flight suggestions (parity: tune/search ConcurrencyLimiter)."""

    def __init__(self, searcher: Searcher, max_concurrent: int = 8,
                 batch: bool = False):
        super().__init__(searcher.metric, searcher.mode)
        self.searcher = searcher
        self.max_concurrent = max_concurrent
        self._live: set = set()

    def set_search_properties(self, metric, mode, config, **kw):
        ok = self.searcher.set_search_properties(metric, mode, config, **kw)
        self.metric, self.mode = self.searcher.metric, self.searcher.mode
        return ok

    def suggest(self, trial_id):
        if len(self._live) >= self.max_concurrent:
            return None
        cfg = self.searcher.suggest(trial_id)
        if cfg is not None:
            self._live.add(trial_id)
        return cfg

    def on_trial_complete(self, trial_id, result=None, error=False):
        self._live.discard(trial_id)
        self.searcher.on_trial_complete(trial_id, result, error)


class Repeater(Searcher):
    """Repeat each suggested config `repeat` times and report the MEAN
    metric to the wrapped searcher (parity: tune/search/repeater.py —
    de-noises stochastic objectives)."""

    def __init__(self, searcher: Searcher, repeat: int = 3,
                 set_index: bool = True):
        super().__init__(searcher.metric, searcher.mode)
        self.searcher = searcher
        self.repeat = repeat
        self.set_index = set_index
        self._groups: Dict[str, dict] = {}   # group lead id -> state
        self._of: Dict[str, str] = {}        # trial id -> group lead id
        self._open: Optional[dict] = None

    def set_search_properties(self, metric, mode, config, **kw):
        ok = self.searcher.set_search_properties(metric, mode, config, **kw)
        self.metric, self.mode = self.searcher.metric, self.searcher.mode
        return ok

    def suggest(self, trial_id):
        if self._open is None or self._open["n"] >= self.repeat:
            cfg = self.searcher.suggest(trial_id)
            if cfg is None:
                return None
            self._open = {"lead": trial_id, "cfg": cfg, "n": 0, "vals": [],
                          "done": 0}
            self._groups[trial_id] = self._open
        g = self._open
        g["n"] += 1
        self._of[trial_id] = g["lead"]
        cfg = dict(g["cfg"])
        if self.set_index:
            cfg["__trial_index__"] = g["n"] - 1
        if g["n"] >= self.repeat:
            self._open = None
        return cfg

    def on_trial_complete(self, trial_id, result=None, error=False):
        lead = self._of.pop(trial_id, None)
        if lead is None:
            return
        g = self._groups.get(lead)
        if g is None:
            return
        g["done"] += 1
        if result and not error:
            v = result.get(self.searcher.metric or self.metric)
            if v is not None:
                g["vals"].append(float(v))
        if g["done"] >= self.repeat:
            self._groups.pop(lead, None)
            if g["vals"]:
                mean = sum(g["vals"]) / len(g["vals"])
                self.searcher.on_trial_complete(
                    lead, {self.searcher.metric or self.metric: mean})
            else:
                self.searcher.on_trial_complete(lead, error=True)
