"""Search space + basic variant generation.

Role parity: reference python/ray/tune/search/ (basic_variant.py grid/random
generation, sample.py Domain classes). Wrapper algorithms (optuna/hyperopt/
…) are out of scope offline; BasicVariantGenerator covers grid_search +
random sampling, the defaults the reference uses without extra installs.
"""
from __future__ import annotations

import itertools
import random
from typing import Any, Dict, List, Optional


class Domain:
    def sample(self, rng: random.Random):
        raise NotImplementedError


class Categorical(Domain):
    def __init__(self, categories):
        self.categories = list(categories)

    def sample(self, rng):
        return rng.choice(self.categories)


class Uniform(Domain):
    def __init__(self, low, high):
        self.low, self.high = low, high

    def sample(self, rng):
        return rng.uniform(self.low, self.high)


class LogUniform(Domain):
    def __init__(self, low, high):
        import math

        self.lo, self.hi = math.log(low), math.log(high)

    def sample(self, rng):
        import math

        return math.exp(rng.uniform(self.lo, self.hi))


class Randint(Domain):
    def __init__(self, low, high):
        self.low, self.high = low, high

    def sample(self, rng):
        return rng.randrange(self.low, self.high)


class GridSearch:
    def __init__(self, values):
        self.values = list(values)


def grid_search(values) -> GridSearch:
    return GridSearch(values)


def choice(categories) -> Categorical:
    return Categorical(categories)


def uniform(low, high) -> Uniform:
    return Uniform(low, high)


def loguniform(low, high) -> LogUniform:
    return LogUniform(low, high)


def randint(low, high) -> Randint:
    return Randint(low, high)


class BasicVariantGenerator:
    """Expands grid_search cross-products; samples Domains num_samples times.
    Parity: tune/search/basic_variant.py."""

    def __init__(self, param_space: Dict[str, Any], num_samples: int = 1,
                 seed: Optional[int] = None):
        self.param_space = param_space
        self.num_samples = num_samples
        self.rng = random.Random(seed)

    def variants(self) -> List[Dict[str, Any]]:
        grid_keys = [k for k, v in self.param_space.items()
                     if isinstance(v, GridSearch)]
        grids = [self.param_space[k].values for k in grid_keys]
        out = []
        for _ in range(self.num_samples):
            for combo in itertools.product(*grids) if grids else [()]:
                cfg = {}
                for k, v in self.param_space.items():
                    if isinstance(v, GridSearch):
                        cfg[k] = combo[grid_keys.index(k)]
                    elif isinstance(v, Domain):
                        cfg[k] = v.sample(self.rng)
                    else:
                        cfg[k] = v
                out.append(cfg)
        return out


class Normal(Domain):
    def __init__(self, mean, sd):
        self.mean, self.sd = mean, sd

    def sample(self, rng):
        return rng.gauss(self.mean, self.sd)


class LogRandint(Domain):
    def __init__(self, low, high):
        import math

        self.lo, self.hi = math.log(low), math.log(high)

    def sample(self, rng):
        import math

        return int(math.exp(rng.uniform(self.lo, self.hi)))


class Quantized(Domain):
    """Round another domain's samples to multiples of q (reference
    sample.py quantized spaces)."""

    def __init__(self, inner: Domain, q):
        self.inner, self.q = inner, q

    def sample(self, rng):
        v = round(self.inner.sample(rng) / self.q) * self.q
        if isinstance(self.inner, (Randint, LogRandint)):
            return int(v)
        return v


class Function(Domain):
    """tune.sample_from — draw from a user callable; the callable may take
    an optional spec argument (ignored here, passed as None)."""

    def __init__(self, fn):
        self.fn = fn

    def sample(self, rng):
        try:
            return self.fn(None)
        except TypeError:
            return self.fn()


def randn(mean: float = 0.0, sd: float = 1.0) -> Normal:
    return Normal(mean, sd)


def qrandn(mean, sd, q) -> Quantized:
    return Quantized(Normal(mean, sd), q)


def quniform(low, high, q) -> Quantized:
    return Quantized(Uniform(low, high), q)


def qloguniform(low, high, q) -> Quantized:
    return Quantized(LogUniform(low, high), q)


def lograndint(low, high) -> LogRandint:
    return LogRandint(low, high)


def qrandint(low, high, q) -> Quantized:
    return Quantized(Randint(low, high), q)


def qlograndint(low, high, q) -> Quantized:
    return Quantized(LogRandint(low, high), q)


def sample_from(fn) -> Function:
    return Function(fn)


# sequential searchers (suggest/observe) + meta-searchers
from ant_ray_amd.tune.search.searcher import (  # noqa: F401,E402
    BayesOptSearch,
    ConcurrencyLimiter,
    Repeater,
    Searcher,
    TPESearch,
)
