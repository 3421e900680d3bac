"""Trainable / env registry + factory helpers.

Role parity: reference python/ray/tune/registry.py (register_trainable,
register_env) and tune/schedulers/__init__.py create_scheduler /
tune/search/__init__.py create_searcher.
"""
from typing import Any, Callable, Dict

_trainables: Dict[str, Any] = {}
_envs: Dict[str, Callable] = {}


def register_trainable(name: str, trainable):
    """Register a trainable under a string name usable in tune.run /
    Tuner(trainable="name")."""
    _trainables[name] = trainable


def register_env(name: str, env_creator: Callable):
    """Register an RLlib environment constructor under a string name."""
    _envs[name] = env_creator


def get_trainable_cls(name: str):
    if name not in _trainables:
        raise ValueError(f"unknown trainable {name!r}; "
                         f"registered: {sorted(_trainables)}")
    return _trainables[name]


def get_env_creator(name: str):
    if name not in _envs:
        raise ValueError(f"unknown env {name!r}; registered: {sorted(_envs)}")
    return _envs[name]


def create_scheduler(name: str, **kwargs):
    from ant_ray_amd.tune.schedulers import (ASHAScheduler, FIFOScheduler,
                                             HyperBandForBOHB,
                                             HyperBandScheduler,
                                             MedianStoppingRule, PB2,
                                             PopulationBasedTraining,
                                             ResourceChangingScheduler)

    table = {"fifo": FIFOScheduler, "asha": ASHAScheduler,
             "async_hyperband": ASHAScheduler,
             "hyperband": HyperBandScheduler,
             "hb_bohb": HyperBandForBOHB,
             "median_stopping_rule": MedianStoppingRule,
             "pbt": PopulationBasedTraining,
             "pb2": PB2,
             "resource_changing": ResourceChangingScheduler}
    if name not in table:
        raise ValueError(f"unknown scheduler {name!r}; "
                         f"available: {sorted(table)}")
    return table[name](**kwargs)


def create_searcher(name: str, **kwargs):
    """Factory (parity tune/search create_searcher). Sequential searchers
    return INSTANCES (the Tuner drives suggest/observe); the basic variant
    generator returns its class (Tuner constructs it with the space)."""
    from ant_ray_amd.tune.search import BasicVariantGenerator
    from ant_ray_amd.tune.search.bayesopt import BayesOptSearch
    from ant_ray_amd.tune.search.bohb import TuneBOHB
    from ant_ray_amd.tune.search.hyperopt import HyperOptSearch
    from ant_ray_amd.tune.search.optuna import OptunaSearch
    from ant_ray_amd.tune.search.searcher import TPESearch

    table = {"variant_generator": BasicVariantGenerator,
             "random": BasicVariantGenerator,
             "bayesopt": BayesOptSearch,
             "tpe": TPESearch,
             "hyperopt": HyperOptSearch,
             "optuna": OptunaSearch,
             "bohb": TuneBOHB}
    if name not in table:
        raise ValueError(
            f"unknown searcher {name!r}; available: {sorted(table)}")
    cls = table[name]
    if cls is BasicVariantGenerator:
        return cls
    return cls(**kwargs)
