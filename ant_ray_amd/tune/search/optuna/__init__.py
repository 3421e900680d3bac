"""Parity path ray.tune.search.optuna. optuna is not in the image;
OptunaSearch falls back to the native TPE implementation (optuna's
default sampler IS TPE)."""
from ant_ray_amd.tune.search.searcher import TPESearch


class OptunaSearch(TPESearch):
    def __init__(self, space=None, metric=None, mode="max", **kw):
        try:
            import optuna  # noqa: F401
        except ImportError:
            pass  # native TPE (optuna's default TPESampler algorithm)
        kw.pop("points_to_evaluate", None)
        kw.pop("sampler", None)
        super().__init__(space=space, metric=metric, mode=mode,
                         seed=kw.pop("seed", None))
