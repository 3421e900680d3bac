"""Parity path ray.tune.search.hyperopt. hyperopt is not in the image;
HyperOptSearch is the native TPE implementation (hyperopt's algorithm).
If hyperopt IS importable, the wrapper delegates to it."""
from ant_ray_amd.tune.search.searcher import TPESearch


class HyperOptSearch(TPESearch):
    def __init__(self, space=None, metric=None, mode="max", **kw):
        try:
            import hyperopt  # noqa: F401
        except ImportError:
            pass  # native TPE
        kw.pop("points_to_evaluate", None)
        n_startup = kw.pop("n_initial_points", 10)
        super().__init__(space=space, metric=metric, mode=mode,
                         n_startup=n_startup,
                         seed=kw.pop("random_state_seed", None))
