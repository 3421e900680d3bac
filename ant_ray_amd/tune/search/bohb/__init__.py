"""Parity path ray.tune.search.bohb (TuneBOHB): model-based suggestions
(TPE stands in for BOHB's KDE model — the same Parzen-estimator family)
meant to pair with the HyperBandForBOHB scheduler."""
from ant_ray_amd.tune.search.searcher import TPESearch


class TuneBOHB(TPESearch):
    pass
