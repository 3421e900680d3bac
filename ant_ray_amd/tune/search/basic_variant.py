"""Parity path ray.tune.search.basic_variant."""
from ant_ray_amd.tune.search import BasicVariantGenerator  # noqa: F401
