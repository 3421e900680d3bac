"""Parity path ray.tune.search.bayesopt. The reference wraps the
`bayesian-optimization` package; this image has none, so BayesOptSearch
IS the native sklearn GP-EI implementation (same algorithm family)."""
from ant_ray_amd.tune.search.searcher import BayesOptSearch  # noqa: F401
