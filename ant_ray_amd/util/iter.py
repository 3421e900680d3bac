"""ray.util.iter — legacy parallel iterators (deprecated upstream).

Role parity: reference python/ray/util/iter.py (ParallelIterator). The
reference deprecated this API in favor of Ray Data; here from_items /
from_range are provided as thin adapters over ant_ray_amd.data so old
call sites keep working, and everything else should use ray.data
directly.
"""
import warnings


def _warn():
    warnings.warn(
        "ray.util.iter is deprecated; use ray.data instead",
        DeprecationWarning, stacklevel=3)


def from_items(items, num_shards: int = 2, repeat: bool = False):
    _warn()
    import ant_ray_amd.data as data

    return data.from_items(list(items), override_num_blocks=num_shards)


def from_range(n: int, num_shards: int = 2, repeat: bool = False):
    _warn()
    import ant_ray_amd.data as data

    return data.range(n, override_num_blocks=num_shards)
