"""ant_ray_amd.workflow — stub, matching the reference exactly.

The reference fork GUTTED ray.workflow (python/ray/workflow/__init__.py is
4 lines; the upstream feature was removed). Kept as an importable stub for
the same reason: old imports fail loudly with a pointer, not an
ImportError at the package level.
"""

def __getattr__(name):
    raise AttributeError(
        "ray.workflow was removed (the reference ships it gutted); use "
        "plain tasks/actors or ray.dag instead"
    )
