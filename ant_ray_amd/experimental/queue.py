"""ray.experimental.queue (parity alias: the maintained Queue is
ray.util.queue)."""
from ant_ray_amd.util.queue import Queue  # noqa: F401
from queue import Empty, Full  # noqa: F401
