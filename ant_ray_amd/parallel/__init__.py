from ant_ray_amd.parallel.flat import (  # noqa: F401
    FlatAdamW,
    FlatDDP,
    FlatParamManager,
)
