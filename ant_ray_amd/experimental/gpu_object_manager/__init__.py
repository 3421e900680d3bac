from ant_ray_amd.experimental.gpu_object_manager.gpu_object_store import (  # noqa: F401
    GPUObjectStore,
    gpu_object_store,
)
