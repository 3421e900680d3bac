from ant_ray_amd.train.huggingface.transformers import (  # noqa: F401
    RayTrainReportCallback,
    prepare_trainer,
)
