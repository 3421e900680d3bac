from ant_ray_amd._common.usage import usage_lib  # noqa: F401
