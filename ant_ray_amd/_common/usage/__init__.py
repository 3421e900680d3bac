from ant_ray_amd._common.usage.usage_lib import (  # noqa: F401
    TagKey,
    record_extra_usage_tag,
    usage_stats_enabled,
)
