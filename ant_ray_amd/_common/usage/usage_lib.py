"""Usage stats (opt-out telemetry) — collection parity, no egress.

Role parity: reference python/ray/_common/usage/usage_lib.py: collects
cluster metadata + feature-usage tags and reports to a telemetry endpoint
unless RAY_USAGE_STATS_ENABLED=0. This deployment is air-gapped: the same
report payload is assembled and written to the session dir instead of
being POSTed anywhere.
"""
from __future__ import annotations

import json
import os
import time
from typing import Any, Dict

_tags: Dict[str, str] = {}


def usage_stats_enabled() -> bool:
    return os.environ.get("RAY_USAGE_STATS_ENABLED", "1") == "1"


def record_library_usage(library: str):
    _tags[f"library_{library}"] = "1"


def record_extra_usage_tag(key: str, value: str):
    _tags[str(key)] = str(value)


def generate_report() -> Dict[str, Any]:
    import platform

    import ant_ray_amd

    return {
        "schema_version": "0.1",
        "source": "ant_ray_amd",
        "session_start_timestamp_ms": int(time.time() * 1000),
        "os": platform.system().lower(),
        "python_version": platform.python_version(),
        "version": ant_ray_amd.__version__,
        "extra_usage_tags": dict(_tags),
        "total_num_cpus": os.cpu_count(),
    }


def write_report(session_dir: str) -> str:
    if not usage_stats_enabled():
        return ""
    path = os.path.join(session_dir, "usage_stats.json")
    try:
        with open(path, "w") as f:
            json.dump(generate_report(), f)
    except OSError:
        return ""
    return path
