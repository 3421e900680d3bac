"""Usage-stats lib (parity: reference python/ray/_common/usage/usage_lib.py).

The reference collects opt-out telemetry and reports it to a public
endpoint; this air-gapped build keeps the same recording API but only
persists locally (session_dir/usage_stats.json) and NEVER transmits.
Disabled entirely with RAY_USAGE_STATS_ENABLED=0 (same knob).
"""
from __future__ import annotations

import json
import os
import threading
import time

_lock = threading.Lock()
_tags: dict = {}


class TagKey:
    """Well-known extra-usage tag names (subset of usage_pb2.TagKey)."""

    RLLIB_FRAMEWORK = "rllib_framework"
    TRAIN_TRAINER = "train_trainer"
    SERVE_API_VERSION = "serve_api_version"
    DATA_API = "data_api"
    CORE_STATE_API = "core_state_api"


def usage_stats_enabled() -> bool:
    return os.environ.get("RAY_USAGE_STATS_ENABLED", "1") not in ("0", "false")


def record_extra_usage_tag(key: str, value: str):
    """Record a library-usage tag (no-op when disabled; local-only)."""
    if not usage_stats_enabled():
        return
    with _lock:
        _tags[str(key)] = str(value)
    _flush()


def get_extra_usage_tags() -> dict:
    with _lock:
        return dict(_tags)


def _flush():
    try:
        from ant_ray_amd._private.worker import global_worker

        cw = global_worker.core_worker
        sd = getattr(cw, "session_dir", "") if cw else ""
        if not sd:
            return
        with _lock:
            snap = {"tags": dict(_tags), "ts": time.time(),
                    "transmitted": False}  # air-gapped: local record only
        with open(os.path.join(sd, "usage_stats.json"), "w") as f:
            json.dump(snap, f)
    except Exception:
        pass
