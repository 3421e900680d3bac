"""Serve shared types. Parity: reference python/ray/serve/_private/common.py
(DeploymentID, ReplicaState) and serve/config.py (AutoscalingConfig)."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

CONTROLLER_NAME = "SERVE_CONTROLLER_ACTOR"
DEFAULT_APP_NAME = "default"


@dataclass
class AutoscalingConfig:
    min_replicas: int = 1
    max_replicas: int = 10
    target_ongoing_requests: float = 2.0
    upscale_delay_s: float = 3.0
    downscale_delay_s: float = 30.0
    metrics_interval_s: float = 2.0

    @classmethod
    def coerce(cls, v):
        if v is None or isinstance(v, cls):
            return v
        return cls(**v)


@dataclass
class DeploymentConfig:
    name: str
    num_replicas: int = 1
    max_ongoing_requests: int = 100
    ray_actor_options: Dict[str, Any] = field(default_factory=dict)
    user_config: Any = None
    autoscaling_config: Optional[AutoscalingConfig] = None
    health_check_period_s: float = 10.0
    graceful_shutdown_timeout_s: float = 20.0


@dataclass
class ReplicaInfo:
    replica_id: str
    deployment: str
    app: str
    actor_id: bytes
