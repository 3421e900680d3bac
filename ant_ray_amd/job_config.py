"""JobConfig (parity: reference job_config.py — per-job runtime env,
namespace and metadata carried into ray.init / the client builder)."""
from typing import Any, Dict, Optional


class JobConfig:
    def __init__(self, jvm_options: Optional[list] = None,
                 code_search_path: Optional[list] = None,
                 runtime_env: Optional[dict] = None,
                 metadata: Optional[Dict[str, str]] = None,
                 ray_namespace: Optional[str] = None,
                 default_actor_lifetime: str = "non_detached", **_):
        self.jvm_options = jvm_options or []
        self.code_search_path = code_search_path or []
        self.runtime_env = runtime_env or {}
        self.metadata = metadata or {}
        self.ray_namespace = ray_namespace
        self.default_actor_lifetime = default_actor_lifetime

    def set_runtime_env(self, runtime_env: Optional[dict]) -> "JobConfig":
        self.runtime_env = runtime_env or {}
        return self

    def set_ray_namespace(self, ns: str) -> "JobConfig":
        self.ray_namespace = ns
        return self

    def set_metadata(self, key: str, value: str) -> "JobConfig":
        self.metadata[key] = value
        return self

    def serialize(self) -> Dict[str, Any]:
        return {"runtime_env": self.runtime_env, "metadata": self.metadata,
                "ray_namespace": self.ray_namespace}
