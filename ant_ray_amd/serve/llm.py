"""ray.serve.llm parity: public wrappers over ant_ray_amd.llm
(reference python/ray/serve/llm/ re-exports the internal serve builders)."""
from ant_ray_amd.llm import (  # noqa: F401
    LLMConfig,
    build_llm_deployment,
    build_openai_app,
)
