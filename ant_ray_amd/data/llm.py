"""ray.data.llm parity: batch-inference processor builder
(reference python/ray/data/llm.py wraps the internal batch stages)."""
from ant_ray_amd.llm import LLMConfig, build_llm_processor  # noqa: F401
