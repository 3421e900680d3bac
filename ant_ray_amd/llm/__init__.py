"""ant_ray_amd.llm — LLM serving/batch APIs (vLLM-ROCm passthrough).

Role parity: reference python/ray/llm/ (~31k LoC): serve side wraps vLLM
engines behind Serve deployments (llm/_internal/serve/engines/vllm/
vllm_engine.py, OpenAI-compatible ingress), batch side runs vLLM inside
Ray Data stages (llm/_internal/batch/stages/vllm_engine_stage.py). Per
SURVEY.md §2.4/§2.5 the reference delegates ALL LLM compute to vLLM —
TP/PP/EP knobs are engine kwargs — so this module is the same thin shell:
config types + deployment/processor builders that hand off to vllm when
it is importable. This image has no vLLM; the builders raise a clear
ImportError at use (the API surface stays importable for parity).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


def _require_vllm():
    try:
        import vllm  # noqa: F401

        return vllm
    except ImportError as e:
        raise ImportError(
            "ray.llm delegates LLM compute to vLLM (reference design: "
            "llm/_internal/serve/engines/vllm/vllm_engine.py); install "
            "vllm-rocm in the image to use LLMServer/build_llm_processor. "
            "For native-kernel LLM inference without vLLM see "
            "ant_ray_amd.models + ops.attention."
        ) from e


@dataclass
class LLMConfig:
    """Parity: ray.serve.llm LLMConfig (model id + engine + scaling)."""

    model_loading_config: Dict[str, Any] = field(default_factory=dict)
    engine_kwargs: Dict[str, Any] = field(default_factory=dict)
    deployment_config: Dict[str, Any] = field(default_factory=dict)
    accelerator_type: Optional[str] = None
    runtime_env: Optional[dict] = None

    @property
    def model_id(self) -> str:
        return self.model_loading_config.get("model_id", "unknown")

    def tensor_parallel_size(self) -> int:
        return int(self.engine_kwargs.get("tensor_parallel_size", 1))

    def pipeline_parallel_size(self) -> int:
        return int(self.engine_kwargs.get("pipeline_parallel_size", 1))

    def num_gpus_per_replica(self) -> int:
        return self.tensor_parallel_size() * self.pipeline_parallel_size()


_NATIVE_MODELS = ("llama3-8b", "llama-3-8b", "llama3_8b", "llama-tiny",
                  "tiny", "llama-tiny-d128", "tiny-d128")


def build_llm_deployment(config: LLMConfig):
    """Serve deployment for an LLM. In-tree model ids (llama3-8b family)
    are served by the NATIVE MI355X engine (ant_ray_amd.llm.native_engine:
    hand-written prefill/decode kernels, KV cache, dynamic batching);
    other model ids delegate to vLLM when importable (parity:
    serve/llm build_llm_deployment -> vllm_engine.py)."""
    if config.model_id.lower() in _NATIVE_MODELS:
        from ant_ray_amd.llm.native_engine import build_native_llm_deployment

        return build_native_llm_deployment(
            model_name=config.model_id.lower(),
            num_replicas=config.deployment_config.get("num_replicas", 1),
            num_gpus=config.num_gpus_per_replica(),
            max_seq=int(config.engine_kwargs.get("max_model_len", 4096)),
            max_batch_size=int(config.engine_kwargs.get("max_num_seqs", 16)),
            # vLLM serves with continuous batching unconditionally; the
            # native engine keeps it opt-in (greedy-only token scheduler)
            continuous=bool(config.engine_kwargs.get(
                "continuous_batching", False)),
        )
    vllm = _require_vllm()
    from ant_ray_amd import serve

    @serve.deployment(
        num_replicas=config.deployment_config.get("num_replicas", 1),
        ray_actor_options={"num_gpus": config.num_gpus_per_replica()},
    )
    class LLMServer:
        def __init__(self):
            from vllm import LLM

            self.engine = LLM(model=config.model_id, **config.engine_kwargs)

        def __call__(self, prompt: str, **params):
            out = self.engine.generate([prompt], **params)
            return out[0].outputs[0].text

    return LLMServer.bind()


def build_openai_app(configs: List[LLMConfig]):
    """OpenAI-compatible ingress (parity: reference llm/_internal/serve/
    core/ingress/ — /v1/models + /v1/completions). In-tree model ids are
    served by the NATIVE engine; `prompt` may be a string (byte-level
    fallback tokenization — the in-tree models run random-init weights,
    there is no pretrained tokenizer in this air-gapped image) or a list
    of token ids; the response carries both `text` (byte-rendered) and
    `token_ids`. Other model ids require vllm (as the reference)."""
    native = [c for c in configs if c.model_id.lower() in _NATIVE_MODELS]
    if not native:
        _require_vllm()
        raise NotImplementedError(
            "OpenAI ingress for non-native models requires vllm; see "
            "reference llm/_internal/serve/core/ingress/")
    from ant_ray_amd import serve

    cfg = native[0]

    def _build_api():
        # built INSIDE the replica (a function-local FastAPI app cannot
        # survive cloudpickle — starlette State recursion); route
        # functions reach the replica through app.state.serve_self
        from fastapi import FastAPI

        api = FastAPI()

        @api.get("/v1/models")
        async def models():  # noqa: ANN202
            replica = api.state.serve_self
            return {"object": "list",
                    "data": [{"id": replica.model_id, "object": "model",
                              "owned_by": "ant-ray-amd"}]}

        @api.post("/v1/completions")
        async def completions(body: dict):  # noqa: ANN202
            import asyncio
            import time as _time

            replica = api.state.serve_self
            prompt = body.get("prompt", "")
            if isinstance(prompt, str):
                ids = [b % replica.engine.vocab
                       for b in prompt.encode("utf-8")] or [0]
            else:
                ids = [int(t) % replica.engine.vocab for t in prompt] or [0]
            max_tokens = int(body.get("max_tokens", 16))
            temperature = float(body.get("temperature", 0.0))
            loop = asyncio.get_running_loop()
            out = await loop.run_in_executor(
                None, lambda: replica.engine.generate_tokens(
                    [ids], max_tokens, temperature)[0])
            text = bytes(t % 256 for t in out).decode("utf-8",
                                                      errors="replace")
            return {
                "id": f"cmpl-{int(_time.time() * 1e6):x}",
                "object": "text_completion",
                "created": int(_time.time()),
                "model": replica.model_id,
                "choices": [{"index": 0, "text": text, "token_ids": out,
                             "finish_reason": "length"}],
                "usage": {"prompt_tokens": len(ids),
                          "completion_tokens": len(out),
                          "total_tokens": len(ids) + len(out)},
            }

        return api

    @serve.deployment(
        num_replicas=cfg.deployment_config.get("num_replicas", 1),
        ray_actor_options={"num_gpus": cfg.num_gpus_per_replica()},
    )
    @serve.ingress(_build_api)
    class OpenAIIngress:
        def __init__(self):
            import torch

            from ant_ray_amd.llm.native_engine import NativeLLMEngine

            device = ("cuda" if (cfg.num_gpus_per_replica() > 0
                                 and torch.cuda.is_available()) else "cpu")
            self.engine = NativeLLMEngine(
                cfg.model_id.lower(),
                max_seq=int(cfg.engine_kwargs.get("max_model_len", 4096)),
                device=device)
            self.model_id = cfg.model_id

    return OpenAIIngress.bind()


def build_llm_processor(config: LLMConfig, preprocess=None, postprocess=None):
    """Ray Data batch-inference processor (parity: ray.data.llm
    build_llm_processor -> vllm_engine_stage)."""
    vllm = _require_vllm()

    def processor(dataset):
        from ant_ray_amd.data.plan import ActorPoolStrategy

        class _VLLMStage:
            def __init__(self):
                from vllm import LLM

                self.engine = LLM(model=config.model_id,
                                  **config.engine_kwargs)

            def __call__(self, batch):
                prompts = list(batch["prompt"])
                outs = self.engine.generate(prompts)
                batch["generated_text"] = [o.outputs[0].text for o in outs]
                return batch

        ds = dataset
        if preprocess:
            ds = ds.map(preprocess)
        ds = ds.map_batches(
            _VLLMStage,
            compute=ActorPoolStrategy(
                size=config.deployment_config.get("num_replicas", 1)),
            num_gpus=config.num_gpus_per_replica(), batch_size=64,
        )
        if postprocess:
            ds = ds.map(postprocess)
        return ds

    return processor
