"""Native LLM engine + Serve deployment (CPU: reference ops; GPU twin in
test_train_serve_gpu.py runs the flash-decode kernels)."""
import random

import pytest


@pytest.fixture()
def ray_cpu():
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()
    ray.init(num_cpus=4)
    yield ray
    ray.shutdown()


def test_native_engine_ragged_batch():
    import torch

    from ant_ray_amd.llm.native_engine import NativeLLMEngine

    eng = NativeLLMEngine("llama-tiny", max_seq=64, device="cpu")
    rng = random.Random(1)
    prompts = [[rng.randrange(1024) for _ in range(n)] for n in (5, 9, 5)]
    outs = eng.generate_tokens(prompts, max_new_tokens=4)
    assert [len(o) for o in outs] == [4, 4, 4]
    # same-length prompts ran as one group; must equal individual runs
    solo = eng.generate_tokens([prompts[0]], max_new_tokens=4)
    assert outs[0] == solo[0]


def test_serve_native_llm_deployment(ray_cpu):
    from ant_ray_amd import serve
    from ant_ray_amd.llm import LLMConfig, build_llm_deployment

    app = build_llm_deployment(LLMConfig(
        model_loading_config={"model_id": "llama-tiny"},
        engine_kwargs={"max_model_len": 64, "max_num_seqs": 4,
                       "tensor_parallel_size": 0},  # 0 GPUs -> CPU engine
        deployment_config={"num_replicas": 1},
    ))
    h = serve.run(app, name="llm-t", route_prefix="/llm-t")
    rng = random.Random(0)
    reqs = [h.remote({"prompt_ids": [rng.randrange(1024) for _ in range(12)],
                      "max_new_tokens": 5}) for _ in range(6)]
    outs = [r.result(timeout_s=300) for r in reqs]
    assert all(len(o["token_ids"]) == 5 for o in outs)
    serve.shutdown()


def test_serve_native_llm_continuous(ray_cpu):
    """Token-level continuous batching through Serve (engine_kwargs
    continuous_batching=True -> ContinuousLLMEngine replica)."""
    from ant_ray_amd import serve
    from ant_ray_amd.llm import LLMConfig, build_llm_deployment

    app = build_llm_deployment(LLMConfig(
        model_loading_config={"model_id": "llama-tiny"},
        engine_kwargs={"max_model_len": 64, "max_num_seqs": 4,
                       "tensor_parallel_size": 0,
                       "continuous_batching": True},
        deployment_config={"num_replicas": 1},
    ))
    h = serve.run(app, name="llm-cb", route_prefix="/llm-cb")
    rng = random.Random(1)
    prompts = [[rng.randrange(1024) for _ in range(rng.randrange(4, 16))]
               for _ in range(8)]
    reqs = [h.remote({"prompt_ids": p, "max_new_tokens": 6})
            for p in prompts]
    outs = [r.result(timeout_s=300) for r in reqs]
    assert all(len(o["token_ids"]) == 6 for o in outs)
    # exactness vs a local reference model (same seed -> same weights)
    import torch

    from ant_ray_amd.models import build_model

    torch.manual_seed(0)
    m = build_model("llama-tiny", device="cpu", seq_len=64)
    m.eval()
    for p, o in zip(prompts, outs):
        ref = m.generate(torch.tensor([p]), 6)[0, len(p):].tolist()
        assert o["token_ids"] == ref
    serve.shutdown()


def test_openai_ingress_native_cpu(ray_cpu):
    """OpenAI-compatible /v1/completions over the native engine (CPU,
    tiny model): request/response shapes mirror the OpenAI API."""
    import urllib.request
    import json as _json

    from ant_ray_amd import serve
    from ant_ray_amd.llm import LLMConfig, build_openai_app

    app = build_openai_app([LLMConfig(
        model_loading_config={"model_id": "llama-tiny-d128"},
        engine_kwargs={"max_model_len": 128,
                       "tensor_parallel_size": 0},  # 0 GPUs -> CPU engine
        deployment_config={"num_replicas": 1},
    )])
    serve.run(app, name="oai", route_prefix="/")
    try:
        base = "http://127.0.0.1:8000"
        with urllib.request.urlopen(base + "/v1/models", timeout=30) as r:
            models = _json.load(r)
        assert models["data"][0]["id"] == "llama-tiny-d128"
        body = _json.dumps({"prompt": "hello world", "max_tokens": 4,
                            "temperature": 0.0}).encode()
        req = urllib.request.Request(
            base + "/v1/completions", data=body,
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=120) as r:
            out = _json.load(r)
        assert out["object"] == "text_completion"
        assert len(out["choices"][0]["token_ids"]) == 4
        assert out["usage"]["completion_tokens"] == 4
        # token-ids prompt form
        body = _json.dumps({"prompt": [1, 2, 3, 4], "max_tokens": 3}).encode()
        req = urllib.request.Request(
            base + "/v1/completions", data=body,
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=120) as r:
            out = _json.load(r)
        assert len(out["choices"][0]["token_ids"]) == 3
    finally:
        serve.shutdown()
