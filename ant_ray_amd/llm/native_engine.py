"""Native MI355X LLM engine: in-tree model family + flash-decode kernels.

Role parity: the reference delegates all LLM serving compute to vLLM
(reference python/ray/llm/_internal/serve/engines/vllm/vllm_engine.py:1);
this image has no vLLM, so the framework ships its OWN engine for the
models it implements natively (ant_ray_amd.models): bf16 weights, the
hand-written CDNA4 kernel family for prefill (attention.hip) and decode
(attention_decode.hip), greedy/temperature sampling, KV cache.

Batching model: requests arriving within a batching window are grouped by
prompt length (the prefill kernel is dense-causal; no pad masking), each
group runs one prefill + shared decode loop. This is dynamic batching at
request granularity — continuous (token-level) batching is future work.
"""
from __future__ import annotations

import threading
from typing import List, Optional

import torch


class NativeLLMEngine:
    """One engine per GPU replica; thread-safe (serialized) generate."""

    def __init__(self, model_name: str, max_seq: int = 4096,
                 seed: int = 0, device: str = "cuda"):
        from ant_ray_amd.models import build_model, setup_tunableop

        if str(device).startswith("cuda"):
            setup_tunableop()  # tuned hipBLASLt algos for the skinny
            # decode GEMM shapes (profiles/tunableop_gfx950.csv)
        torch.manual_seed(seed)
        self.model = build_model(model_name, device=device, seq_len=max_seq)
        self.model.eval()
        self.device = device
        self.max_seq = max_seq
        self.vocab = self.model.cfg.vocab
        self._lock = threading.Lock()
        # hash-based block prefix cache (vLLM-parity; see prefix_cache.py):
        # shared prompt prefixes skip their prefill FLOPs
        from ant_ray_amd.llm.prefix_cache import prefix_cache_from_env

        self.prefix_cache = prefix_cache_from_env()
        # persistent per-batch-bucket graphed decoders (greedy + CUDA +
        # D=128): KV cache and hipGraph capture are paid once per bucket,
        # then every token of every request is one graph replay
        self._decoders = {}
        if (str(device).startswith("cuda")
                and self.model.cfg.head_dim == 128):
            # pre-capture the B=1 bucket: the process's FIRST hipGraph
            # instantiation costs ~0.7 s — pay it at replica startup,
            # not on the first request
            try:
                dec, _ = self._graphed_decoder(1)
                dummy = torch.zeros(1, 8, dtype=torch.long, device=device)
                dec.generate(dummy, 12)
            except Exception:
                pass

    def _graphed_decoder(self, batch: int):
        from ant_ray_amd.models.llama import GraphedDecoder

        bucket = 1
        while bucket < batch:
            bucket *= 2
        dec = self._decoders.get(bucket)
        if dec is None:
            dec = GraphedDecoder(self.model, bucket, self.max_seq,
                                 self.device)
            self._decoders[bucket] = dec
        return dec, bucket

    @torch.no_grad()
    def generate_tokens(self, prompts: List[List[int]], max_new_tokens: int,
                        temperature: float = 0.0) -> List[List[int]]:
        """prompts: per-request token id lists (ragged ok — grouped by
        length). Returns the NEW tokens per request (without the prompt)."""
        with self._lock:
            by_len = {}
            for i, p in enumerate(prompts):
                by_len.setdefault(len(p), []).append(i)
            out: List[Optional[List[int]]] = [None] * len(prompts)
            use_graph = (temperature == 0
                         and str(self.device).startswith("cuda")
                         and self.model.cfg.head_dim == 128)
            for plen, idxs in by_len.items():
                group = [prompts[i] for i in idxs]
                toks = torch.tensor(group, dtype=torch.long,
                                    device=self.device)
                # prefix-cache lookup on the group's COMMON prefix (serve
                # batches typically share a system prompt); keep >=1 token
                # to prefill
                hit, blocks, common = 0, [], plen
                if self.prefix_cache is not None:
                    for c, col in enumerate(zip(*group)):
                        if any(t != col[0] for t in col):
                            common = c
                            break
                    hit, blocks = self.prefix_cache.lookup(
                        group[0][: min(common, plen - 1)])
                cache = None
                if use_graph:
                    dec, bucket = self._graphed_decoder(len(idxs))
                    if len(idxs) < bucket:  # pad rows replay row 0
                        toks = torch.cat(
                            [toks, toks[:1].expand(bucket - len(idxs), -1)])
                    if hit:
                        self.prefix_cache.seed(dec.cache, blocks)
                    gen = dec.generate(toks, max_new_tokens, prefix_len=hit)
                    cache = dec.cache
                else:
                    from ant_ray_amd.models.llama import (KVCache,
                                                          LlamaForCausalLM)

                    if isinstance(self.model, LlamaForCausalLM):
                        cache = KVCache(
                            self.model.cfg, toks.shape[0],
                            min(self.max_seq, plen + max_new_tokens),
                            self.device)
                        if hit:
                            self.prefix_cache.seed(cache, blocks)
                        gen = self.model.generate(
                            toks, max_new_tokens, cache=cache,
                            temperature=temperature, start_pos=hit)
                    else:  # model family without a llama KVCache
                        gen = self.model.generate(toks, max_new_tokens,
                                                  temperature=temperature)
                if self.prefix_cache is not None and cache is not None:
                    # store the group's common prompt prefix for reuse
                    self.prefix_cache.insert(group[0], cache, row=0,
                                             upto=common)
                new = gen[: len(idxs), plen:].tolist()
                for j, i in enumerate(idxs):
                    out[i] = new[j]
            return out


def build_native_llm_deployment(model_name: str = "llama3-8b",
                                num_replicas: int = 1,
                                num_gpus: float = 1,
                                max_seq: int = 4096,
                                max_batch_size: int = 16,
                                batch_wait_timeout_s: float = 0.02,
                                continuous: bool = False):
    """Serve deployment serving the native engine.

    continuous=False: dynamic request batching (@serve.batch groups by
    prompt length). continuous=True: token-level continuous batching —
    each request joins the running decode batch immediately
    (llm/continuous.py ContinuousLLMEngine; greedy only).

    Request payload: {"prompt_ids": [int], "max_new_tokens": int,
    "temperature": float} -> {"token_ids": [int]}.
    """
    from ant_ray_amd import serve

    if continuous:
        @serve.deployment(
            num_replicas=num_replicas,
            ray_actor_options={"num_gpus": num_gpus},
            max_ongoing_requests=max(64, 4 * max_batch_size),
        )
        class NativeLLMServerCB:
            def __init__(self):
                import asyncio  # noqa: F401

                from ant_ray_amd.llm.continuous import ContinuousLLMEngine

                self.engine = ContinuousLLMEngine(
                    model_name, slots=max_batch_size, max_seq=max_seq,
                    device="cuda" if num_gpus > 0 else "cpu",
                    start_thread=True)

            async def __call__(self, request: dict) -> dict:
                import asyncio

                fut = self.engine.submit(
                    request["prompt_ids"],
                    int(request.get("max_new_tokens", 32)),
                    stop_token_ids=request.get("stop_token_ids"))
                return {"token_ids": await asyncio.wrap_future(fut)}

        return NativeLLMServerCB.bind()

    @serve.deployment(
        num_replicas=num_replicas,
        ray_actor_options={"num_gpus": num_gpus},
        max_ongoing_requests=max(32, 2 * max_batch_size),
    )
    class NativeLLMServer:
        def __init__(self):
            self.engine = NativeLLMEngine(
                model_name, max_seq=max_seq,
                device="cuda" if num_gpus > 0 else "cpu")

        @serve.batch(max_batch_size=max_batch_size,
                     batch_wait_timeout_s=batch_wait_timeout_s)
        async def _generate(self, requests: List[dict]) -> List[dict]:
            prompts = [r["prompt_ids"] for r in requests]
            mnt = max(int(r.get("max_new_tokens", 32)) for r in requests)
            temp = float(requests[0].get("temperature", 0.0))
            outs = self.engine.generate_tokens(prompts, mnt, temp)
            return [{"token_ids": o} for o in outs]

        async def __call__(self, request: dict) -> dict:
            return await self._generate(request)

    return NativeLLMServer.bind()
