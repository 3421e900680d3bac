"""Continuous (token-level) batching scheduler (llm/continuous.py).

Role parity: vLLM continuous batching (reference python/ray/llm/
_internal/serve/engines/vllm/vllm_engine.py:1). On CPU every op falls
back to the exact fp32 reference, so slot-decoded outputs must equal
per-request model.generate() exactly, regardless of what the other
slots are doing.
"""
import time

import torch

from ant_ray_amd.llm.continuous import ContinuousLLMEngine


def _ref_tokens(model, prompt, n):
    toks = torch.tensor([prompt], dtype=torch.long)
    return model.generate(toks, n)[0, len(prompt):].tolist()


def test_continuous_matches_generate_with_queueing():
    eng = ContinuousLLMEngine("llama-tiny", slots=2, max_seq=128,
                              device="cpu")
    prompts = [[5, 6, 7], [9, 8, 7, 6, 5], [11, 12], [3, 3, 3, 3]]
    news = [6, 4, 5, 3]
    futs = [eng.submit(p, n) for p, n in zip(prompts, news)]
    eng.run_until_idle()
    for p, n, f in zip(prompts, news, futs):
        assert f.result(timeout=0) == _ref_tokens(eng.model, p, n), (p, n)
    st = eng.stats()
    assert st["active"] == 0 and st["queued"] == 0
    # 4 requests through 2 slots: continuous admission, not request
    # batching — fewer steps than serial sum of lengths
    assert st["steps"] <= sum(news)


def test_continuous_mid_run_admission():
    """A request submitted while another is mid-decode joins the running
    batch and both outputs stay exact."""
    eng = ContinuousLLMEngine("llama-tiny", slots=4, max_seq=128,
                              device="cpu")
    f1 = eng.submit([42, 17, 8, 100], 10)
    for _ in range(4):
        eng.pump()
    f2 = eng.submit([7, 7, 7], 5)
    eng.run_until_idle()
    assert f1.result(timeout=0) == _ref_tokens(eng.model, [42, 17, 8, 100],
                                               10)
    assert f2.result(timeout=0) == _ref_tokens(eng.model, [7, 7, 7], 5)


def test_continuous_slot_reuse_isolation():
    """A retired slot's stale cache must not leak into the request that
    reuses the slot."""
    eng = ContinuousLLMEngine("llama-tiny", slots=1, max_seq=128,
                              device="cpu")
    f1 = eng.submit([100, 101, 102, 103, 104, 105], 8)
    f2 = eng.submit([55, 44], 6)  # queued; reuses slot 0 after f1
    eng.run_until_idle()
    assert f1.result(timeout=0) == _ref_tokens(
        eng.model, [100, 101, 102, 103, 104, 105], 8)
    assert f2.result(timeout=0) == _ref_tokens(eng.model, [55, 44], 6)


def test_continuous_background_thread():
    eng = ContinuousLLMEngine("llama-tiny", slots=2, max_seq=128,
                              device="cpu", start_thread=True)
    try:
        futs = [eng.submit([i + 1, i + 2, i + 3], 4) for i in range(5)]
        deadline = time.time() + 60
        for f in futs:
            out = f.result(timeout=max(1.0, deadline - time.time()))
            assert len(out) == 4
        for i, f in enumerate(futs):
            assert f.result() == _ref_tokens(
                eng.model, [i + 1, i + 2, i + 3], 4)
    finally:
        eng.shutdown()


def test_continuous_with_prefix_cache(monkeypatch):
    """Requests sharing a system prompt reuse its KV blocks across slot
    admissions; outputs stay exact."""
    monkeypatch.setenv("ANTRAY_PREFIX_CACHE", "1")
    monkeypatch.setenv("ANTRAY_PREFIX_BLOCK", "8")
    eng = ContinuousLLMEngine("llama-tiny", slots=2, max_seq=128,
                              device="cpu")
    assert eng.prefix_cache is not None
    import random

    rng = random.Random(3)
    sysp = [rng.randrange(256) for _ in range(24)]
    prompts = [sysp + [rng.randrange(256) for _ in range(4)]
               for _ in range(4)]
    futs = [eng.submit(p, 5) for p in prompts]
    eng.run_until_idle()
    assert eng.prefix_cache.stats()["tokens_reused"] >= 24 * 2
    for p, f in zip(prompts, futs):
        assert f.result(timeout=0) == _ref_tokens(eng.model, p, 5)


def test_continuous_stop_tokens_and_streaming():
    """stop_token_ids ends a request early (stop token included);
    on_token streams every emitted token; a plain request sharing the
    batch is unaffected."""
    eng = ContinuousLLMEngine("llama-tiny", slots=2, max_seq=128,
                              device="cpu")
    p = [21, 22, 23, 24]
    ref = _ref_tokens(eng.model, p, 10)
    stop = ref[3]
    streamed = []
    f = eng.submit(p, 10, stop_token_ids=[stop],
                   on_token=streamed.append)
    p2 = [9, 9, 9]
    f2 = eng.submit(p2, 6)
    eng.run_until_idle()
    out = f.result(timeout=0)
    k = ref.index(stop) + 1
    assert out == ref[:k], (out, ref)
    assert streamed == out
    assert f2.result(timeout=0) == _ref_tokens(eng.model, p2, 6)


def test_continuous_rejects_oversized():
    eng = ContinuousLLMEngine("llama-tiny", slots=1, max_seq=32,
                              device="cpu")
    try:
        eng.submit(list(range(30)), 10)
        assert False, "expected ValueError"
    except ValueError:
        pass
