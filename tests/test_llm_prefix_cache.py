"""Prefix caching for the native LLM engine (llm/prefix_cache.py).

Role parity: vLLM automatic prefix caching (reference python/ray/llm/
_internal/serve/engines/vllm/vllm_engine.py:1 engine kwargs). CPU tests:
the decode path falls back to the exact fp32 reference op, so cached vs
uncached generation must produce identical tokens.
"""
import random

import pytest
import torch

from ant_ray_amd.llm.prefix_cache import PrefixKVCache
from ant_ray_amd.models import build_model
from ant_ray_amd.models.llama import KVCache


def _mk_engine(enabled=True, monkeypatch=None, seed=0):
    from ant_ray_amd.llm.native_engine import NativeLLMEngine

    if monkeypatch is not None:
        monkeypatch.setenv("ANTRAY_PREFIX_CACHE", "1" if enabled else "0")
        monkeypatch.setenv("ANTRAY_PREFIX_BLOCK", "16")
    return NativeLLMEngine("llama-tiny", max_seq=512, seed=seed,
                           device="cpu")


def test_prefix_cache_unit_lookup_insert_evict():
    model = build_model("llama-tiny", device="cpu", seq_len=256)
    cfg = model.cfg
    cache = KVCache(cfg, 1, 256, "cpu")
    torch.manual_seed(0)
    cache.k.normal_()
    cache.v.normal_()

    pc = PrefixKVCache(block_size=16, budget_mb=1024)
    toks = list(range(100, 180))  # 80 tokens = 5 blocks
    assert pc.insert(toks, cache, row=0) == 5

    # full hit: block-aligned prefix comes back in order
    n, blocks = pc.lookup(toks)
    assert n == 80 and len(blocks) == 5
    assert torch.equal(blocks[0].k, cache.k[:, 0, :, 0:16])
    assert torch.equal(blocks[4].v, cache.v[:, 0, :, 64:80])

    # partial hit: shared 32-token prefix then divergence
    n, blocks = pc.lookup(toks[:32] + [1, 2, 3] + toks[35:])
    assert n == 32 and len(blocks) == 2

    # no hit when the FIRST block differs (chain hashing)
    n, blocks = pc.lookup([9] + toks[1:])
    assert n == 0 and blocks == []

    # sub-block prompts never hit
    n, _ = pc.lookup(toks[:15])
    assert n == 0

    # seeding copies blocks into every row of a live cache
    tgt = KVCache(cfg, 3, 256, "cpu")
    tgt.k.zero_()
    tgt.v.zero_()
    _, blocks = pc.lookup(toks[:48])
    pc.seed(tgt, blocks)
    for r in range(3):
        assert torch.equal(tgt.k[:, r, :, :48], cache.k[:, 0, :, :48])

    st = pc.stats()
    assert st["blocks"] == 5 and st["tokens_reused"] > 0


def test_prefix_cache_lru_eviction():
    model = build_model("llama-tiny", device="cpu", seq_len=256)
    cache = KVCache(model.cfg, 1, 256, "cpu")
    cache.k.normal_()
    cache.v.normal_()
    # one block = n_layers*Hk*bs*D*2(dtypes)*2(k+v) bytes; pick a budget
    # that holds ~3 blocks
    blk_bytes = model.cfg.n_layers * model.cfg.n_kv_heads * 16 * \
        model.cfg.head_dim * 2 * 2
    pc = PrefixKVCache(block_size=16, budget_mb=3.2 * blk_bytes / (1 << 20))
    pc.insert(list(range(0, 48)), cache, row=0)       # 3 blocks
    assert len(pc._blocks) == 3
    pc.insert(list(range(1000, 1032)), cache, row=0)  # 2 more -> evicts
    assert len(pc._blocks) <= 3
    # the oldest chain was evicted; the newest survives
    n, _ = pc.lookup(list(range(1000, 1032)))
    assert n == 32


def test_chunked_prefill_matches_full_prefill():
    """forward(pos=p) over a seeded cache == one-shot prefill (fp32 ref
    path on CPU, exact to bf16 rounding)."""
    torch.manual_seed(1)
    model = build_model("llama-tiny", device="cpu", seq_len=256)
    model.eval()
    toks = torch.randint(0, model.cfg.vocab, (2, 96))
    with torch.no_grad():
        c1 = KVCache(model.cfg, 2, 128, "cpu")
        full = model.forward(toks, cache=c1, pos=0)
        c2 = KVCache(model.cfg, 2, 128, "cpu")
        model.forward(toks[:, :64], cache=c2, pos=0)
        split = model.forward(toks[:, 64:], cache=c2, pos=64)
    assert torch.equal(c1.k[:, :, :, :96], c2.k[:, :, :, :96])
    torch.testing.assert_close(full.float(), split.float(),
                               rtol=2e-2, atol=2e-2)
    assert torch.equal(full.argmax(-1), split.argmax(-1))


def test_engine_prefix_cached_generation_matches(monkeypatch):
    """Same prompts through a prefix-cached engine and a disabled one:
    identical tokens, and the second call actually reuses blocks."""
    random.seed(7)
    sys_prompt = [random.randrange(0, 256) for _ in range(48)]
    p1 = sys_prompt + [random.randrange(0, 256) for _ in range(10)]
    p2 = sys_prompt + [random.randrange(0, 256) for _ in range(10)]

    eng = _mk_engine(enabled=True, monkeypatch=monkeypatch)
    assert eng.prefix_cache is not None
    ref = _mk_engine(enabled=False, monkeypatch=monkeypatch)
    assert ref.prefix_cache is None

    out1 = eng.generate_tokens([p1], 8)
    assert eng.prefix_cache.stats()["blocks"] >= 3  # 48/16 common blocks
    out2 = eng.generate_tokens([p2], 8)            # hits the cached prefix
    assert eng.prefix_cache.stats()["tokens_reused"] >= 48 - 16

    assert out1 == ref.generate_tokens([p1], 8)
    assert out2 == ref.generate_tokens([p2], 8)


def test_engine_batched_common_prefix(monkeypatch):
    """A batch whose rows share a system prompt: the common prefix is
    cached once and a later batch reuses it; outputs match the
    uncached engine row for row."""
    random.seed(11)
    sysp = [random.randrange(0, 256) for _ in range(32)]
    batch = [sysp + [random.randrange(0, 256) for _ in range(6)]
             for _ in range(3)]

    eng = _mk_engine(enabled=True, monkeypatch=monkeypatch)
    ref = _mk_engine(enabled=False, monkeypatch=monkeypatch)
    out_a = eng.generate_tokens(batch, 6)
    assert out_a == ref.generate_tokens(batch, 6)

    batch2 = [sysp + [random.randrange(0, 256) for _ in range(6)]
              for _ in range(2)]
    reused0 = eng.prefix_cache.stats()["tokens_reused"]
    out_b = eng.generate_tokens(batch2, 6)
    assert eng.prefix_cache.stats()["tokens_reused"] > reused0
    assert out_b == ref.generate_tokens(batch2, 6)
