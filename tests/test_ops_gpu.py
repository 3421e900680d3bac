"""Numerics: every HIP kernel vs a plain PyTorch fp32 reference (same inputs).

All tests are @pytest.mark.gpu (MI355X). Tolerances reflect bf16 I/O with
fp32 accumulation in the kernels.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from ant_ray_amd import ops
    from ant_ray_amd.ops import reference as ref

    DEV = "cuda:0"


def _assert_close(a, b, atol=2e-2, rtol=2e-2, what=""):
    a = a.float().cpu()
    b = b.float().cpu()
    torch.testing.assert_close(a, b, atol=atol, rtol=rtol, msg=what)


@pytest.fixture(autouse=True)
def _seed():
    torch.manual_seed(1234)


class TestRMSNorm:
    def test_fwd(self):
        x = torch.randn(512, 4096, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(4096, dtype=torch.bfloat16, device=DEV)
        y = ops.rmsnorm(x, w)
        y_ref = ref.rmsnorm(x, w)
        _assert_close(y, y_ref, what="rmsnorm fwd")

    def test_bwd(self):
        x = torch.randn(256, 1024, dtype=torch.bfloat16, device=DEV, requires_grad=True)
        w = torch.randn(1024, dtype=torch.bfloat16, device=DEV, requires_grad=True)
        y = ops.rmsnorm(x, w)
        dy = torch.randn_like(y)
        y.backward(dy)
        xr = x.detach().clone().float().requires_grad_(True)
        wr = w.detach().clone().float().requires_grad_(True)
        rstd = torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + 1e-5)
        yr = xr * rstd * wr
        yr.backward(dy.float())
        _assert_close(x.grad, xr.grad, what="rmsnorm dx")
        _assert_close(w.grad, wr.grad, atol=5e-2, rtol=5e-2, what="rmsnorm dw")

    def test_fused_add(self):
        x = torch.randn(128, 2048, dtype=torch.bfloat16, device=DEV)
        res = torch.randn(128, 2048, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(2048, dtype=torch.bfloat16, device=DEV)
        res_ref = res.clone()
        y, h = ops.fused_add_rmsnorm(x, res, w)
        y_ref, h_ref = ref.add_rmsnorm(x, res_ref, w)
        _assert_close(h, h_ref, what="fused add h")
        _assert_close(y, y_ref, what="fused add y")


class TestRope:
    def test_qkv_roundtrip(self):
        B, S, Hq, Hk, D = 2, 256, 8, 2, 128
        cos, sin = ops.rope_tables(D, S, device=DEV)
        qkv = torch.randn(B, S, (Hq + 2 * Hk) * D, dtype=torch.bfloat16, device=DEV)
        qkv_in = qkv.clone()
        out = ops.rope_qkv(qkv, cos, sin, Hq, Hk, D)
        q_ref = ref.rope_apply(qkv_in[..., : Hq * D].view(B, S, Hq, D), cos, sin)
        k_ref = ref.rope_apply(
            qkv_in[..., Hq * D : (Hq + Hk) * D].view(B, S, Hk, D), cos, sin
        )
        _assert_close(out[..., : Hq * D].view(B, S, Hq, D), q_ref, what="rope q")
        _assert_close(
            out[..., Hq * D : (Hq + Hk) * D].view(B, S, Hk, D), k_ref, what="rope k"
        )
        # v region untouched
        assert torch.equal(out[..., (Hq + Hk) * D :], qkv_in[..., (Hq + Hk) * D :])

    def test_backward_is_inverse_rotation(self):
        B, S, Hq, Hk, D = 1, 64, 2, 1, 64
        cos, sin = ops.rope_tables(D, S, device=DEV)
        qkv = torch.randn(B, S, (Hq + 2 * Hk) * D, dtype=torch.bfloat16, device=DEV,
                          requires_grad=True)
        out = ops.rope_qkv(qkv.clone(), cos, sin, Hq, Hk, D)
        # autograd-level check: d(sum(out * g))/d qkv == rope^T(g)
        qkv2 = qkv.detach().clone().requires_grad_(True)
        out2 = ops.rope_qkv(qkv2 * 1.0, cos, sin, Hq, Hk, D)
        g = torch.randn_like(out2)
        out2.backward(g)
        gq_ref = ref.rope_apply(g[..., : Hq * D].view(B, S, Hq, D), cos, sin, backward=True)
        _assert_close(qkv2.grad[..., : Hq * D].view(B, S, Hq, D), gq_ref, what="rope bwd q")


class TestSwiGLU:
    def test_fwd_bwd(self):
        N, I = 1024, 2048
        gu = torch.randn(N, 2 * I, dtype=torch.bfloat16, device=DEV, requires_grad=True)
        out = ops.swiglu(gu)
        out_ref = ref.swiglu(gu.detach())
        _assert_close(out, out_ref, what="swiglu fwd")
        dout = torch.randn_like(out)
        out.backward(dout)
        gur = gu.detach().clone().float().requires_grad_(True)
        outr = torch.nn.functional.silu(gur[:, :I]) * gur[:, I:]
        outr.backward(dout.float())
        _assert_close(gu.grad, gur.grad, atol=5e-2, rtol=5e-2, what="swiglu bwd")


class TestCrossEntropy:
    def test_kernel_vs_ref(self):
        N, V = 64, 1024
        logits = torch.randn(N, V, dtype=torch.bfloat16, device=DEV) * 4
        targets = torch.randint(0, V, (N,), device=DEV, dtype=torch.int32)
        targets[::7] = -100
        logits_in = logits.clone()
        loss = ops.hip_ops().cross_entropy_fwd_bwd(logits, targets, 1.0, -100, True)
        loss_ref, dl_ref = ref.cross_entropy(logits_in, targets, -100)
        _assert_close(loss, loss_ref, what="ce loss")
        _assert_close(logits, dl_ref, atol=5e-3, rtol=5e-2, what="ce dlogits")

    def test_linear_ce_end_to_end(self):
        N, H, V = 128, 256, 512
        x = torch.randn(N, H, dtype=torch.bfloat16, device=DEV, requires_grad=True)
        w = (torch.randn(V, H, dtype=torch.bfloat16, device=DEV) * 0.05).requires_grad_(True)
        targets = torch.randint(0, V, (N,), device=DEV)
        loss = ops.linear_cross_entropy(x, w, targets, chunk_rows=32)
        loss.backward()
        xr = x.detach().float().clone().requires_grad_(True)
        wr = w.detach().float().clone().requires_grad_(True)
        loss_ref = torch.nn.functional.cross_entropy(xr @ wr.t(), targets.long())
        loss_ref.backward()
        _assert_close(loss, loss_ref, atol=5e-3, rtol=1e-2, what="lce loss")
        _assert_close(x.grad, xr.grad, atol=2e-2, rtol=5e-2, what="lce dx")
        _assert_close(w.grad, wr.grad, atol=2e-2, rtol=5e-2, what="lce dw")


class TestAdamW:
    def test_vs_reference(self):
        n = 4096
        p32 = torch.randn(n, device=DEV)
        pb = p32.to(torch.bfloat16)
        g = torch.randn(n, dtype=torch.bfloat16, device=DEV)
        m = torch.zeros(n, device=DEV)
        v = torch.zeros(n, device=DEV)
        p32_r, pb_r, m_r, v_r = p32.clone(), pb.clone(), m.clone(), v.clone()
        for step in (1, 2, 3):
            ops.adamw_step(p32, pb, g, m, v, lr=1e-2, b1=0.9, b2=0.95, eps=1e-8,
                           wd=0.1, step=step, grad_scale=0.5)
            ref.adamw_step(p32_r, pb_r, g, m_r, v_r, 1e-2, 0.9, 0.95, 1e-8, 0.1,
                           step, 0.5)
        _assert_close(p32, p32_r, atol=1e-5, rtol=1e-5, what="adamw p32")
        _assert_close(m, m_r, atol=1e-5, rtol=1e-5, what="adamw m")
        assert torch.equal(pb, pb_r)


class TestModelGPU:
    def test_llama_tiny_trains(self):
        from ant_ray_amd.models import build_model
        from ant_ray_amd.parallel import FlatAdamW, FlatParamManager

        torch.manual_seed(0)
        m = build_model("llama-tiny", device=DEV, seq_len=256)
        mgr = FlatParamManager(m)
        opt = FlatAdamW(mgr, lr=3e-3)
        tokens = torch.randint(0, 1024, (4, 256), device=DEV)
        losses = []
        for _ in range(8):
            loss = m(tokens, tokens)
            loss.backward()
            opt.step()
            opt.zero_grad()
            losses.append(loss.item())
        assert losses[-1] < losses[0] - 1.0, f"no learning: {losses}"

    def test_gpt2_fwd_bwd(self):
        from ant_ray_amd.models import build_model

        m = build_model("gpt2", device=DEV, seq_len=256)
        tok = torch.randint(0, 50304, (2, 256), device=DEV)
        loss = m(tok, tok)
        loss.backward()
        assert torch.isfinite(loss)


class TestAttentionDecode:
    """Flash-decode kernel (attention_decode.hip) vs fp32 reference."""

    def test_uniform_len(self):
        B, Hq, Hk, T, D = 4, 32, 8, 777, 128
        q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
        k = torch.randn(B, Hk, 1024, D, dtype=torch.bfloat16, device=DEV)
        v = torch.randn(B, Hk, 1024, D, dtype=torch.bfloat16, device=DEV)
        o = ops.attention_decode(q, k, v, seq_len=T)
        o_ref = ref.attention_decode_ref(q, k[:, :, :T], v[:, :, :T])
        _assert_close(o, o_ref, what="attn_decode uniform")

    def test_long_cache_multichunk(self):
        # T large enough that the two-level (chunked) combine path runs
        B, Hq, Hk, T, D = 1, 32, 8, 4096, 128
        q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
        k = torch.randn(B, Hk, T, D, dtype=torch.bfloat16, device=DEV)
        v = torch.randn(B, Hk, T, D, dtype=torch.bfloat16, device=DEV)
        o = ops.attention_decode(q, k, v, seq_len=T)
        o_ref = ref.attention_decode_ref(q, k, v)
        _assert_close(o, o_ref, what="attn_decode multichunk")

    def test_ragged_lens(self):
        B, Hq, Hk, T, D = 5, 8, 4, 512, 128
        q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
        k = torch.randn(B, Hk, T, D, dtype=torch.bfloat16, device=DEV)
        v = torch.randn(B, Hk, T, D, dtype=torch.bfloat16, device=DEV)
        lens = torch.tensor([3, 100, 512, 77, 256], dtype=torch.int32,
                            device=DEV)
        o = ops.attention_decode(q, k, v, seq_len=T, lens=lens)
        o_ref = ref.attention_decode_ref(q, k, v, lens=lens)
        _assert_close(o, o_ref, what="attn_decode ragged")

    def test_spiked_key(self):
        # forces the defer-max rescale branch (rule 26): one huge score
        B, Hq, Hk, T, D = 2, 4, 2, 300, 128
        q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
        k = torch.randn(B, Hk, T, D, dtype=torch.bfloat16, device=DEV) * 0.1
        v = torch.randn(B, Hk, T, D, dtype=torch.bfloat16, device=DEV)
        k[:, :, 237] = q[:, ::2, :] * 3.0  # aligned spike late in the walk
        o = ops.attention_decode(q, k, v, seq_len=T)
        o_ref = ref.attention_decode_ref(q, k, v)
        _assert_close(o, o_ref, what="attn_decode spiked")


class TestGenerate:
    def test_kv_cache_generate_matches_full_forward(self):
        """Greedy generation with the KV cache + decode kernel must match
        argmax stepping with full no-cache forwards."""
        from ant_ray_amd.models import build_model

        torch.manual_seed(7)
        m = build_model("llama-tiny-d128", device=DEV, seq_len=256)
        m.eval()
        tokens = torch.randint(0, 1024, (2, 17), device=DEV)
        out = m.generate(tokens, max_new_tokens=8)
        # reference: recompute the full sequence each step (no cache)
        cur = tokens.clone()
        with torch.no_grad():
            for _ in range(8):
                logits = m(cur)[:, -1, :]
                cur = torch.cat([cur, logits.argmax(-1, keepdim=True)], 1)
        # bf16 decode vs recompute can differ after an early divergence;
        # require the first few tokens to match exactly
        assert torch.equal(out[:, :17 + 4], cur[:, :17 + 4]), (
            out.tolist(), cur.tolist())

    def test_graphed_decoder_matches_eager_and_reuses_capture(self):
        """The persistent GraphedDecoder (hipGraph token step, device-pos
        rope/cache-write/flash-decode) must produce the SAME tokens as
        the eager host-pos loop, and keep producing them on a SECOND
        generate call that reuses the captured graph + KV cache."""
        import os

        from ant_ray_amd.models import build_model
        from ant_ray_amd.models.llama import GraphedDecoder

        torch.manual_seed(11)
        m = build_model("llama-tiny-d128", device=DEV, seq_len=256)
        m.eval()
        tokens = torch.randint(0, 1024, (2, 13), device=DEV)
        os.environ["ANTRAY_DECODE_GRAPH"] = "0"
        try:
            eager = m.generate(tokens, max_new_tokens=16)
        finally:
            os.environ.pop("ANTRAY_DECODE_GRAPH", None)
        dec = GraphedDecoder(m, batch=2, max_seq=256, device=DEV)
        first = dec.generate(tokens, 16)
        assert dec.graph is not None, "capture did not happen"
        second = dec.generate(tokens, 16)  # replays the SAME graph
        assert torch.equal(first, eager), (first.tolist(), eager.tolist())
        assert torch.equal(second, eager)

    def test_engine_prefix_cache_graphed_path(self):
        """Prefix-cached generation through the NativeLLMEngine on the
        graphed decode path: a second request sharing a system prompt
        reuses cached KV blocks (chunked prefill of the suffix) and
        produces the same leading tokens as an uncached engine."""
        import os

        from ant_ray_amd.llm.native_engine import NativeLLMEngine

        os.environ["ANTRAY_PREFIX_CACHE"] = "1"
        os.environ["ANTRAY_PREFIX_BLOCK"] = "16"
        try:
            eng = NativeLLMEngine("llama-tiny-d128", max_seq=256, seed=3,
                                  device=DEV)
            assert eng.prefix_cache is not None
            os.environ["ANTRAY_PREFIX_CACHE"] = "0"
            ref = NativeLLMEngine("llama-tiny-d128", max_seq=256, seed=3,
                                  device=DEV)
            assert ref.prefix_cache is None
        finally:
            os.environ.pop("ANTRAY_PREFIX_CACHE", None)
            os.environ.pop("ANTRAY_PREFIX_BLOCK", None)
        torch.manual_seed(5)
        sysp = torch.randint(0, 1024, (48,)).tolist()
        p1 = sysp + torch.randint(0, 1024, (9,)).tolist()
        p2 = sysp + torch.randint(0, 1024, (9,)).tolist()
        out1 = eng.generate_tokens([p1], 8)
        assert eng.prefix_cache.stats()["blocks"] >= 3
        out2 = eng.generate_tokens([p2], 8)  # hits the cached prefix
        assert eng.prefix_cache.stats()["tokens_reused"] >= 32
        assert len(out1[0]) == 8 and len(out2[0]) == 8
        r1 = ref.generate_tokens([p1], 8)
        # identical prefill paths (both cold) -> identical leading tokens
        assert out1[0][:4] == r1[0][:4], (out1, r1)
        # the HIT path's numerics: chunked prefill (masked SDPA over the
        # seeded cache) vs one-shot flash prefill. Tokens on a random-init
        # model can tie-flip in bf16, so compare next-token LOGITS with
        # tolerance instead of decoded tails.
        from ant_ray_amd.models.llama import KVCache

        m = eng.model
        toks = torch.tensor([p2], dtype=torch.long, device=DEV)
        with torch.no_grad():
            c1 = KVCache(m.cfg, 1, 256, DEV)
            full = m.forward(toks, cache=c1, pos=0)
            c2 = KVCache(m.cfg, 1, 256, DEV)
            m.forward(toks[:, :32], cache=c2, pos=0)
            split = m.forward(toks[:, 32:], cache=c2, pos=32)
        torch.testing.assert_close(full.float(), split.float(),
                                   rtol=5e-2, atol=5e-2)

    def test_continuous_engine_gpu(self):
        """Slot-based continuous batching on the graphed ragged-lens
        decode path: mixed-length requests through a shared 4-slot
        decoder match per-request generate()."""
        from ant_ray_amd.llm.continuous import ContinuousLLMEngine

        eng = ContinuousLLMEngine("llama-tiny-d128", slots=4, max_seq=256,
                                  device=DEV)
        torch.manual_seed(9)
        prompts = [torch.randint(0, 1024, (n,)).tolist()
                   for n in (9, 13, 7, 21, 5)]
        futs = [eng.submit(p, 8) for p in prompts]
        eng.run_until_idle()
        assert eng.dec.graph is not None, "step graph was not captured"
        assert eng.stats()["active"] == 0
        for p, f in zip(prompts, futs):
            out = f.result(timeout=0)
            assert len(out) == 8
            ref = eng.model.generate(
                torch.tensor([p], dtype=torch.long, device=DEV),
                8)[0, len(p):].tolist()
            # bf16 GEMM batch-shape effects can tie-flip late tokens on a
            # random-init model; require the leading tokens to agree
            assert out[:4] == ref[:4], (p, out, ref)


class TestDataTransforms:
    """Fused Data-plane kernels (data_transform.hip) vs torch reference."""

    def test_cast_affine_u8_scalar(self):
        x = torch.randint(0, 256, (4096, 128), dtype=torch.uint8, device=DEV)
        y = ops.cast_affine(x, scale=1 / 255.0, shift=127.5,
                            out_dtype=torch.bfloat16)
        y_ref = ((x.float() - 127.5) / 255.0).to(torch.bfloat16)
        _assert_close(y, y_ref, atol=1e-3, rtol=1e-2, what="cast_affine u8")

    def test_cast_affine_u8_per_channel_f32_out(self):
        C = 16
        x = torch.randint(0, 256, (1000, C), dtype=torch.uint8, device=DEV)
        scale = torch.rand(C, device=DEV) + 0.5
        shift = torch.rand(C, device=DEV) * 100
        y = ops.cast_affine(x, scale, shift, out_dtype=torch.float32)
        y_ref = (x.float() - shift) * scale
        _assert_close(y, y_ref, atol=1e-4, rtol=1e-4, what="cast_affine chan")

    def test_cast_affine_ragged_tail(self):
        x = torch.randint(0, 256, (3, 7, 13), dtype=torch.uint8, device=DEV)
        y = ops.cast_affine(x, 2.0, 1.0, out_dtype=torch.float32)
        _assert_close(y, (x.float() - 1.0) * 2.0, atol=1e-5, rtol=1e-5,
                      what="cast_affine tail")

    def test_nhwc_to_nchw(self):
        x = torch.randint(0, 256, (8, 32, 32, 3), dtype=torch.uint8,
                          device=DEV)
        mean = [123.7, 116.3, 103.5]
        std = [58.4, 57.1, 57.4]
        y = ops.nhwc_to_nchw(x, mean, std, out_dtype=torch.float32)
        m = torch.tensor(mean, device=DEV).view(1, 1, 1, 3)
        s = torch.tensor(std, device=DEV).view(1, 1, 1, 3)
        y_ref = ((x.float() - m) / s).permute(0, 3, 1, 2).contiguous()
        assert y.shape == (8, 3, 32, 32)
        _assert_close(y, y_ref, atol=1e-3, rtol=1e-3, what="nhwc_to_nchw")
