"""Time-breakdown of the chunked-prefill path on the llama3-8b shape.

python tools/prof_chunked.py  (GPU) -> one JSON line
"""
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timeit(f, n=10):
    for _ in range(2):
        f()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / n


def main():
    from ant_ray_amd import ops
    from ant_ray_amd.models import build_model, setup_tunableop
    from ant_ray_amd.models.llama import KVCache

    setup_tunableop()
    m = build_model("llama3-8b", device="cuda", seq_len=4096)
    m.eval()
    B, S, pos = 1, 64, 3072
    toks = torch.randint(0, 128000, (1, pos + S), device="cuda")
    out = {}
    with torch.no_grad():
        c = KVCache(m.cfg, 1, 4096, "cuda")
        out["full_prefill_ms"] = timeit(
            lambda: m.forward(toks, cache=c, pos=0), 5) * 1e3
        m.forward(toks[:, :pos], cache=c, pos=0)
        out["chunked_fast_ms"] = timeit(
            lambda: m.forward(toks[:, pos:], cache=c, pos=pos), 5) * 1e3
        os.environ["ANTRAY_FLASH"] = "0"
        out["chunked_sdpa_ms"] = timeit(
            lambda: m.forward(toks[:, pos:], cache=c, pos=pos), 5) * 1e3
        os.environ.pop("ANTRAY_FLASH", None)

        # isolated attention op at layer shape
        Hq, Hk, D = m.cfg.n_heads, m.cfg.n_kv_heads, m.cfg.head_dim
        ck, cv = c.layer(0)
        q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
        kn = ck[:, :, pos:pos + S]
        vn = cv[:, :, pos:pos + S]
        out["op_total_us"] = timeit(
            lambda: ops.chunked_prefill_attention(q, kn, vn, ck, cv, pos),
            50) * 1e6

        # pieces
        from ant_ray_amd.ops import _hip  # noqa
        hip = ops.functional._hip()
        scale = D ** -0.5
        out["op_flash_us"] = timeit(
            lambda: hip.attn_fwd(q, kn, vn, scale, True, True), 50) * 1e6
        G = Hq // Hk
        qg = q.reshape(B, Hk, G * S, D)
        kp = ck[:, :, :pos]
        vp = cv[:, :, :pos]
        out["op_qg_reshape_us"] = timeit(
            lambda: q.reshape(B, Hk, G * S, D), 50) * 1e6
        out["op_qk_matmul_us"] = timeit(
            lambda: torch.matmul(qg, kp.transpose(-1, -2)), 50) * 1e6
        sp = torch.matmul(qg, kp.transpose(-1, -2)).float() * scale
        out["op_softmax_stats_us"] = timeit(
            lambda: (sp.amax(-1, keepdim=True),
                     torch.exp(sp - sp.amax(-1, keepdim=True)).sum(-1)),
            50) * 1e6
        eA = torch.exp(sp - sp.amax(-1, keepdim=True))
        out["op_pv_matmul_us"] = timeit(
            lambda: torch.matmul(eA.to(torch.bfloat16), vp), 50) * 1e6
    print(json.dumps({k: round(v, 2) for k, v in out.items()}))


if __name__ == "__main__":
    main()
