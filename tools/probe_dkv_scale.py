import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from ant_ray_amd.ops.functional import _hip

D, S = 128, 4096
scale = D ** -0.5
for (B, Hq, Hk) in [(1,1,1), (1,4,1), (1,32,8), (6,32,8)]:
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Hk, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Hk, S, D, device="cuda", dtype=torch.bfloat16)
    dout = torch.randn_like(q)
    o, lse = _hip().attn_fwd(q, k, v, scale, True, True)
    for _ in range(3):
        _hip().attn_bwd(dout, q, k, v, o, lse, scale, True)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(5):
        _hip().attn_bwd(dout, q, k, v, o, lse, scale, True)
    torch.cuda.synchronize()
    wgs = (S//128) * Hq * B
    print(f"B{B} Hq{Hq}: {(time.time()-t0)/5*1e3:8.2f} ms   ({wgs} WGs)")
