"""Standalone dkv kernel micro-bench (for PMC counter runs)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

import ant_ray_amd.ops as ops
from ant_ray_amd.ops.functional import _hip


def main():
    B, Hq, Hk, S, D = 6, 32, 8, 4096, 128
    scale = D ** -0.5
    torch.manual_seed(0)
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Hk, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Hk, S, D, device="cuda", dtype=torch.bfloat16)
    dout = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    o, lse = _hip().attn_fwd(q, k, v, scale, True, True)
    delta = (dout.float() * o.float()).sum(-1)

    import ctypes
    # call only the dkv launch through attn_bwd but time pieces via events
    for trial in range(3):
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(5):
            dq, dk, dv = _hip().attn_bwd(dout, q, k, v, o, lse, scale, True)
        torch.cuda.synchronize()
        print(f"attn_bwd x5: {(time.time()-t0)/5*1e3:.2f} ms")


if __name__ == "__main__":
    main()
