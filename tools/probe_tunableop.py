"""Does TunableOp beat the default hipBLASLt heuristics on the model's GEMM
shapes? (B6 S4096 llama-3-8B: M=24576, layers' N/K.)"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

shapes = [  # (M, N, K) fwd shapes + transposed bwd shapes
    (24576, 5120, 4096),   # qkv proj (32+2*8 heads * 128)
    (24576, 4096, 4096),   # o proj
    (24576, 28672, 4096),  # gate_up (2*14336)
    (24576, 4096, 14336),  # down
]

def bench(fn, iters=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters

def run(tag):
    total = 0
    for (M, N, K) in shapes:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        t = bench(lambda: a @ w.t())
        tf = 2 * M * N * K / t / 1e12
        total += t
        print(f"{tag} {M}x{N}x{K}: {t*1e3:7.3f} ms {tf:6.1f} TF/s", flush=True)
    print(f"{tag} total: {total*1e3:.3f} ms", flush=True)

run("default")
import torch.cuda.tunable as tun
tun.enable(True)
tun.tuning_enable(True)
tun.set_max_tuning_duration(500)
print("tunable on:", tun.is_enabled(), flush=True)
run("tuning")   # first pass tunes
tun.tuning_enable(False)
run("tuned")
