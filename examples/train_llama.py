"""Ray Train: Llama-3-8B DDP on N MI355X GPUs (1 worker actor per GPU).

    python examples/train_llama.py --workers 8 --steps 20
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ray
from ray.train import RunConfig, ScalingConfig
from ray.train.torch import TorchTrainer


def train_fn(config):
    import torch

    from ant_ray_amd import train
    from ant_ray_amd.models import build_model
    from ant_ray_amd.parallel import FlatAdamW, FlatDDP, FlatParamManager

    device = train.torch.get_device()
    torch.cuda.set_device(device)
    model = build_model(config["model"], device=str(device),
                        seq_len=config["seq"])
    mgr = FlatParamManager(model, device=device)
    ddp = FlatDDP(model, manager=mgr, bucket_mb=64)
    opt = FlatAdamW(mgr, lr=3e-4, weight_decay=0.1)
    vocab = model.cfg.vocab
    for step in range(config["steps"]):
        tokens = torch.randint(0, vocab, (config["batch"], config["seq"]),
                               device=device)
        loss = ddp(tokens, tokens)
        loss.backward()
        ddp.finish_grad_sync()
        opt.step()
        opt.zero_grad()
        train.report({"step": step, "loss": float(loss)})


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--batch", type=int, default=6)
    ap.add_argument("--seq", type=int, default=4096)
    args = ap.parse_args()
    ray.init()
    result = TorchTrainer(
        train_fn,
        train_loop_config=vars(args) | {"steps": args.steps},
        scaling_config=ScalingConfig(num_workers=args.workers, use_gpu=True),
        run_config=RunConfig(name="llama_ddp"),
    ).fit()
    print("final:", result.metrics)
