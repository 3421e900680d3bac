"""FlatDDP correctness on CPU with gloo, world_size=2 (the distributed path
the driver's 8-GPU bench exercises, minus the RCCL transport)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from ant_ray_amd.models import build_model
        from ant_ray_amd.parallel import FlatAdamW, FlatDDP, FlatParamManager

        torch.manual_seed(7)  # same init on both ranks (then broadcast anyway)
        m = build_model("llama-tiny", device="cpu", seq_len=64)
        mgr = FlatParamManager(m)
        ddp = FlatDDP(m, manager=mgr, bucket_mb=1)
        opt = FlatAdamW(mgr, lr=1e-3)

        g = torch.Generator().manual_seed(100 + rank)
        tokens = torch.randint(0, 1024, (2, 64), generator=g)
        loss = ddp(tokens, tokens)
        loss.backward()
        ddp.finish_grad_sync()
        gradsum = mgr.flat_grad.float().sum().item()
        opt.step()
        psum = mgr.flat_param.float().sum().item()
        q.put((rank, gradsum, psum, float(loss)))
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4, 8])
def test_flatddp_grad_sync(world):
    """2/4/8-rank gloo: the rank counts the driver's round-end multi-GPU
    scaling bench uses (8-GPU readiness pre-staged on CPU — VERDICT r01
    item 5). bucket_mb=1 gives several buckets incl. a partial tail."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29511 + world
    procs = [ctx.Process(target=_worker, args=(r, world, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, gradsum, psum, loss = q.get(timeout=240)
        results[rank] = (gradsum, psum, loss)
    for p in procs:
        p.join(timeout=60)
    # after all-reduce every rank sees identical summed grads and params
    for r in range(1, world):
        assert results[0][0] == pytest.approx(results[r][0], rel=1e-5)
        assert results[0][1] == pytest.approx(results[r][1], rel=1e-6)
    # losses differ (different shards) -> the all-reduce really combined them
    assert results[0][2] != results[1][2]


def test_flatddp_matches_full_batch():
    """2-rank DDP step == single-process step on the concatenated batch."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29512
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, gradsum, psum, loss = q.get(timeout=240)
        results[rank] = (gradsum, psum, loss)
    for p in procs:
        p.join(timeout=60)

    from ant_ray_amd.models import build_model
    from ant_ray_amd.parallel import FlatAdamW, FlatParamManager

    torch.manual_seed(7)
    m = build_model("llama-tiny", device="cpu", seq_len=64)
    mgr = FlatParamManager(m)
    opt = FlatAdamW(mgr, lr=1e-3, world_size=1)
    batches = []
    for rank in range(2):
        g = torch.Generator().manual_seed(100 + rank)
        batches.append(torch.randint(0, 1024, (2, 64), generator=g))
    tokens = torch.cat(batches)
    loss = m(tokens, tokens)
    loss.backward()
    opt.step()
    psum = mgr.flat_param.float().sum().item()
    # bf16 grads: loose but meaningful agreement
    assert psum == pytest.approx(results[0][1], rel=1e-3)
