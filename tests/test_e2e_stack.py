"""End-to-end stack test: Data -> Train -> Serve on one cluster session
(the full library chain over the same runtime, CPU/gloo)."""
import time

import pytest


def test_data_train_serve_roundtrip(tmp_path_factory):
    import numpy as np
    import torch

    import ant_ray_amd as ray
    import ant_ray_amd.data as data
    from ant_ray_amd import serve, train
    from ant_ray_amd.train import Checkpoint, RunConfig, ScalingConfig
    from ant_ray_amd.train.torch import TorchTrainer

    if ray.is_initialized():
        ray.shutdown()
    ray.init(num_cpus=8)
    storage = str(tmp_path_factory.mktemp("e2e"))

    # ---- Data: synthesize a labeled dataset, transform it
    ds = data.range(512).map_batches(
        lambda b: {"x": b["id"].astype("float32") / 512.0,
                   "y": (b["id"] % 2).astype("float32")},
        batch_format="numpy")

    # ---- Train: 2-worker DDP on streaming shards, checkpoint the model
    def train_fn(config):
        import torch.nn as nn

        shard = train.get_dataset_shard("train")
        model = nn.Linear(1, 1)
        model = train.torch.prepare_model(model)
        opt = torch.optim.SGD(model.parameters(), lr=0.1)
        rows = 0
        for epoch in range(2):
            for batch in shard.iter_torch_batches(batch_size=32):
                x = batch["x"].unsqueeze(-1)
                y = batch["y"].unsqueeze(-1)
                loss = ((model(x) - y) ** 2).mean()
                opt.zero_grad()
                loss.backward()
                opt.step()
                rows += len(x)
        if train.get_context().get_world_rank() == 0:
            import tempfile

            with tempfile.TemporaryDirectory() as d:
                sd = {k: v.cpu() for k, v in model.state_dict().items()}
                torch.save(sd, f"{d}/model.pt")
                train.report({"rows": rows, "loss": float(loss)},
                             checkpoint=Checkpoint.from_directory(d))
        else:
            train.report({"rows": rows})

    result = TorchTrainer(
        train_fn,
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(name="e2e", storage_path=storage),
        datasets={"train": ds},
    ).fit()
    assert result.error is None
    assert result.checkpoint is not None
    assert result.metrics["rows"] > 0

    # ---- Serve: deploy the trained model from the checkpoint
    ckpt_path = result.checkpoint.path

    @serve.deployment
    class Scorer:
        def __init__(self, path):
            import torch.nn as nn

            self.model = nn.Linear(1, 1)
            sd = torch.load(f"{path}/model.pt", weights_only=False)
            self.model.load_state_dict(
                {k.replace("module.", ""): v for k, v in sd.items()})

        def __call__(self, x: float) -> float:
            with torch.no_grad():
                return float(self.model(torch.tensor([[x]])).item())

    h = serve.run(Scorer.bind(ckpt_path), name="scorer",
                  route_prefix="/score")
    preds = [h.remote(x).result(timeout_s=60) for x in (0.1, 0.9)]
    assert all(isinstance(p, float) for p in preds)

    # ---- state API sees the whole thing
    from ant_ray_amd.util import state

    assert any(a["state"] == "ALIVE" for a in state.list_actors())
    serve.shutdown()
    ray.shutdown()


def test_dataset_shard_epochs():
    """get_dataset_shard-style iterators re-execute the dataset per pass
    (epochs), staying row-equal across shards every epoch."""
    import ant_ray_amd as ray
    import ant_ray_amd.data as data

    if ray.is_initialized():
        ray.shutdown()
    ray.init(num_cpus=4)
    ds = data.range(100)
    s0, s1 = ds.streaming_split(2, equal=True)
    for epoch in range(3):
        rows0 = sum(len(b["id"]) for b in s0.iter_batches(batch_size=16))
        rows1 = sum(len(b["id"]) for b in s1.iter_batches(batch_size=16))
        assert rows0 == rows1 == 50, (epoch, rows0, rows1)
    ray.shutdown()
