"""GPU data pipeline: map_batches on a GPU actor pool running HIP kernels,
batches landing on-device (BASELINE config 5 slice: Data -> Train input)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ray_mod():
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()
    ray.init(num_cpus=8, num_gpus=1)
    yield ray
    ray.shutdown()


def test_gpu_map_batches_pipeline(ray_mod):
    import ant_ray_amd.data as data

    class GpuNormalize:
        """Stateful GPU stage: rmsnorm each row on the MI355X."""

        def __init__(self):
            import ant_ray_amd.ops as ops

            assert torch.cuda.is_available(), "stage must own the GPU"
            assert ops.have_hip()
            self.w = torch.ones(256, device="cuda", dtype=torch.bfloat16)

        def __call__(self, batch):
            import ant_ray_amd.ops as ops

            x = torch.from_numpy(
                np.ascontiguousarray(batch["data"])).to(
                    "cuda", dtype=torch.bfloat16)
            y = ops.rmsnorm(x, self.w)
            batch["normed"] = y.float().cpu().numpy()
            return batch

    arr = np.random.rand(512, 256).astype(np.float32)
    ds = data.from_numpy(arr).map_batches(
        GpuNormalize, concurrency=1, num_gpus=1, batch_size=128)
    out = ds.take_batch(512)
    normed = out["normed"].reshape(512, 256)
    # numerics: matches a torch fp32 rmsnorm reference
    ref = (arr / np.sqrt((arr ** 2).mean(-1, keepdims=True) + 1e-5))
    assert np.abs(normed - ref).mean() < 2e-2

    # batches iterate straight onto the device
    got = next(iter(ds.iter_torch_batches(batch_size=64, device="cuda")))
    assert got["normed"].is_cuda and got["normed"].shape == (64, 256)
