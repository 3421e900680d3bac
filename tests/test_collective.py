"""ray.util.collective over actors (gloo on CPU; same path is RCCL on GPU)."""
import numpy as np
import pytest
import torch


@pytest.fixture(scope="module")
def ray_cluster():
    import ant_ray_amd as ray

    ray.init(num_cpus=4)
    yield ray
    ray.shutdown()


def _make_worker(ray):
    @ray.remote
    class ColWorker:
        def __init__(self, rank, world):
            self.rank = rank
            self.world = world

        def init_collective_group(self, world_size, rank, backend, group_name):
            from ant_ray_amd.util import collective as col

            col.init_collective_group(world_size, rank, backend, group_name)
            return True

        def do_allreduce(self, group_name="default"):
            from ant_ray_amd.util import collective as col

            t = torch.full((8,), float(self.rank + 1))
            col.allreduce(t, group_name)
            return t.numpy()

        def do_broadcast(self, group_name="default"):
            from ant_ray_amd.util import collective as col

            t = torch.full((4,), float(self.rank * 10))
            col.broadcast(t, src_rank=0, group_name=group_name)
            return t.numpy()

        def do_sendrecv(self, group_name="default"):
            from ant_ray_amd.util import collective as col

            if self.rank == 0:
                t = torch.arange(4, dtype=torch.float32)
                col.send(t, 1, group_name)
                return t.numpy()
            t = torch.zeros(4)
            col.recv(t, 0, group_name)
            return t.numpy()

        def do_allgather(self, group_name="default"):
            from ant_ray_amd.util import collective as col

            t = torch.full((2,), float(self.rank))
            out = [torch.zeros(2) for _ in range(self.world)]
            col.allgather(out, t, group_name)
            return [o.numpy() for o in out]

        def rank_of(self, group_name="default"):
            from ant_ray_amd.util import collective as col

            return col.get_rank(group_name)

    return ColWorker


def test_collective_group(ray_cluster):
    ray = ray_cluster
    from ant_ray_amd.util import collective as col

    ColWorker = _make_worker(ray)
    world = 2
    actors = [ColWorker.remote(r, world) for r in range(world)]
    col.create_collective_group(actors, world, list(range(world)),
                                backend="gloo", group_name="g1")
    # allreduce: 1 + 2 = 3
    out = ray.get([a.do_allreduce.remote("g1") for a in actors])
    for o in out:
        assert np.allclose(o, 3.0)
    # broadcast from rank 0
    out = ray.get([a.do_broadcast.remote("g1") for a in actors])
    for o in out:
        assert np.allclose(o, 0.0)
    # send/recv
    out = ray.get([a.do_sendrecv.remote("g1") for a in actors])
    assert np.allclose(out[1], np.arange(4))
    # allgather
    out = ray.get([a.do_allgather.remote("g1") for a in actors])
    assert np.allclose(out[0][0], 0.0) and np.allclose(out[0][1], 1.0)
    # ranks
    assert ray.get([a.rank_of.remote("g1") for a in actors]) == [0, 1]
