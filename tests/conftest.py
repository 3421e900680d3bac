import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


@pytest.fixture
def ray_local():
    import ant_ray_amd as ray

    ray.init(local_mode=True, ignore_reinit_error=True)
    yield ray
    ray.shutdown()


@pytest.fixture(scope="module")
def ray_start_regular_module():
    """Module-scoped real cluster (head subprocess + workers)."""
    import ant_ray_amd as ray

    ray.init(num_cpus=4)
    yield ray
    ray.shutdown()


@pytest.fixture
def ray_start_regular():
    import ant_ray_amd as ray

    if not ray.is_initialized():
        ray.init(num_cpus=4)
    yield ray
    ray.shutdown()
