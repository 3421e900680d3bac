"""The in-tree native extensions must be complete and loadable on CPU.

Guards against the failure mode where a partial rebuild relinks
`_hip_ops` without some kernel objects: the .so links fine but dies at
import time on the GPU box with `undefined symbol: launch_*`. Importing
here (torch first, so libtorch is resolvable) catches that on every
driver CPU run.
"""
import pytest


def test_hip_ops_complete():
    torch = pytest.importorskip("torch")  # noqa: F841

    import ant_ray_amd._hip_ops as m

    for fn in ("rmsnorm_fwd", "rmsnorm_bwd", "rope_", "swiglu_fwd",
               "swiglu_bwd", "adamw_", "cross_entropy_fwd_bwd",
               "attn_fwd", "attn_bwd"):
        assert hasattr(m, fn), f"_hip_ops is missing {fn} — stale/partial link"


def test_shm_store_loads():
    from ant_ray_amd._shm_store import ShmStore  # noqa: F401


def test_gpu_ipc_loads():
    pytest.importorskip("torch")
    import ant_ray_amd._gpu_ipc  # noqa: F401


def test_ray_alias_deep_imports():
    """Arbitrary-depth `ray.*` imports resolve to ant_ray_amd modules via
    the alias meta-path finder (user code written for the reference works
    unchanged)."""
    import importlib

    import ray

    for name in ["ray.runtime_context", "ray.types", "ray.cross_language",
                 "ray.job_config", "ray.workflow", "ray.air",
                 "ray.serve.handle", "ray.data.datasource",
                 "ray.tune.stopper", "ray.util.debug",
                 "ray._private.serialization"]:
        mod = importlib.import_module(name)
        assert mod.__name__.startswith("ant_ray_amd"), (name, mod.__name__)
    from ray.job_config import JobConfig

    assert JobConfig(ray_namespace="n").serialize()["ray_namespace"] == "n"
    assert ray.types.ObjectRef is not None


def test_experimental_namespace():
    """ray.experimental parity members: tqdm_ray, compiled_dag_ref, queue,
    object locations (reference python/ray/experimental/)."""
    import io

    import ray
    from ray.experimental import tqdm_ray
    from ray.experimental.compiled_dag_ref import CompiledDAGRef  # noqa: F401
    from ray.experimental.queue import Empty, Queue  # noqa: F401

    bar = tqdm_ray.tqdm(total=5, desc="x")
    for _ in range(5):
        bar.update(1)
    bar.set_description("done")
    bar.close()
    tqdm_ray.safe_print("safe")
    assert list(tqdm_ray.tqdm(iterable=[1, 2, 3])) == [1, 2, 3]
    tqdm_ray.instance().unhide_bars()
