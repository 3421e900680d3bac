"""runtime_env plugin tests: env_vars on pooled workers, working_dir /
py_modules on dedicated workers, offline pip rejection."""
import os

import pytest


@pytest.fixture(scope="module")
def ray_mod():
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()  # never inherit another module's (possibly dying) session
    if not ray.is_initialized():
        ray.init(num_cpus=8)
    yield ray
    ray.shutdown()


def test_env_vars(ray_mod):
    ray = ray_mod

    @ray.remote(runtime_env={"env_vars": {"ANTRAY_TEST_VAR": "hello42"}})
    def read_env():
        return os.environ.get("ANTRAY_TEST_VAR")

    assert ray.get(read_env.remote(), timeout=60) == "hello42"


def test_working_dir_and_py_modules(ray_mod, tmp_path):
    ray = ray_mod
    mod_dir = tmp_path / "mods"
    mod_dir.mkdir()
    (mod_dir / "my_rt_module.py").write_text("VALUE = 'from_py_modules'\n")
    wd = tmp_path / "wd"
    wd.mkdir()
    (wd / "marker.txt").write_text("working-dir-marker")

    @ray.remote(runtime_env={"working_dir": str(wd),
                             "py_modules": [str(mod_dir)]})
    def probe():
        import my_rt_module

        with open("marker.txt") as f:
            return my_rt_module.VALUE, f.read(), os.getcwd()

    val, marker, cwd = ray.get(probe.remote(), timeout=120)
    assert val == "from_py_modules"
    assert marker == "working-dir-marker"
    assert cwd == str(wd)


def test_actor_runtime_env(ray_mod, tmp_path):
    ray = ray_mod

    @ray.remote(runtime_env={"env_vars": {"ACTOR_RT": "yes"}})
    class A:
        def get(self):
            return os.environ.get("ACTOR_RT")

    a = A.remote()
    assert ray.get(a.get.remote(), timeout=60) == "yes"


def test_pip_rejected_offline():
    from ant_ray_amd._private.runtime_env import (
        RuntimeEnvSetupError,
        build_worker_spawn,
    )

    with pytest.raises(RuntimeEnvSetupError, match="air-gapped"):
        build_worker_spawn(["python"], {}, {"pip": ["requests"]})
    with pytest.raises(RuntimeEnvSetupError, match="unknown"):
        build_worker_spawn(["python"], {}, {"bogus_plugin": 1})


def test_runtime_env_class(ray_mod):
    """ray.runtime_env.RuntimeEnv public type: validates fields, passes
    through the normal runtime_env plumbing as a dict."""
    import pytest as _pt

    import ant_ray_amd as ray
    from ant_ray_amd.runtime_env import RuntimeEnv

    with _pt.raises(ValueError):
        RuntimeEnv(pip=["requests"])  # cloud-only: rejected up front
    with _pt.raises(ValueError):
        RuntimeEnv(bogus_field=1)
    with _pt.raises(TypeError):
        RuntimeEnv(env_vars={"A": 1})

    env = RuntimeEnv(env_vars={"MARKER_VAR": "via-class"})

    @ray.remote(runtime_env=env)
    def read():
        import os

        return os.environ.get("MARKER_VAR")

    assert ray.get(read.remote(), timeout=60) == "via-class"
