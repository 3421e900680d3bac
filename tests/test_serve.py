"""Ray Serve parity tests: deployments, handles, composition, HTTP proxy,
batching, autoscaling policy math, status/delete."""
import asyncio
import time

import pytest


@pytest.fixture(scope="module")
def serve_mod():
    import ant_ray_amd as ray
    from ant_ray_amd import serve

    if ray.is_initialized():
        ray.shutdown()  # never inherit another module's (possibly dying) session
    if not ray.is_initialized():
        ray.init(num_cpus=8)
    yield serve
    serve.shutdown()
    ray.shutdown()


def test_deploy_and_handle_call(serve_mod):
    serve = serve_mod

    @serve.deployment(num_replicas=2)
    class Doubler:
        def __call__(self, x):
            return 2 * x

        def triple(self, x):
            return 3 * x

    h = serve.run(Doubler.bind(), name="doubler", route_prefix="/doubler")
    assert h.remote(21).result(timeout_s=30) == 42
    assert h.triple.remote(5).result(timeout_s=30) == 15
    # load-balances over 2 replicas without error
    assert [h.remote(i).result() for i in range(10)] == [2 * i for i in range(10)]


def test_function_deployment_and_composition(serve_mod):
    serve = serve_mod

    @serve.deployment
    def magnitude(x):
        return abs(x)

    @serve.deployment
    class Pipeline:
        def __init__(self, mag_handle):
            self.mag = mag_handle

        async def __call__(self, x):
            return await self.mag.remote(x) + 1

    h = serve.run(Pipeline.bind(magnitude.bind()), name="pipe",
                  route_prefix="/pipe")
    assert h.remote(-41).result(timeout_s=30) == 42


def test_http_proxy(serve_mod):
    serve = serve_mod
    import ant_ray_amd as ray

    @serve.deployment
    class Echo:
        async def __call__(self, request):
            body = await request.body()
            return {"path": request.url.path, "body": body.decode(),
                    "method": request.method}

    serve.run(Echo.bind(), name="echo", route_prefix="/echo")
    port = ray.get(ray.get_actor("SERVE_PROXY_ACTOR").ready.remote())
    import urllib.request

    req = urllib.request.Request(f"http://127.0.0.1:{port}/echo/sub?q=1",
                                 data=b"hello", method="POST")
    with urllib.request.urlopen(req, timeout=30) as resp:
        import json

        out = json.loads(resp.read())
    assert out == {"path": "/sub", "body": "hello", "method": "POST"}
    # healthz + routes endpoints
    with urllib.request.urlopen(f"http://127.0.0.1:{port}/-/healthz",
                                timeout=10) as r:
        assert r.read() == b"ok"


def test_starlette_response_passthrough(serve_mod):
    serve = serve_mod
    import ant_ray_amd as ray

    @serve.deployment
    class Custom:
        def __call__(self, request):
            from starlette.responses import PlainTextResponse

            return PlainTextResponse("teapot", status_code=418)

    serve.run(Custom.bind(), name="custom", route_prefix="/custom")
    port = ray.get(ray.get_actor("SERVE_PROXY_ACTOR").ready.remote())
    import urllib.error
    import urllib.request

    try:
        urllib.request.urlopen(f"http://127.0.0.1:{port}/custom", timeout=30)
        assert False, "expected 418"
    except urllib.error.HTTPError as e:
        assert e.code == 418
        assert e.read() == b"teapot"


def test_batching(serve_mod):
    serve = serve_mod

    @serve.deployment
    class Batched:
        def __init__(self):
            self.batch_sizes = []

        @serve.batch(max_batch_size=8, batch_wait_timeout_s=0.2)
        async def __call__(self, items):
            self.batch_sizes.append(len(items))
            return [i * 10 for i in items]

        def get_sizes(self):
            return self.batch_sizes

    h = serve.run(Batched.bind(), name="batched", route_prefix="/batched")
    responses = [h.remote(i) for i in range(8)]
    results = [r.result(timeout_s=30) for r in responses]
    assert sorted(results) == [i * 10 for i in range(8)]
    sizes = h.get_sizes.remote().result(timeout_s=30)
    assert max(sizes) > 1  # at least some calls were actually batched


def test_status_and_delete(serve_mod):
    serve = serve_mod

    @serve.deployment
    def f():
        return "ok"

    serve.run(f.bind(), name="tmp_app", route_prefix="/tmp")
    st = serve.status()["applications"]
    assert "tmp_app" in st
    assert st["tmp_app"]["deployments"]["f"]["replicas"] == 1
    serve.delete("tmp_app")
    st = serve.status()["applications"]
    assert "tmp_app" not in st


def test_autoscaling_policy_math():
    from ant_ray_amd.serve._private.common import AutoscalingConfig
    from ant_ray_amd.serve._private.controller import (
        _calculate_desired_num_replicas,
    )

    cfg = AutoscalingConfig(min_replicas=1, max_replicas=10,
                            target_ongoing_requests=2.0)
    assert _calculate_desired_num_replicas(cfg, total_ongoing=0, current=2) == 1
    assert _calculate_desired_num_replicas(cfg, total_ongoing=8, current=2) == 4
    assert _calculate_desired_num_replicas(cfg, total_ongoing=100, current=2) == 10
    assert _calculate_desired_num_replicas(cfg, total_ongoing=3, current=2) == 2


def test_multiplexed(serve_mod):
    serve = serve_mod

    @serve.deployment
    class MultiModel:
        def __init__(self):
            self.loads = []

        @serve.multiplexed(max_num_models_per_replica=2)
        async def get_model(self, model_id: str):
            self.loads.append(model_id)
            return f"model:{model_id}"

        async def __call__(self, model_id: str):
            m = await self.get_model(model_id)
            return {"model": m, "loads": list(self.loads)}

    h = serve.run(MultiModel.bind(), name="mux", route_prefix="/mux")
    r1 = h.remote("a").result(timeout_s=30)
    r2 = h.remote("a").result(timeout_s=30)
    assert r2["model"] == "model:a" and r2["loads"].count("a") == 1  # cached
    h.remote("b").result(timeout_s=30)
    out = h.remote("c").result(timeout_s=30)  # evicts LRU ("a")
    out = h.remote("a").result(timeout_s=30)  # reloads a
    assert out["loads"].count("a") == 2


def test_serve_cli(serve_mod, tmp_path, monkeypatch):
    """serve CLI: run (non-blocking) an import-path app, status, delete."""
    from click.testing import CliRunner

    from ant_ray_amd.serve.scripts import cli

    app_mod = tmp_path / "cli_app_mod.py"
    app_mod.write_text(
        "from ant_ray_amd import serve\n"
        "@serve.deployment\n"
        "def hello(request=None):\n"
        "    return 'hi-from-cli'\n"
        "app = hello.bind()\n"
    )
    monkeypatch.chdir(tmp_path)
    import sys

    # earlier module tests leave their apps (and 1-CPU replicas) running;
    # clear them so this app's replica can schedule in the 8-CPU session
    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    sys.path.insert(0, str(tmp_path))
    try:
        r = CliRunner().invoke(cli, [
            "run", "cli_app_mod:app", "--name", "cliapp",
            "--route-prefix", "/cliapp", "--non-blocking"])
        assert r.exit_code == 0, r.output
        assert "deployed" in r.output

        r = CliRunner().invoke(cli, ["status"])
        assert r.exit_code == 0, r.output
        assert "cliapp" in r.output

        import urllib.request

        body = urllib.request.urlopen(
            "http://127.0.0.1:8000/cliapp", timeout=30).read().decode()
        assert "hi-from-cli" in body

        r = CliRunner().invoke(cli, ["delete", "cliapp"])
        assert r.exit_code == 0, r.output
    finally:
        sys.path.remove(str(tmp_path))


def test_streaming_response(serve_mod):
    """A generator ingress streams chunks: via handle.options(stream=True)
    AND over HTTP with chunked transfer — first chunk arrives while the
    replica is still producing."""
    import time as _t
    import urllib.request

    import ant_ray_amd as ray

    serve = serve_mod

    @serve.deployment
    def streamer(request=None):
        for i in range(4):
            yield f"chunk{i}|"
            _t.sleep(0.3)

    serve.run(streamer.bind(), name="streamapp", route_prefix="/stream")

    # handle-level streaming
    h = serve.get_app_handle("streamapp")
    chunks = list(h.options(stream=True).remote())
    assert chunks == [f"chunk{i}|" for i in range(4)]

    # HTTP chunked: read incrementally, first chunk must arrive early
    port = ray.get(ray.get_actor("SERVE_PROXY_ACTOR").ready.remote())
    t0 = _t.time()
    with urllib.request.urlopen(f"http://127.0.0.1:{port}/stream",
                                timeout=30) as r:
        first = r.read(7)
        first_latency = _t.time() - t0
        rest = r.read()
    assert first == b"chunk0|"
    assert rest == b"chunk1|chunk2|chunk3|"
    assert first_latency < 1.0, f"first chunk took {first_latency:.2f}s"
    serve.delete("streamapp")


def test_serve_deploy_yaml(serve_mod, tmp_path, monkeypatch):
    """serve deploy <config.yaml>: multi-app declarative deploy."""
    from click.testing import CliRunner

    from ant_ray_amd.serve.scripts import cli

    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    (tmp_path / "yaml_apps.py").write_text(
        "from ant_ray_amd import serve\n"
        "@serve.deployment\n"
        "def alpha(request=None):\n"
        "    return 'from-alpha'\n"
        "@serve.deployment\n"
        "def beta(request=None):\n"
        "    return 'from-beta'\n"
        "app_a = alpha.bind()\n"
        "app_b = beta.bind()\n"
    )
    (tmp_path / "serve_config.yaml").write_text(
        "applications:\n"
        "  - name: appa\n"
        "    route_prefix: /a\n"
        "    import_path: yaml_apps:app_a\n"
        "  - name: appb\n"
        "    route_prefix: /b\n"
        "    import_path: yaml_apps:app_b\n"
    )
    monkeypatch.chdir(tmp_path)
    import sys

    sys.path.insert(0, str(tmp_path))
    try:
        r = CliRunner().invoke(cli, ["deploy", "serve_config.yaml"])
        assert r.exit_code == 0, r.output
        import urllib.request

        a = urllib.request.urlopen("http://127.0.0.1:8000/a",
                                   timeout=30).read().decode()
        b = urllib.request.urlopen("http://127.0.0.1:8000/b",
                                   timeout=30).read().decode()
        assert a == "from-alpha" and b == "from-beta"
        serve.delete("appa")
        serve.delete("appb")
    finally:
        sys.path.remove(str(tmp_path))


def test_autoscaling_e2e(serve_mod):
    """Request-rate autoscaling end to end: concurrent slow requests grow
    the replica set above min; draining traffic shrinks it back."""
    import threading

    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    @serve.deployment(autoscaling_config={
        "min_replicas": 1, "max_replicas": 3,
        "target_ongoing_requests": 1.0,
        "upscale_delay_s": 0.2, "downscale_delay_s": 0.5,
    }, max_ongoing_requests=10)
    class Slow:
        async def __call__(self, x):
            import asyncio as _a

            await _a.sleep(0.4)
            return x

    h = serve.run(Slow.bind(), name="auto", route_prefix="/auto")

    stop = threading.Event()

    def hammer():
        while not stop.is_set():
            try:
                h.remote(1).result(timeout_s=30)
            except Exception:
                pass

    threads = [threading.Thread(target=hammer, daemon=True)
               for _ in range(8)]
    for t in threads:
        t.start()
    try:
        deadline = time.time() + 60
        grew = False
        while time.time() < deadline:
            st = serve.status()["applications"]["auto"]["deployments"]["Slow"]
            if st["replicas"] > 1:
                grew = True
                break
            time.sleep(0.5)
        assert grew, "autoscaler never scaled up under load"
    finally:
        stop.set()
        for t in threads:
            t.join(timeout=10)

    deadline = time.time() + 60
    shrunk = False
    while time.time() < deadline:
        st = serve.status()["applications"]["auto"]["deployments"]["Slow"]
        if st["replicas"] == 1:
            shrunk = True
            break
        time.sleep(0.5)
    assert shrunk, "autoscaler never scaled back down"
    serve.delete("auto")


def test_sync_handlers_run_concurrently(serve_mod):
    """A BLOCKING sync handler must not serialize the replica: concurrent
    requests overlap on the replica's thread pool (the reference runs
    sync handlers off the event loop)."""
    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    @serve.deployment(max_ongoing_requests=8)
    class Blocking:
        def __call__(self, x):
            import time as _t

            _t.sleep(0.5)
            return x

    h = serve.run(Blocking.bind(), name="blocky", route_prefix="/blocky")
    t0 = time.time()
    rs = [h.remote(i) for i in range(6)]
    out = sorted(r.result(timeout_s=60) for r in rs)
    dt = time.time() - t0
    assert out == list(range(6))
    assert dt < 2.0, f"6x0.5s blocking requests took {dt:.2f}s (serialized)"
    serve.delete("blocky")


def test_controller_crash_recovery(serve_mod):
    """The Serve controller checkpoints app configs to the GCS KV; after
    its process is SIGKILLed, the restarted controller (max_restarts=-1)
    recovers every app and fresh replicas serve traffic (reference
    _recover_state_from_checkpoint). The dead incarnation's replicas die
    with their owner (actor owner fate-sharing)."""
    import os
    import signal

    import ant_ray_amd as ray

    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    @serve.deployment
    def resilient(request=None):
        return "still-here"

    h = serve.run(resilient.bind(), name="recov", route_prefix="/recov")
    assert h.remote(None).result(timeout_s=30) == "still-here"

    controller = ray.get_actor("SERVE_CONTROLLER_ACTOR")
    pid = ray.get(controller.getpid.remote(), timeout=30)
    os.kill(pid, signal.SIGKILL)

    deadline = time.time() + 90
    ok = False
    while time.time() < deadline:
        try:
            apps = serve.status().get("applications", {})
            if "recov" in apps and \
                    apps["recov"]["deployments"]["resilient"]["replicas"] >= 1:
                h2 = serve.get_app_handle("recov")
                if h2.remote(None).result(timeout_s=20) == "still-here":
                    ok = True
                    break
        except Exception:
            pass
        time.sleep(1.0)
    assert ok, "controller did not recover the app after SIGKILL"
    serve.delete("recov")


def test_grpc_ingress(serve_mod):
    """gRPC proxy parity: a stub-less unary call to
    /ray.serve.UserApplicationService/<app> reaches the deployment."""
    import pickle

    import grpc

    import ant_ray_amd as ray

    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    @serve.deployment
    def echo_upper(payload):
        return {"upper": payload["text"].upper()}

    serve.run(echo_upper.bind(), name="grpcapp", route_prefix="/grpcapp")
    from ant_ray_amd.serve._private.grpc_proxy import start_grpc_proxy

    port = start_grpc_proxy(port=0)
    assert port > 0

    ch = grpc.insecure_channel(f"127.0.0.1:{port}")
    call = ch.unary_unary(
        "/ray.serve.UserApplicationService/grpcapp",
        request_serializer=None, response_deserializer=None)
    reply = call(pickle.dumps({"text": "grpc works"}), timeout=60)
    assert pickle.loads(reply) == {"upper": "GRPC WORKS"}

    # unknown app -> NOT_FOUND
    bad = ch.unary_unary("/ray.serve.UserApplicationService/nope",
                         request_serializer=None, response_deserializer=None)
    import pytest as _pt

    with _pt.raises(grpc.RpcError):
        bad(b"x", timeout=30)
    ch.close()
    serve.delete("grpcapp")


def test_multiplexed_model_affinity_routing(serve_mod):
    """handle.options(multiplexed_model_id=...): repeated calls for one
    model stick to one replica (each model loads on exactly one of the
    replicas instead of everywhere)."""
    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    @serve.deployment(num_replicas=2)
    class MultiModel:
        def __init__(self):
            self.loaded = set()

        @serve.multiplexed(max_num_models_per_replica=4)
        async def get_model(self, model_id: str):
            self.loaded.add(model_id)
            return f"model:{model_id}"

        async def __call__(self, model_id: str):
            import os

            await self.get_model(model_id)
            return (os.getpid(), sorted(self.loaded))

    h = serve.run(MultiModel.bind(), name="muxaff", route_prefix="/muxaff")
    pids_a = {h.options(multiplexed_model_id="a").remote("a")
              .result(timeout_s=30)[0] for _ in range(6)}
    pids_b = {h.options(multiplexed_model_id="b").remote("b")
              .result(timeout_s=30)[0] for _ in range(6)}
    assert len(pids_a) == 1, "model 'a' requests should stick to one replica"
    assert len(pids_b) == 1
    serve.delete("muxaff")


def test_graceful_replica_shutdown(serve_mod):
    """Scaling down / deleting drains in-flight requests before killing
    replicas: a slow request completes across the delete."""
    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    @serve.deployment(graceful_shutdown_timeout_s=30)
    class SlowFinish:
        async def __call__(self, x):
            import asyncio as _a

            await _a.sleep(2.0)
            return f"done-{x}"

    h = serve.run(SlowFinish.bind(), name="drainapp", route_prefix="/drain")
    resp = h.remote(1)  # in flight...
    time.sleep(0.3)
    serve.delete("drainapp", _blocking=False)  # removal starts mid-request
    assert resp.result(timeout_s=60) == "done-1"


def test_unhealthy_replica_replaced(serve_mod):
    """A replica whose user check_health starts failing is killed and
    replaced by the controller's periodic health loop."""
    import os

    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    poison = "/tmp/antray_unhealthy_marker"
    try:
        os.unlink(poison)
    except FileNotFoundError:
        pass

    @serve.deployment(health_check_period_s=1.0)
    class Fragile:
        def __call__(self, x):
            import os as _os

            return _os.getpid()

        def check_health(self):
            import os as _os

            if _os.path.exists(poison):
                raise RuntimeError("simulated sickness")
            return True

    h = serve.run(Fragile.bind(), name="fragile", route_prefix="/fragile")
    pid1 = h.remote(1).result(timeout_s=60)
    open(poison, "w").close()  # every current replica now reports sick
    time.sleep(2.5)
    os.unlink(poison)  # replacements come up healthy
    deadline = time.time() + 60
    pid2 = None
    while time.time() < deadline:
        try:
            pid2 = h.remote(1).result(timeout_s=20)
            if pid2 != pid1:
                break
        except Exception:
            pass
        time.sleep(0.5)
    assert pid2 is not None and pid2 != pid1, (pid1, pid2)
    serve.delete("fragile")


def test_async_check_health(serve_mod):
    """An ASYNC user check_health must be awaited, not run_until_complete
    (which explodes inside the replica's running loop)."""
    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    @serve.deployment(health_check_period_s=0.5)
    class AsyncHealthy:
        async def check_health(self):
            import asyncio as _a

            await _a.sleep(0.01)
            return True

        def __call__(self, x):
            return x + 1

    h = serve.run(AsyncHealthy.bind(), name="ahealth", route_prefix="/ah")
    assert h.remote(1).result(timeout_s=60) == 2
    time.sleep(2.0)  # several health periods pass without replica churn
    assert h.remote(2).result(timeout_s=60) == 3
    st = serve.status()["applications"]["ahealth"]["deployments"]
    assert st["AsyncHealthy"]["replicas"] == 1
    serve.delete("ahealth")


def test_router_max_ongoing_backpressure(serve_mod):
    """max_ongoing_requests is enforced at the ROUTER: with one replica
    capped at 2, a burst of 6 requests never has more than 2 in flight
    replica-side."""
    serve = serve_mod
    for app_name in list(serve.status().get("applications", {})):
        serve.delete(app_name)

    @serve.deployment(max_ongoing_requests=2)
    class Gauged:
        def __init__(self):
            self.inflight = 0
            self.peak = 0

        async def __call__(self, x):
            import asyncio as _a

            self.inflight += 1
            self.peak = max(self.peak, self.inflight)
            await _a.sleep(0.3)
            self.inflight -= 1
            return self.peak

    import threading

    h = serve.run(Gauged.bind(), name="gauged", route_prefix="/gauged")
    results = []

    def call():
        results.append(h.remote(0).result(timeout_s=60))

    ts = [threading.Thread(target=call) for _ in range(6)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=90)
    assert len(results) == 6
    assert max(results) <= 2, f"router admitted beyond the cap: {results}"
    serve.delete("gauged")


def test_request_timeout(serve_mod):
    """http_options request_timeout_s: slow requests get 408 at the proxy
    instead of holding the connection open forever."""
    import subprocess
    import sys
    import textwrap

    # separate session: the module fixture's proxy has no timeout
    script = textwrap.dedent("""
        import urllib.error
        import urllib.request

        import ant_ray_amd as ray
        from ant_ray_amd import serve

        ray.init(num_cpus=4)
        serve.start(http_options={"port": 8899, "request_timeout_s": 1.0})

        @serve.deployment
        class Sluggish:
            async def __call__(self, request):
                import asyncio

                await asyncio.sleep(10)
                return "late"

        serve.run(Sluggish.bind(), name="slug", route_prefix="/slug")
        try:
            urllib.request.urlopen("http://127.0.0.1:8899/slug", timeout=30)
            print("NO_TIMEOUT")
        except urllib.error.HTTPError as e:
            print("STATUS", e.code)
        serve.shutdown()
        ray.shutdown()
    """)
    out = subprocess.run([sys.executable, "-c", script], capture_output=True,
                         text=True, timeout=180, cwd="/root/repo")
    assert "STATUS 408" in out.stdout, out.stdout[-800:] + out.stderr[-800:]


def test_replica_context_and_run_many(serve_mod):
    """get_replica_context inside replicas; run_many deploys several apps;
    HTTPOptions accepted by start (parity: serve/context.py:37,
    serve/api.py:520/645)."""
    serve = serve_mod

    @serve.deployment(num_replicas=2)
    class WhoAmI:
        def __init__(self):
            ctx = serve.get_replica_context()
            self.boot_tag = ctx.replica_tag

        def __call__(self, _):
            ctx = serve.get_replica_context()
            assert ctx.servable_object is self
            return {"app": ctx.app_name, "dep": ctx.deployment,
                    "tag": ctx.replica_tag, "rank": ctx.rank,
                    "ws": ctx.world_size, "boot": self.boot_tag}

    @serve.deployment
    def plain(_):
        return "plain-ok"

    # outside a replica -> error
    with pytest.raises(RuntimeError):
        serve.get_replica_context()

    h1, h2 = serve.run_many([
        serve.RunTarget(WhoAmI.bind(), name="who"),
        serve.RunTarget(plain.bind(), name="plain_app"),
    ])
    seen = {h1.remote(None).result(timeout_s=30)["tag"] for _ in range(10)}
    one = h1.remote(None).result(timeout_s=30)
    assert one["app"] == "who" and one["dep"] == "WhoAmI"
    assert one["ws"] == 2 and one["boot"] == one["tag"]
    assert len(seen) == 2  # both ranks answered across 10 calls
    assert h2.remote(None).result(timeout_s=30) == "plain-ok"
    serve.delete("who")
    serve.delete("plain_app")


def test_serve_cli_build_and_config(serve_mod, tmp_path, monkeypatch):
    """serve build generates a deployable YAML; serve config prints the
    live state (parity: reference serve CLI build/config)."""
    import sys

    import yaml
    from click.testing import CliRunner

    from ant_ray_amd.serve.scripts import cli

    serve = serve_mod
    app_mod = tmp_path / "cli_bc_mod.py"
    app_mod.write_text(
        "from ant_ray_amd import serve\n"
        "@serve.deployment\n"
        "def ping(request=None):\n"
        "    return 'pong'\n"
        "app = ping.bind()\n"
    )
    monkeypatch.chdir(tmp_path)
    sys.path.insert(0, str(tmp_path))
    try:
        out_yaml = tmp_path / "built.yaml"
        r = CliRunner().invoke(cli, ["build", "cli_bc_mod:app",
                                     "-o", str(out_yaml)])
        assert r.exit_code == 0, r.output
        doc = yaml.safe_load(out_yaml.read_text())
        assert doc["applications"][0]["import_path"] == "cli_bc_mod:app"

        for app_name in list(serve.status().get("applications", {})):
            serve.delete(app_name)
        r = CliRunner().invoke(cli, ["deploy", str(out_yaml)])
        assert r.exit_code == 0, r.output

        r = CliRunner().invoke(cli, ["config"])
        assert r.exit_code == 0, r.output
        cfg = yaml.safe_load(r.output)
        names = [a["name"] for a in cfg["applications"]]
        assert "default" in names
        serve.delete("default")
    finally:
        sys.path.remove(str(tmp_path))
