"""GCS fault tolerance: durable tables survive a GCS restart; raylets
reconnect and re-register instead of fate-sharing.

Role parity: reference GCS restartable with Redis persistence
(gcs/store_client/redis_store_client.cc), raylets notified via
NotifyGCSRestart (node_manager.proto:446); here the store client is a
file snapshot (no Redis in this image) and raylets reconnect on
connection loss (raylet.py _reconnect_gcs)."""
import os
import signal
import socket
import subprocess
import sys
import time

import pytest


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _spawn_gcs(port, persist):
    return subprocess.Popen(
        [sys.executable, "-m", "ant_ray_amd._private.gcs",
         "--host", "127.0.0.1", "--port", str(port), "--persist", persist],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)


def _wait_port(port, timeout=30):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            socket.create_connection(("127.0.0.1", port), timeout=1).close()
            return True
        except OSError:
            time.sleep(0.2)
    return False


def test_gcs_restart_with_persistence(tmp_path):
    port = _free_port()
    persist = str(tmp_path / "gcs_tables.msgpack")
    session = str(tmp_path / "session")
    os.makedirs(session, exist_ok=True)
    store = str(tmp_path / "store_shm")

    gcs = _spawn_gcs(port, persist)
    raylet = None
    try:
        assert _wait_port(port)
        raylet = subprocess.Popen(
            [sys.executable, "-m", "ant_ray_amd._private.raylet",
             "--gcs", f"127.0.0.1:{port}", "--num-cpus", "2",
             "--num-gpus", "0", "--store-path", store,
             "--store-capacity", str(256 * 1024 * 1024),
             "--session-dir", session],
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
            env={**os.environ, "ANTRAY_GCS_RECONNECT_TIMEOUT_S": "60"})

        import ant_ray_amd as ray

        if ray.is_initialized():
            ray.shutdown()
        deadline = time.time() + 30
        while True:
            try:
                ray.init(address=f"127.0.0.1:{port}")
                break
            except Exception:
                if time.time() > deadline:
                    raise
                time.sleep(0.5)

        from ant_ray_amd.experimental import internal_kv

        internal_kv._internal_kv_put(b"durable_key", b"durable_value")

        from ant_ray_amd.util import virtual_cluster as vc

        node_id = [n["NodeID"] for n in ray.nodes()][0]
        vc.create_or_update_virtual_cluster("vc_persist", node_ids=[node_id])

        @ray.remote
        class Keeper:
            def __init__(self):
                self.v = "alive-across-gcs-restart"

            def get(self):
                return self.v

        k = Keeper.options(name="keeper", lifetime="detached").remote()
        assert ray.get(k.get.remote(), timeout=60) == "alive-across-gcs-restart"
        time.sleep(0.5)  # let the debounced snapshot land
        assert os.path.exists(persist)

        # ---- kill the GCS hard, restart it on the same port; the LIVE
        # driver must reconnect on its own (worker._reconnect_gcs)
        gcs.send_signal(signal.SIGKILL)
        gcs.wait(timeout=10)
        gcs = _spawn_gcs(port, persist)
        assert _wait_port(port)

        deadline = time.time() + 60
        while True:
            try:
                assert internal_kv._internal_kv_get(b"durable_key") == \
                    b"durable_value"
                break
            except Exception:
                if time.time() > deadline:
                    raise
                time.sleep(0.5)
        ray.shutdown()

        # fresh driver: KV restored, named detached actor reachable, and
        # the raylet re-registered (its worker kept the actor alive)
        deadline = time.time() + 60
        while True:
            try:
                ray.init(address=f"127.0.0.1:{port}")
                break
            except Exception:
                if time.time() > deadline:
                    raise
                time.sleep(0.5)
        assert internal_kv._internal_kv_get(b"durable_key") == b"durable_value"

        from ant_ray_amd.util import virtual_cluster as vc

        vcs = {v["virtual_cluster_id"] for v in vc.list_virtual_clusters()}
        assert "vc_persist" in vcs, "virtual cluster lost across GCS restart"

        deadline = time.time() + 60
        nodes = []
        while time.time() < deadline:
            nodes = [n for n in ray.nodes() if n.get("Alive")]
            if nodes:
                break
            time.sleep(0.5)
        assert nodes, "raylet should re-register with the restarted GCS"

        k2 = ray.get_actor("keeper")
        assert ray.get(k2.get.remote(), timeout=60) == "alive-across-gcs-restart"
        ray.shutdown()
    finally:
        for proc in (gcs, raylet):
            if proc is not None and proc.poll() is None:
                proc.kill()


def test_head_fate_shares_with_driver(tmp_path):
    """A head started by ray.init must die when its driver is SIGKILLed
    (no orphaned head+worker process trees); `ray start --head` stays
    detached (owner_pid=0)."""
    import textwrap

    script = textwrap.dedent("""
        import os, sys, time
        import ant_ray_amd as ray

        ray.init(num_cpus=2)
        from ant_ray_amd._private.worker import global_worker

        # print the head pid for the parent test, then hang until killed
        import subprocess
        print("SESSION", global_worker.core_worker.session_dir, flush=True)
        sys.stdout.flush()
        time.sleep(300)
    """)
    proc = subprocess.Popen([sys.executable, "-c", script],
                            stdout=subprocess.PIPE, text=True)
    try:
        line = proc.stdout.readline()
        assert line.startswith("SESSION"), line
        session_dir = line.split(" ", 1)[1].strip()

        # find the head process serving that session
        out = subprocess.run(["ps", "-ww", "-eo", "pid,args"], capture_output=True,
                             text=True).stdout
        head_pid = None
        for row in out.splitlines():
            if "_private.head" in row and session_dir in row:
                head_pid = int(row.split()[0])
                break
        assert head_pid is not None, f"no head for {session_dir}"

        proc.kill()  # SIGKILL the driver: no ray.shutdown runs
        proc.wait(timeout=10)

        deadline = time.time() + 30
        alive = True
        while time.time() < deadline:
            try:
                os.kill(head_pid, 0)
            except ProcessLookupError:
                alive = False
                break
            time.sleep(0.5)
        assert not alive, "orphaned head survived its driver's death"
    finally:
        if proc.poll() is None:
            proc.kill()


def test_driver_death_cleans_up_actors(tmp_path):
    """A SIGKILLed driver's non-detached actors die with it (owner
    fate-sharing on driver disconnect); detached actors survive."""
    import textwrap

    from ant_ray_amd.cluster_utils import Cluster

    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()
    c = Cluster(initialize_head=True, head_node_args={"num_cpus": 4})
    try:
        script = textwrap.dedent(f"""
            import time
            import ant_ray_amd as ray
            ray.init(address="{c.address}")

            @ray.remote
            class A:
                def ping(self):
                    return 1

            plain = A.remote()
            kept = A.options(name="survivor", lifetime="detached").remote()
            ray.get([plain.ping.remote(), kept.ping.remote()], timeout=60)
            print("READY", flush=True)
            time.sleep(300)
        """)
        proc = subprocess.Popen([sys.executable, "-c", script],
                                stdout=subprocess.PIPE, text=True,
                                cwd="/root/repo")
        assert proc.stdout.readline().startswith("READY")
        proc.kill()
        proc.wait(timeout=10)

        c.connect()
        from ant_ray_amd.util import state

        deadline = time.time() + 60
        ok = False
        while time.time() < deadline:
            rows = state.list_actors()
            named = {r["name"]: r["state"] for r in rows}
            anon_alive = [r for r in rows
                          if not r["name"] and r["state"] == "ALIVE"]
            if (named.get("survivor") == "ALIVE" and not anon_alive):
                ok = True
                break
            time.sleep(1.0)
        assert ok, rows
        # the detached actor still answers
        s = ray.get_actor("survivor")
        assert ray.get(s.ping.remote(), timeout=60) == 1
    finally:
        try:
            ray.shutdown()
        except Exception:
            pass
        c.shutdown()


def test_restored_actor_with_dead_worker_recovers(tmp_path):
    """An actor whose WORKER dies while the GCS is down must be detected
    by the restarted GCS's liveness verification and restarted (its
    max_restarts budget), not left ALIVE-with-stale-addr forever."""
    port = _free_port()
    persist = str(tmp_path / "gcs.msgpack")
    session = str(tmp_path / "session")
    os.makedirs(session, exist_ok=True)

    gcs = _spawn_gcs(port, persist)
    raylet = None
    try:
        assert _wait_port(port)
        raylet = subprocess.Popen(
            [sys.executable, "-m", "ant_ray_amd._private.raylet",
             "--gcs", f"127.0.0.1:{port}", "--num-cpus", "2",
             "--num-gpus", "0", "--store-path", str(tmp_path / "store"),
             "--store-capacity", str(256 * 1024 * 1024),
             "--session-dir", session],
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)

        import ant_ray_amd as ray

        if ray.is_initialized():
            ray.shutdown()
        deadline = time.time() + 30
        while True:
            try:
                ray.init(address=f"127.0.0.1:{port}")
                break
            except Exception:
                if time.time() > deadline:
                    raise
                time.sleep(0.5)

        @ray.remote(max_restarts=2)
        class Phoenix:
            def pid(self):
                import os as _os

                return _os.getpid()

        p = Phoenix.options(name="phoenix", lifetime="detached").remote()
        pid1 = ray.get(p.pid.remote(), timeout=60)
        time.sleep(0.5)  # snapshot lands

        gcs.send_signal(signal.SIGKILL)
        gcs.wait(timeout=10)
        os.kill(pid1, signal.SIGKILL)  # the actor dies DURING the outage
        time.sleep(1.0)
        gcs = _spawn_gcs(port, persist)
        assert _wait_port(port)

        deadline = time.time() + 90
        pid2 = None
        while time.time() < deadline:
            try:
                p2 = ray.get_actor("phoenix")
                pid2 = ray.get(p2.pid.remote(), timeout=20)
                break
            except Exception:
                time.sleep(1.0)
        assert pid2 is not None and pid2 != pid1, (pid1, pid2)
        ray.shutdown()
    finally:
        for proc in (gcs, raylet):
            if proc is not None and proc.poll() is None:
                proc.kill()
