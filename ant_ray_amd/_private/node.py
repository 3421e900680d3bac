"""Driver-side node bootstrap: start/connect a head process.

Role parity: reference python/ray/_private/node.py:53 (class Node; head
startup sequence at node.py:1367-1457) — here the head is one subprocess
(GCS + raylet, see head.py).
"""
from __future__ import annotations

import json
import os
import signal
import subprocess
import sys
import time
from typing import Optional


class HeadProcess:
    def __init__(self, proc: subprocess.Popen, info: dict):
        self.proc = proc
        self.info = info

    def terminate(self):
        try:
            os.killpg(self.proc.pid, signal.SIGKILL)
        except Exception:
            try:
                self.proc.kill()
            except Exception:
                pass


def start_head(
    num_cpus: Optional[int] = None,
    num_gpus: Optional[int] = None,
    object_store_memory: Optional[int] = None,
    resources: Optional[dict] = None,
    host: str = "127.0.0.1",
    port: int = 0,
    session_dir: str = "",
    prestart: int = 0,  # 0 = one per CPU
    timeout: float = 60.0,
    owner_pid: Optional[int] = -1,  # -1 = this process; 0/None = detached
) -> HeadProcess:
    if not session_dir:
        session_dir = os.path.join("/tmp/antray", f"session_{int(time.time()*1000)}_{os.getpid()}")
    os.makedirs(os.path.join(session_dir, "logs"), exist_ok=True)
    cmd = [
        sys.executable, "-m", "ant_ray_amd._private.head",
        "--host", host,
        "--port", str(port),
        "--session-dir", session_dir,
        "--prestart", str(prestart),
    ]
    if owner_pid == -1:
        owner_pid = os.getpid()
    if owner_pid:
        # a head started by ray.init fate-shares with its driver: if the
        # driver dies without ray.shutdown (SIGKILL, crash), the cluster
        # must not linger as orphaned processes. `ray start --head`
        # passes owner_pid=0 and stays detached.
        cmd += ["--owner-pid", str(owner_pid)]
    if num_cpus is not None:
        cmd += ["--num-cpus", str(num_cpus)]
    if num_gpus is not None:
        cmd += ["--num-gpus", str(num_gpus)]
    if object_store_memory:
        cmd += ["--object-store-memory", str(int(object_store_memory))]
    if resources:
        cmd += ["--resources", json.dumps(resources)]
    err_log = open(os.path.join(session_dir, "logs", "head.err"), "wb")
    proc = subprocess.Popen(
        cmd,
        stdout=subprocess.PIPE,
        stderr=err_log,
        start_new_session=True,
    )
    err_log.close()
    deadline = time.monotonic() + timeout
    info = None
    while time.monotonic() < deadline:
        line = proc.stdout.readline()
        if not line:
            if proc.poll() is not None:
                with open(os.path.join(session_dir, "logs", "head.err"), "rb") as f:
                    tail = f.read()[-4000:].decode(errors="replace")
                raise RuntimeError(f"head process exited: {tail}")
            time.sleep(0.05)
            continue
        line = line.decode(errors="replace").strip()
        if line.startswith("ANTRAY_HEAD "):
            info = json.loads(line[len("ANTRAY_HEAD "):])
            break
    if info is None:
        proc.kill()
        raise RuntimeError("timed out waiting for head process to start")
    return HeadProcess(proc, info)
