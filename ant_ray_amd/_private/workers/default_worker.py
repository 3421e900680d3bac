"""Worker process entrypoint.

Role parity: reference python/ray/_private/workers/default_worker.py (spawned
by the raylet WorkerPool; connects CoreWorker and executes pushed tasks).
Configuration comes via ANTRAY_* env vars set by the raylet at spawn.
"""
from __future__ import annotations

import logging
import os
import threading


def main():
    from ant_ray_amd._private.stack_dump import install as _stack_install

    _stack_install()
    logging.basicConfig(level=logging.INFO)
    worker_id = bytes.fromhex(os.environ["ANTRAY_WORKER_ID"])
    gcs_host, gcs_port = os.environ["ANTRAY_GCS"].rsplit(":", 1)
    raylet_host, raylet_port = os.environ["ANTRAY_RAYLET"].rsplit(":", 1)
    node_id = bytes.fromhex(os.environ["ANTRAY_NODE_ID"])
    store_path = os.environ["ANTRAY_STORE"]
    session_dir = os.environ.get("ANTRAY_SESSION_DIR", "")

    from ant_ray_amd._private.task_executor import TaskExecutor
    from ant_ray_amd._private.worker import WORKER_MODE, CoreWorker, global_worker

    cw = CoreWorker(
        WORKER_MODE,
        node_ip=raylet_host,
        worker_id=worker_id,
        session_dir=session_dir,
    )
    global_worker.core_worker = cw
    global_worker.mode = WORKER_MODE
    cw.executor = TaskExecutor(cw)
    cw.connect(
        (gcs_host, int(gcs_port)),
        is_driver=False,
        raylet_addr=(raylet_host, int(raylet_port)),
        store_path=store_path,
        node_id=node_id,
    )
    try:
        # driver-pushed structured logging config (ray.init(logging_config=))
        r = cw.io.run(cw.gcs.call("kv_get", {"ns": "_cluster",
                                             "key": b"logging_config"},
                                  timeout=5), timeout=8)
        if r and r.get("value"):
            import json as _json

            from ant_ray_amd._private.logging_config import LoggingConfig

            LoggingConfig._from_dict(_json.loads(r["value"].decode()))._apply()
    except Exception:
        pass
    logging.getLogger("antray.worker").info(
        "worker %s ready on %s", worker_id.hex()[:8], cw.addr
    )
    threading.Event().wait()


if __name__ == "__main__":
    main()
