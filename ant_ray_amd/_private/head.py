"""Head process: GCS + the head node's raylet in one process.

Role parity: reference `ray start --head` which launches gcs_server + raylet
as separate binaries (python/ray/_private/node.py:1367 start_head_processes).
We co-locate them in one asyncio process; worker nodes run raylet.py alone.
Prints a JSON line with the session info on startup (consumed by ray.init or
by `ant-ray start --head`).
"""
from __future__ import annotations

import argparse
import asyncio
import json
import logging
import os
import sys
import time


def default_store_capacity() -> int:
    try:
        st = os.statvfs("/dev/shm")
        free = st.f_bavail * st.f_frsize
        return max(256 * 1024 * 1024, min(int(free * 0.6), 64 * 1024**3))
    except Exception:
        return 2 * 1024**3


async def run_head(args):
    from ant_ray_amd._private.gcs import GcsServer
    from ant_ray_amd._private.raylet import Raylet, detect_num_gpus

    session_dir = args.session_dir
    os.makedirs(os.path.join(session_dir, "logs"), exist_ok=True)

    gcs = GcsServer()
    if getattr(args, "owner_pid", 0):
        async def _watch_owner():
            # the driver that started this head is gone (crash/SIGKILL)
            # -> tear the whole local cluster down instead of leaking
            # orphaned head+worker processes
            while True:
                await asyncio.sleep(2.0)
                try:
                    os.kill(args.owner_pid, 0)
                except ProcessLookupError:
                    logging.getLogger("antray.head").warning(
                        "owner pid %d gone; shutting down session",
                        args.owner_pid)
                    os._exit(0)  # raylet workers fate-share via conn close
                except PermissionError:
                    pass  # pid exists under another uid: still alive

        asyncio.get_running_loop().create_task(_watch_owner())
    gcs_port = await gcs.start(
        args.host, args.port,
        persist_path=os.path.join(session_dir, "gcs_tables.msgpack"))

    resources = json.loads(args.resources) if args.resources else {}
    ncpu = args.num_cpus if args.num_cpus >= 0 else os.cpu_count()
    ngpu = args.num_gpus if args.num_gpus >= 0 else detect_num_gpus()
    resources.setdefault("CPU", float(ncpu))
    resources.setdefault("GPU", float(ngpu))
    resources.setdefault("memory", float(os.sysconf("SC_PHYS_PAGES") * os.sysconf("SC_PAGE_SIZE")))
    resources.setdefault(f"node:{args.host}", 1.0)
    capacity = args.object_store_memory or default_store_capacity()
    resources.setdefault("object_store_memory", float(capacity))
    # GC arenas leaked by SIGKILLed sessions (name = antray_<headpid>_<ts>;
    # if that pid is gone, nothing can still map the store legitimately)
    try:
        for f in os.listdir("/dev/shm"):
            if not f.startswith("antray_"):
                continue
            try:
                pid = int(f.split("_")[1])
                os.kill(pid, 0)
            except (ValueError, IndexError):
                continue
            except ProcessLookupError:
                try:
                    os.unlink(os.path.join("/dev/shm", f))
                except OSError:
                    pass
            except PermissionError:
                pass  # pid alive under another uid: leave it
    except OSError:
        pass
    store_path = os.path.join("/dev/shm", f"antray_{os.getpid()}_{int(time.time())}")

    raylet = Raylet(args.host, (args.host, gcs_port), resources, store_path, capacity, session_dir)
    await raylet.start(0)
    # warm the worker pool
    try:
        # one worker per CPU up to a spawn-storm cap (large hosts fork the
        # rest on demand)
        n_pre = args.prestart if args.prestart > 0 else min(int(ncpu), 16)
        await raylet.rpc_prestart_workers(None, {"n": min(int(ncpu), n_pre)})
    except Exception:
        pass

    info = {
        "gcs_addr": f"{args.host}:{gcs_port}",
        "raylet_addr": f"{args.host}:{raylet.port}",
        "node_id": raylet.node_id.hex(),
        "store_path": store_path,
        "session_dir": session_dir,
        "pid": os.getpid(),
    }
    print("ANTRAY_HEAD " + json.dumps(info), flush=True)
    with open(os.path.join(session_dir, "head.json"), "w") as f:
        json.dump(info, f)
    await gcs._shutdown.wait()


def main():
    from ant_ray_amd._private.stack_dump import install as _stack_install

    _stack_install()
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=0)
    ap.add_argument("--num-cpus", type=int, default=-1)
    ap.add_argument("--num-gpus", type=int, default=-1)
    ap.add_argument("--resources", default="")
    ap.add_argument("--object-store-memory", type=int, default=0)
    ap.add_argument("--session-dir", default="")
    ap.add_argument("--prestart", type=int, default=0)
    ap.add_argument("--owner-pid", type=int, default=0,
                    help="fate-share: exit when this pid dies (0=detached)")
    args = ap.parse_args()
    if not args.session_dir:
        args.session_dir = os.path.join(
            "/tmp/antray", f"session_{int(time.time())}_{os.getpid()}"
        )
    log_path = os.path.join(args.session_dir, "logs")
    os.makedirs(log_path, exist_ok=True)
    logging.basicConfig(
        level=logging.INFO,
        filename=os.path.join(log_path, "head.log"),
    )
    asyncio.run(run_head(args))


if __name__ == "__main__":
    main()
