"""Per-node reporter: physical stats (psutil) + GPU stats (amdsmi /
rocm-smi) published to the GCS.

Role parity: reference python/ray/dashboard/modules/reporter/
reporter_agent.py:393 ReporterAgent — a per-node agent sampling
cpu/mem/disk/net + GPU utilization and shipping them to the dashboard.
Here the reporter runs as a task inside each raylet (the lite equivalent
of the reference's per-node agent process) and publishes one JSON record
per node into the GCS KV (ns "node_stats", key = node id hex); the
dashboard head serves them at /api/node_stats and folds the latest
sample into /api/nodes.
"""
from __future__ import annotations

import json
import os
import time
from typing import List, Optional


def _gpu_stats_amdsmi() -> Optional[List[dict]]:
    try:
        import amdsmi

        amdsmi.amdsmi_init()
        try:
            out = []
            for i, h in enumerate(amdsmi.amdsmi_get_processor_handles()):
                rec = {"index": i}
                try:
                    e = amdsmi.amdsmi_get_gpu_activity(h)
                    rec["gfx_busy_pct"] = e.get("gfx_activity")
                    rec["mem_busy_pct"] = e.get("umc_activity")
                except Exception:
                    pass
                try:
                    m = amdsmi.amdsmi_get_gpu_vram_usage(h)
                    rec["vram_used_mb"] = m.get("vram_used")
                    rec["vram_total_mb"] = m.get("vram_total")
                except Exception:
                    pass
                try:
                    p = amdsmi.amdsmi_get_power_info(h)
                    rec["power_w"] = p.get("average_socket_power")
                except Exception:
                    pass
                try:
                    t = amdsmi.amdsmi_get_temp_metric(
                        h, amdsmi.AmdSmiTemperatureType.JUNCTION,
                        amdsmi.AmdSmiTemperatureMetric.CURRENT)
                    rec["temp_c"] = t
                except Exception:
                    pass
                out.append(rec)
            return out
        finally:
            try:
                amdsmi.amdsmi_shut_down()
            except Exception:
                pass
    except Exception:
        return None


def _gpu_stats_rocm_smi() -> Optional[List[dict]]:
    import subprocess

    try:
        r = subprocess.run(
            ["rocm-smi", "--showuse", "--showmemuse", "--showmeminfo",
             "vram", "--json"],
            capture_output=True, text=True, timeout=10)
        data = json.loads(r.stdout or "{}")
        out = []
        for key, v in sorted(data.items()):
            if not key.startswith("card"):
                continue
            rec = {"index": int(key[4:]) if key[4:].isdigit() else key}
            for src, dst in (("GPU use (%)", "gfx_busy_pct"),
                             ("GPU Memory Allocated (VRAM%)", "vram_pct"),
                             ("VRAM Total Memory (B)", "vram_total_b"),
                             ("VRAM Total Used Memory (B)", "vram_used_b")):
                if src in v:
                    try:
                        rec[dst] = float(v[src])
                    except (TypeError, ValueError):
                        pass
            out.append(rec)
        return out or None
    except Exception:
        return None


def sample_node_stats(node_id_hex: str, node_ip: str,
                      with_gpu: bool = True) -> dict:
    """One sample of this node's physical stats (reference ReporterAgent
    _get_all_stats shape, reduced)."""
    import psutil

    la1, la5, la15 = (os.getloadavg() if hasattr(os, "getloadavg")
                      else (0.0, 0.0, 0.0))
    vm = psutil.virtual_memory()
    net = psutil.net_io_counters()
    try:
        disk = psutil.disk_usage("/")
        disk_rec = {"total": disk.total, "used": disk.used,
                    "percent": disk.percent}
    except Exception:
        disk_rec = {}
    rec = {
        "node_id": node_id_hex,
        "ip": node_ip,
        "ts": time.time(),
        "cpu_percent": psutil.cpu_percent(interval=None),
        "cpus": psutil.cpu_count(),
        "load_avg": [la1, la5, la15],
        "mem": {"total": vm.total, "available": vm.available,
                "percent": vm.percent},
        "disk": disk_rec,
        "net": {"sent": net.bytes_sent, "recv": net.bytes_recv},
        "pid": os.getpid(),
    }
    gpus = None
    if with_gpu:
        gpus = _gpu_stats_amdsmi()
        if gpus is None:
            gpus = _gpu_stats_rocm_smi()
    rec["gpus"] = gpus or []
    # shm object-store arena usage for this node's store
    try:
        import shutil

        shm = shutil.disk_usage("/dev/shm")
        rec["shm"] = {"total": shm.total, "used": shm.used}
    except Exception:
        pass
    return rec


async def reporter_loop(raylet, interval_s: float = 5.0):
    """Runs inside the raylet's event loop; publishes to GCS KV."""
    import asyncio

    # amdsmi init/shutdown per sample is slow; only poll GPU when the node
    # actually advertises GPUs
    with_gpu = raylet.resources_total.get("GPU", 0) > 0
    seq = 0
    while not raylet._shutdown.is_set():
        try:
            loop = asyncio.get_running_loop()
            rec = await loop.run_in_executor(
                None, sample_node_stats, raylet.node_id.hex(),
                raylet.node_ip, with_gpu)
            seq += 1
            await raylet.gcs_conn.call("kv_put", {
                "ns": "node_stats", "key": raylet.node_id.hex().encode(),
                "value": json.dumps(rec).encode(), "overwrite": True,
                "seq": seq, "seq_id": raylet.node_id,
            }, timeout=5)
        except Exception:
            pass
        try:
            await asyncio.sleep(interval_s)
        except Exception:
            return
