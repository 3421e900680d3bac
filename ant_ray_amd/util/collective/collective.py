"""ray.util.collective on RCCL-over-xGMI / gloo.

Role parity: reference python/ray/util/collective/collective.py:328-725 and
the NCCL group implementation (collective_group/nccl_collective_group.py).
MI355X-first design differences:

  * groups are standalone torch.distributed ProcessGroups built over a
    TCPStore — backend "nccl" IS RCCL on ROCm. No cupy, no per-group
    ncclUniqueId actor: rendezvous goes through the GCS KV (the reference
    stores the ncclUniqueId in a named actor — nccl_collective_group.py:29).
  * one process per GPU (xGMI point-to-point topology: 7 links x ~153 GB/s
    per GPU); multi-gpu-per-process variants are compatibility loops.
"""
from __future__ import annotations

import datetime
import logging
import os
import socket
import threading
import time
from typing import Dict, List, Optional

from ant_ray_amd.util.collective.types import Backend, ReduceOp, torch_reduce_op

logger = logging.getLogger("antray.collective")

_groups: Dict[str, "Group"] = {}
_lock = threading.Lock()


class Group:
    def __init__(self, name, backend, rank, world_size, pg, store):
        self.name = name
        self.backend = backend
        self.rank = rank
        self.world_size = world_size
        self.pg = pg
        self.store = store


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _kv_rendezvous(group_name: str, rank: int, timeout: float = 120.0) -> str:
    """Rank 0 publishes host:port in the GCS KV; others poll it."""
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    key = f"collective/{group_name}".encode()
    if rank == 0:
        host = os.environ.get("ANTRAY_NODE_IP", "127.0.0.1")
        addr = f"{host}:{_free_port()}"
        cw.io.run(cw.gcs.call("kv_put", {"ns": "collective", "key": key,
                                         "value": addr.encode(), "overwrite": True}))
        return addr
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        r = cw.io.run(cw.gcs.call("kv_get", {"ns": "collective", "key": key}))
        if r.get("value"):
            return r["value"].decode()
        time.sleep(0.05)
    raise TimeoutError(f"collective rendezvous for '{group_name}' timed out")


def init_collective_group(
    world_size: int,
    rank: int,
    backend="nccl",
    group_name: str = "default",
    master_addr: Optional[str] = None,
) -> None:
    """Initialize this process's membership in a named collective group."""
    import torch.distributed as dist

    backend = Backend.parse(backend)
    with _lock:
        if group_name in _groups:
            raise RuntimeError(f"group '{group_name}' already initialized")
    if master_addr is None:
        env_key = f"ANTRAY_COLLECTIVE_{group_name.upper()}"
        master_addr = os.environ.get(env_key)
    if master_addr is None:
        try:
            master_addr = _kv_rendezvous(group_name, rank)
        except Exception:
            if rank == 0:
                master_addr = f"127.0.0.1:{_free_port()}"
            else:
                raise
    host, port = master_addr.rsplit(":", 1)
    store = dist.TCPStore(
        host, int(port), world_size, is_master=(rank == 0),
        timeout=datetime.timedelta(seconds=120), wait_for_workers=False,
    )
    if backend == Backend.GLOO:
        pg = dist.ProcessGroupGloo(store, rank, world_size,
                                   datetime.timedelta(seconds=120))
    else:
        from torch.distributed import ProcessGroupNCCL

        opts = ProcessGroupNCCL.Options()
        pg = ProcessGroupNCCL(store, rank, world_size, opts)
    with _lock:
        _groups[group_name] = Group(group_name, backend, rank, world_size, pg, store)
    logger.info("collective group '%s' rank %d/%d (%s) ready",
                group_name, rank, world_size, backend.value)


def create_collective_group(actors, world_size: int, ranks: List[int],
                            backend="nccl", group_name: str = "default"):
    """Driver-side declarative setup: tells each actor to init its rank.
    Actors must expose an `init_collective_group`-calling method or be plain
    actors — we invoke the module-level init in their process via
    __ray_call__-style helper method `_antray_init_collective` if present,
    else `init_collective_group`."""
    import ant_ray_amd as ray

    refs = []
    for actor, rank in zip(actors, ranks):
        m = getattr(actor, "_antray_init_collective", None) or getattr(
            actor, "init_collective_group", None
        )
        if m is None:
            raise ValueError(
                "actor must define init_collective_group(world_size, rank, "
                "backend, group_name) to join a collective group"
            )
        refs.append(m.remote(world_size, rank, backend, group_name))
    ray.get(refs)


def _get(group_name: str) -> Group:
    with _lock:
        g = _groups.get(group_name)
    if g is None:
        raise RuntimeError(f"collective group '{group_name}' is not initialized")
    return g


def is_group_initialized(group_name: str = "default") -> bool:
    with _lock:
        return group_name in _groups


def destroy_collective_group(group_name: str = "default"):
    with _lock:
        g = _groups.pop(group_name, None)
    if g is not None:
        del g.pg
        del g.store


def get_rank(group_name: str = "default") -> int:
    return _get(group_name).rank


def get_collective_group_size(group_name: str = "default") -> int:
    return _get(group_name).world_size


# ----------------------------------------------------------- collective ops


def allreduce(tensor, group_name: str = "default", op=ReduceOp.SUM):
    g = _get(group_name)
    opts_op = torch_reduce_op(op if isinstance(op, ReduceOp) else ReduceOp(op))
    import torch.distributed as dist

    o = dist.AllreduceOptions()
    o.reduceOp = opts_op
    g.pg.allreduce([tensor], o).wait()


def allreduce_multigpu(tensor_list, group_name: str = "default", op=ReduceOp.SUM):
    for t in tensor_list:
        allreduce(t, group_name, op)


def reduce(tensor, dst_rank: int = 0, group_name: str = "default", op=ReduceOp.SUM):
    g = _get(group_name)
    import torch.distributed as dist

    o = dist.ReduceOptions()
    o.reduceOp = torch_reduce_op(op if isinstance(op, ReduceOp) else ReduceOp(op))
    o.rootRank = dst_rank
    g.pg.reduce([tensor], o).wait()


def broadcast(tensor, src_rank: int = 0, group_name: str = "default"):
    g = _get(group_name)
    import torch.distributed as dist

    o = dist.BroadcastOptions()
    o.rootRank = src_rank
    o.rootTensor = 0
    g.pg.broadcast([tensor], o).wait()


def allgather(tensor_list: list, tensor, group_name: str = "default"):
    g = _get(group_name)
    g.pg.allgather([tensor_list], [tensor]).wait()


def allgather_multigpu(output_lists, input_list, group_name: str = "default"):
    for out, t in zip(output_lists, input_list):
        allgather(out, t, group_name)


def reducescatter(tensor, tensor_list: list, group_name: str = "default",
                  op=ReduceOp.SUM):
    g = _get(group_name)
    import torch.distributed as dist

    o = dist.ReduceScatterOptions()
    o.reduceOp = torch_reduce_op(op if isinstance(op, ReduceOp) else ReduceOp(op))
    g.pg.reduce_scatter([tensor], [tensor_list], o).wait()


def send(tensor, dst_rank: int, group_name: str = "default"):
    g = _get(group_name)
    g.pg.send([tensor], dst_rank, 0).wait()


def recv(tensor, src_rank: int, group_name: str = "default"):
    g = _get(group_name)
    g.pg.recv([tensor], src_rank, 0).wait()


def barrier(group_name: str = "default"):
    g = _get(group_name)
    g.pg.barrier().wait()


def get_group_handle(group_name: str = "default"):
    """The underlying torch ProcessGroup (for torch.distributed interop)."""
    return _get(group_name).pg
