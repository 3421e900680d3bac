"""Train worker group: one actor per worker, placement-group pinned.

Role parity: reference python/ray/train/v2/_internal/execution/worker_group/
worker_group.py (actor creation + dist bootstrap) and thread_runner.py (user
train_func runs on a thread so the actor stays responsive to poll()).
MI355X shape: one worker per GPU (`num_gpus=1`), ranks laid out so
LOCAL_RANK == HIP device index on the node; torch.distributed over RCCL.
"""
from __future__ import annotations

import os
import socket
import threading
import traceback
from typing import Any, Callable, Dict, List, Optional

import ant_ray_amd as ray
from ant_ray_amd.train._checkpoint import Checkpoint
from ant_ray_amd.train.session import TrainContext, set_train_context


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


class TrainWorker:
    """Actor hosting one training rank. max_concurrency>1 so poll() works
    while the training thread runs."""

    def __init__(self):
        self._thread: Optional[threading.Thread] = None
        self._ctx: Optional[TrainContext] = None
        self._error: Optional[str] = None
        self._done = False

    # ------------------------------------------------------- rendezvous

    def get_node_ip(self) -> str:
        return os.environ.get("ANTRAY_NODE_IP", "127.0.0.1")

    def get_free_port(self) -> int:
        return _free_port()

    def get_gpu_ids(self) -> List[int]:
        return ray.get_gpu_ids()

    # ------------------------------------------------------------ setup

    def init_dist(self, world_rank: int, local_rank: int, world_size: int,
                  local_world_size: int, node_rank: int, master_addr: str,
                  master_port: int, backend: str, timeout_s: int = 1800):
        """torch.distributed.init_process_group; backend nccl IS RCCL."""
        import datetime

        import torch
        import torch.distributed as dist

        # a leaked TORCHELASTIC_USE_AGENT_STORE (e.g. the cluster was
        # started from inside a torchrun rank) would make this join a
        # nonexistent agent store at MASTER_PORT and hang the rendezvous
        for k in [k for k in os.environ if k.startswith("TORCHELASTIC")]:
            os.environ.pop(k, None)
        os.environ["RANK"] = str(world_rank)
        os.environ["LOCAL_RANK"] = str(local_rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["LOCAL_WORLD_SIZE"] = str(local_world_size)
        os.environ["NODE_RANK"] = str(node_rank)
        os.environ["MASTER_ADDR"] = master_addr
        os.environ["MASTER_PORT"] = str(master_port)
        self._dist_info = (world_rank, local_rank, world_size, local_world_size,
                           node_rank)
        if backend == "nccl" and torch.cuda.is_available():
            torch.cuda.set_device(local_rank % torch.cuda.device_count())
        if world_size > 1 or backend == "gloo":
            dist.init_process_group(
                backend=backend,
                init_method=f"tcp://{master_addr}:{master_port}",
                rank=world_rank,
                world_size=world_size,
                timeout=datetime.timedelta(seconds=timeout_s),
            )
        return True

    # --------------------------------------------------------- training

    def start_training(self, train_fn: Callable, config: Optional[dict],
                       experiment_name: str, experiment_path: str,
                       restore_checkpoint_path: Optional[str],
                       dataset_shards: Optional[Dict[str, Any]] = None):
        assert self._thread is None, "training already started on this worker"
        r, lr, ws, lws, nr = getattr(self, "_dist_info", (0, 0, 1, 1, 0))
        self._ctx = TrainContext(
            experiment_name=experiment_name,
            experiment_path=experiment_path,
            world_rank=r, local_rank=lr, world_size=ws,
            local_world_size=lws, node_rank=nr,
            restore_checkpoint=(
                Checkpoint(restore_checkpoint_path)
                if restore_checkpoint_path else None
            ),
            dataset_shards=dataset_shards or {},
        )

        def run():
            set_train_context(self._ctx)
            try:
                if config is not None:
                    train_fn(config)
                else:
                    try:
                        train_fn({})
                    except TypeError:
                        train_fn()
            except BaseException:
                self._error = traceback.format_exc()
            finally:
                self._done = True

        self._thread = threading.Thread(target=run, daemon=True, name="train_fn")
        self._thread.start()
        return True

    def poll(self) -> Dict[str, Any]:
        reports = []
        if self._ctx is not None:
            while True:
                try:
                    reports.append(self._ctx.report_queue.get_nowait())
                except Exception:
                    break
        status = "running"
        if self._done:
            status = "errored" if self._error else "finished"
        elif self._thread is None:
            status = "idle"
        return {"status": status, "reports": reports, "error": self._error}

    def shutdown(self):
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()
        return True


class WorkerGroup:
    """Driver-side handle on the N TrainWorker actors of one attempt."""

    def __init__(self, scaling, torch_config, experiment_name: str,
                 experiment_path: str):
        self.scaling = scaling
        self.torch_config = torch_config
        self.experiment_name = experiment_name
        self.experiment_path = experiment_path
        self.workers: List[Any] = []
        self.pg = None

    def start(self):
        from ant_ray_amd.util.placement_group import (
            placement_group,
            remove_placement_group,  # noqa: F401
        )
        from ant_ray_amd.util.scheduling_strategies import (
            PlacementGroupSchedulingStrategy,
        )

        n = self.scaling.num_workers
        res = self.scaling._resources_per_worker_not_none
        bundles = [dict(res) for _ in range(n)]
        self.pg = placement_group(bundles, strategy=self.scaling.placement_strategy)
        self.pg.wait(timeout_seconds=60)
        WorkerCls = ray.remote(TrainWorker)
        self.workers = [
            WorkerCls.options(
                num_cpus=res.get("CPU", 1),
                num_gpus=res.get("GPU", 0),
                max_concurrency=4,
                scheduling_strategy=PlacementGroupSchedulingStrategy(
                    placement_group=self.pg, placement_group_bundle_index=i
                ),
            ).remote()
            for i in range(n)
        ]
        # rank layout: group workers by node ip so LOCAL_RANK is contiguous
        ips = ray.get([w.get_node_ip.remote() for w in self.workers])
        order = sorted(range(n), key=lambda i: (ips[i], i))
        self.workers = [self.workers[i] for i in order]
        ips = [ips[i] for i in order]
        master_addr = ips[0]
        master_port = ray.get(self.workers[0].get_free_port.remote())
        backend = self.torch_config.resolved_backend(self.scaling.use_gpu)
        node_of, local_rank, seen = [], [], {}
        local_count: Dict[str, int] = {}
        for ip in ips:
            node_of.append(seen.setdefault(ip, len(seen)))
            local_rank.append(local_count.get(ip, 0))
            local_count[ip] = local_rank[-1] + 1
        ray.get([
            w.init_dist.remote(
                i, local_rank[i], n, local_count[ips[i]], node_of[i],
                master_addr, master_port, backend, self.torch_config.timeout_s,
            )
            for i, w in enumerate(self.workers)
        ])

    def start_training(self, train_fn, config, restore_path,
                       dataset_shards_per_worker=None):
        refs = []
        for i, w in enumerate(self.workers):
            shards = (dataset_shards_per_worker[i]
                      if dataset_shards_per_worker else None)
            refs.append(w.start_training.remote(
                train_fn, config, self.experiment_name, self.experiment_path,
                restore_path, shards,
            ))
        ray.get(refs)

    def poll(self) -> List[Dict[str, Any]]:
        return ray.get([w.poll.remote() for w in self.workers])

    def shutdown(self):
        try:
            ray.get([w.shutdown.remote() for w in self.workers], timeout=10)
        except Exception:
            pass
        for w in self.workers:
            try:
                ray.kill(w)
            except Exception:
                pass
        if self.pg is not None:
            try:
                from ant_ray_amd.util.placement_group import remove_placement_group

                remove_placement_group(self.pg)
            except Exception:
                pass
        self.workers = []
        self.pg = None
