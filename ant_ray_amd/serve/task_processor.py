"""Serve task queue: async task processing consumed by deployments.

Role parity: ant fork python/ray/serve/task_processor.py:41
(CeleryTaskProcessorAdapter) + task_consumer.py:95 (@task_consumer): Serve
deployments consume tasks from a durable queue. This image has no Celery
broker; QueueTaskProcessorAdapter implements the same adapter interface on
ray.util.queue (actor-backed). CeleryTaskProcessorAdapter keeps the
reference class name and activates when celery is importable.
"""
from __future__ import annotations

import threading
import time
import uuid
from typing import Any, Callable, Dict, Optional


class TaskProcessorAdapter:
    """Adapter interface (parity: serve/task_processor.py)."""

    def enqueue(self, task_name: str, *args, **kwargs) -> str:
        raise NotImplementedError

    def register(self, task_name: str, fn: Callable):
        raise NotImplementedError

    def start_consumer(self):
        raise NotImplementedError

    def status(self, task_id: str) -> Dict[str, Any]:
        raise NotImplementedError


class QueueTaskProcessorAdapter(TaskProcessorAdapter):
    """Queue-actor-backed adapter: durable within the cluster session."""

    def __init__(self, queue_name: str = "serve_tasks"):
        from ant_ray_amd.util.queue import Queue

        self.queue = Queue(actor_options={"name": f"_taskq:{queue_name}",
                                          "get_if_exists": True})
        self._handlers: Dict[str, Callable] = {}
        self._results: Dict[str, Dict[str, Any]] = {}
        self._consumer: Optional[threading.Thread] = None
        self._stop = threading.Event()

    def register(self, task_name: str, fn: Callable):
        self._handlers[task_name] = fn

    def enqueue(self, task_name: str, *args, **kwargs) -> str:
        task_id = uuid.uuid4().hex
        # record BEFORE the put: the consumer may finish the task first and
        # a later write would clobber its result
        self._results[task_id] = {"status": "PENDING"}
        self.queue.put({"id": task_id, "name": task_name, "args": args,
                        "kwargs": kwargs})
        return task_id

    def _consume_loop(self):
        from queue import Empty

        while not self._stop.is_set():
            try:
                item = self.queue.get(timeout=0.5)
            except Empty:
                continue
            fn = self._handlers.get(item["name"])
            rec = self._results.setdefault(item["id"], {})
            if fn is None:
                rec.update(status="FAILED", error=f"no handler {item['name']}")
                continue
            rec["status"] = "RUNNING"
            try:
                rec["result"] = fn(*item["args"], **item["kwargs"])
                rec["status"] = "SUCCEEDED"
            except Exception as e:  # noqa: BLE001
                rec.update(status="FAILED", error=str(e))

    def start_consumer(self):
        self._consumer = threading.Thread(target=self._consume_loop,
                                          daemon=True)
        self._consumer.start()

    def stop_consumer(self):
        self._stop.set()

    def status(self, task_id: str) -> Dict[str, Any]:
        return dict(self._results.get(task_id, {"status": "UNKNOWN"}))

    def wait(self, task_id: str, timeout: float = 30) -> Dict[str, Any]:
        deadline = time.time() + timeout
        while time.time() < deadline:
            st = self.status(task_id)
            if st.get("status") in ("SUCCEEDED", "FAILED"):
                return st
            time.sleep(0.05)
        return self.status(task_id)


class CeleryTaskProcessorAdapter(TaskProcessorAdapter):
    """Reference-named adapter; needs a celery broker (not in this image)."""

    def __init__(self, *a, **kw):
        try:
            import celery  # noqa: F401
        except ImportError as e:
            raise ImportError(
                "CeleryTaskProcessorAdapter needs celery + a broker; use "
                "QueueTaskProcessorAdapter in this deployment") from e


def task_consumer(queue_name: str = "serve_tasks"):
    """@task_consumer class decorator (parity task_consumer.py:95): methods
    marked @task_handler are registered and consumed from the queue."""

    def wrap(cls):
        orig_init = cls.__init__

        def __init__(self, *args, **kwargs):
            orig_init(self, *args, **kwargs)
            self._task_adapter = QueueTaskProcessorAdapter(queue_name)
            for name in dir(self):
                m = getattr(self, name)
                if getattr(m, "__task_handler__", False):
                    self._task_adapter.register(
                        getattr(m, "__task_name__", name), m)
            self._task_adapter.start_consumer()

        cls.__init__ = __init__
        return cls

    return wrap


def task_handler(_fn=None, *, name: Optional[str] = None):
    def wrap(fn):
        fn.__task_handler__ = True
        if name:
            fn.__task_name__ = name
        return fn

    return wrap(_fn) if _fn else wrap
