"""Local-mode execution: tasks/actors run synchronously in-process.

Role parity: reference ray.init(local_mode=True) (used by BASELINE config 1:
"@ray.remote task+actor round-trip on CPU"). Values go through the memory
store; errors surface at ray.get like in distributed mode.
"""
from __future__ import annotations

import threading
from typing import Any, Dict

from ant_ray_amd._private.ids import ActorID, JobID, ObjectID, TaskID
from ant_ray_amd._private.object_ref import ObjectRef
from ant_ray_amd.exceptions import RayActorError, RayTaskError


class LocalModeExecutor:
    def __init__(self, core_worker):
        self.cw = core_worker
        self.actors: Dict[bytes, Any] = {}
        self._lock = threading.Lock()

    def _resolve(self, v):
        if isinstance(v, ObjectRef):
            return self.cw.get([v], None)[0]
        return v

    def _store_result(self, task_id: bytes, n_returns: int, fn, args, kwargs, name):
        refs = [
            ObjectRef(ObjectID.for_return(TaskID(task_id), i).binary(), None, worker=self.cw)
            for i in range(max(n_returns, 1))
        ]
        try:
            args = tuple(self._resolve(a) for a in args)
            kwargs = {k: self._resolve(v) for k, v in kwargs.items()}
            result = fn(*args, **kwargs)
            if n_returns <= 1:
                self.cw.store.memory.put(refs[0].binary(), result)
            else:
                values = tuple(result)
                if len(values) != n_returns:
                    raise ValueError(
                        f"task returned {len(values)} values, expected {n_returns}"
                    )
                for r, v in zip(refs, values):
                    self.cw.store.memory.put(r.binary(), v)
        except Exception as e:  # noqa: BLE001
            err = e if isinstance(e, RayTaskError) else RayTaskError.from_exception(e, name)
            from ant_ray_amd._private.worker import _ErrorResult

            for r in refs:
                self.cw.store.memory.put(r.binary(), _ErrorResult(err))
        return refs if n_returns > 1 else refs[:1]

    def submit_task(self, fn, args, kwargs, opts):
        task_id = TaskID.for_task(JobID.from_int(self.cw.job_id or 0)).binary()
        n = opts.get("num_returns", 1)
        if n in ("streaming", "dynamic"):
            return self._store_stream(task_id, fn, args, kwargs)
        return self._store_result(task_id, n, fn, args, kwargs, getattr(fn, "__name__", "task"))

    def _store_stream(self, task_id, fn, args, kwargs):
        """local_mode streaming: materialize the generator eagerly and hand
        back an ObjectRefGenerator over the stored items."""
        from ant_ray_amd._private.worker import ObjectRefGenerator, _StreamState

        st = _StreamState()
        self.cw._streams = getattr(self.cw, "_streams", {})
        self.cw._streams[task_id] = st
        try:
            args = tuple(self._resolve(a) for a in args)
            kwargs = {k: self._resolve(v) for k, v in kwargs.items()}
            count = 0
            for v in fn(*args, **kwargs):
                oid = ObjectID.for_return(TaskID(task_id), count).binary()
                self.cw.store.memory.put(oid, v)
                st.refs.append(oid)
                count += 1
            st.total = count
        except Exception as e:  # noqa: BLE001
            st.error = (e if isinstance(e, RayTaskError)
                        else RayTaskError.from_exception(
                            e, getattr(fn, "__name__", "task")))
        return ObjectRefGenerator(task_id, self.cw)

    def create_actor(self, cls, actor_id, args, kwargs, opts):
        args = tuple(self._resolve(a) for a in args)
        kwargs = {k: self._resolve(v) for k, v in kwargs.items()}
        instance = cls(*args, **kwargs)
        with self._lock:
            self.actors[actor_id] = instance
        return actor_id

    def submit_actor_task(self, actor_id, method_name, args, kwargs, opts):
        instance = self.actors.get(actor_id)
        if instance is None:
            raise RayActorError(f"local actor {actor_id.hex()[:8]} not found")
        seq_task = TaskID.for_task(JobID.from_int(self.cw.job_id or 0)).binary()
        n = opts.get("num_returns", 1)
        method = getattr(instance, method_name)
        import inspect

        if inspect.iscoroutinefunction(method):
            import asyncio

            def runner(*a, **kw):
                return asyncio.run(method(*a, **kw))

            return self._store_result(seq_task, n, runner, args, kwargs, method_name)
        return self._store_result(seq_task, n, method, args, kwargs, method_name)

    def kill_actor(self, actor_id, no_restart=True):
        with self._lock:
            self.actors.pop(actor_id, None)

    def request_exit(self):
        pass
