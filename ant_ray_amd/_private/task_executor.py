"""Task execution engine for worker processes.

Role parity: reference TaskReceiver + scheduling queues
(src/ray/core_worker/task_execution/task_receiver.h:44,
actor_scheduling_queue.cc, concurrency_group_manager.cc, fiber.h for async
actors) and the Cython execution path (_raylet.pyx:2141
task_execution_handler). Execution models:

  * normal tasks + plain actors: one serial executor thread (strict order),
  * threaded actors (max_concurrency>1): ThreadPoolExecutor,
  * async actors (coroutine methods): dedicated asyncio loop thread with a
    max_concurrency semaphore,
  * actor task ordering: per-caller sequence numbers; out-of-order arrivals
    are buffered until their turn (parity with SequentialActorSubmitQueue).
"""
from __future__ import annotations

import asyncio
import concurrent.futures
import inspect
import logging
import os
import queue
import threading
import time
from typing import Any, Dict, Optional

from ant_ray_amd._private import serialization
from ant_ray_amd._private.ids import ObjectID, TaskID
from ant_ray_amd._private.object_ref import ObjectRef
from ant_ray_amd._private.object_store import INLINE_OBJECT_MAX
from ant_ray_amd.exceptions import RayTaskError

logger = logging.getLogger("antray.executor")


class TaskExecutor:
    def __init__(self, core_worker):
        self.cw = core_worker
        self._fn_cache: Dict[bytes, Any] = {}
        self.actor_instance = None
        self.actor_id: Optional[bytes] = None
        self.max_concurrency = 1
        self._serial_q: "queue.Queue" = queue.Queue()
        self._cancelled: set = set()        # task_ids cancelled pre-start
        self._running_serial: bytes = None  # task running on the serial thread
        self._serial_cancel_lock = threading.Lock()
        self._serial_thread = threading.Thread(target=self._serial_loop, daemon=True)
        self._serial_thread.start()
        self._pool: Optional[concurrent.futures.ThreadPoolExecutor] = None
        self._actor_loop: Optional[Any] = None  # EventLoopThread for async actors
        self._async_sem: Optional[asyncio.Semaphore] = None
        self._exit_requested = False
        self._inflight = 0
        self._inflight_cv = threading.Condition()
        # actor-task ordering: caller -> expected next seq / buffered payloads
        self._expected_seq: Dict[bytes, int] = {}
        self._buffered: Dict[bytes, Dict[int, tuple]] = {}

    # ------------------------------------------------------------- entrypoint
    async def submit(self, payload: dict) -> dict:
        """Called on the io loop; returns the reply dict when the task is done."""
        loop = asyncio.get_running_loop()
        fut: asyncio.Future = loop.create_future()
        ttype = payload.get("type")
        if ttype == "actor_task":
            self._submit_actor_ordered(payload, fut, loop)
        else:
            self._dispatch(payload, fut, loop)
        return await fut

    def _submit_actor_ordered(self, payload, fut, loop):
        caller = payload["caller"]
        seq = payload["seq"]
        expected = self._expected_seq.get(caller, 1)
        if seq == expected:
            self._dispatch(payload, fut, loop)
            self._expected_seq[caller] = expected + 1
            # drain any buffered successors
            buf = self._buffered.get(caller)
            while buf:
                nxt = self._expected_seq[caller]
                item = buf.pop(nxt, None)
                if item is None:
                    break
                self._dispatch(item[0], item[1], item[2])
                self._expected_seq[caller] = nxt + 1
        else:
            self._buffered.setdefault(caller, {})[seq] = (payload, fut, loop)

    def _dispatch(self, payload, fut, loop):
        with self._inflight_cv:
            self._inflight += 1

        def done(reply):
            with self._inflight_cv:
                self._inflight -= 1
                self._inflight_cv.notify_all()
            loop.call_soon_threadsafe(
                lambda: fut.set_result(reply) if not fut.done() else None
            )
            if self._exit_requested:
                self._maybe_exit()

        method = payload.get("method", "")
        group = payload.get("concurrency_group")
        if (
            payload.get("type") == "actor_task"
            and self.actor_instance is not None
            and self._is_async_method(method)
        ):
            self._run_async_actor_task(payload, done)
        elif (payload.get("type") == "actor_task" and group
                and group in getattr(self, "_groups", {})):
            self._groups[group].submit(self._run_and_reply, payload, done)
        elif payload.get("type") == "actor_task" and (
                self.max_concurrency > 1 and self._pool is not None):
            self._pool.submit(self._run_and_reply, payload, done)
        else:
            self._serial_q.put((payload, done))

    def _serial_loop(self):
        while True:
            try:
                payload, done = self._serial_q.get()
                if payload is None:
                    return
                with self._serial_cancel_lock:
                    self._running_serial = payload.get("task_id")
                try:
                    self._run_and_reply(payload, done)
                finally:
                    with self._serial_cancel_lock:
                        self._running_serial = None
            except KeyboardInterrupt:
                # a soft-cancel async-exc landed AFTER its target task
                # finished (between cancel_task's _running_serial check and
                # delivery at the next bytecode boundary): the interrupt is
                # stale — swallow it and keep serving, never let it kill
                # the serial thread (callers would hang forever)
                continue

    def cancel_task(self, task_id: bytes, force: bool):
        """Cancel a queued or running task in this worker (parity:
        CoreWorkerService CancelTask). Queued -> dropped before start;
        running on the serial thread -> KeyboardInterrupt raised into it
        via PyThreadState_SetAsyncExc (the reference sends SIGINT to the
        worker's main thread); force -> the whole worker exits (reference
        force kill), the owner maps the dead push to TaskCancelledError.
        Tasks running on a concurrency-group/thread-pool thread are NOT
        interrupted (no per-thread soft cancel, matching the reference's
        main-thread-only SIGINT); they are dropped if still queued and
        force-kill works regardless."""
        self._cancelled.add(task_id)
        t = getattr(self, "_async_tasks", {}).get(task_id)
        if t is not None and not force:
            # running async actor method: cancel its coroutine
            try:
                t.cancel()
            except Exception:
                pass
            return {"cancelled": "async task cancelled"}
        if force:
            import threading as _t

            _t.Timer(0.05, lambda: os._exit(1)).start()  # let the RPC reply
            return {"cancelled": "force-killing worker"}
        with self._serial_cancel_lock:
            # the lock pins _running_serial across the check+inject so the
            # interrupt can only be pended while the task is still the
            # current one (the residual stale-delivery window is handled by
            # _serial_loop's KeyboardInterrupt guard)
            if self._running_serial == task_id:
                import ctypes

                ctypes.pythonapi.PyThreadState_SetAsyncExc(
                    ctypes.c_long(self._serial_thread.ident),
                    ctypes.py_object(KeyboardInterrupt))
                return {"cancelled": "interrupted running task"}
        return {"cancelled": "queued"}

    def request_exit(self):
        self._exit_requested = True
        self._maybe_exit()

    def _maybe_exit(self):
        with self._inflight_cv:
            busy = self._inflight > 0
        if not busy:
            try:
                if self.actor_instance is not None:
                    ray_terminate = getattr(self.actor_instance, "__ray_terminate__", None)
                    if ray_terminate:
                        ray_terminate()
            except Exception:
                pass
            os._exit(0)

    # -------------------------------------------------------------- execution
    def _is_async_method(self, name: str) -> bool:
        if self.actor_instance is None:
            return False
        m = getattr(type(self.actor_instance), name, None)
        return m is not None and (
            inspect.iscoroutinefunction(m)
            or inspect.isasyncgenfunction(m)
        )

    def _resolve_fn(self, payload):
        fn_id = payload.get("fn_id")
        if payload.get("fn") is not None:
            import cloudpickle

            fn = cloudpickle.loads(payload["fn"])
            if fn_id:
                self._fn_cache[fn_id] = fn
            return fn
        if fn_id in self._fn_cache:
            return self._fn_cache[fn_id]
        return None

    def _deserialize_args(self, payload):
        args, kwargs = serialization.deserialize(
            memoryview(payload["args"]), serialization.META_PICKLE
        )
        # top-level ObjectRef args are fetched before the call (reference
        # semantics: dependencies resolved by the worker before execution)
        def resolve(v):
            if isinstance(v, ObjectRef):
                return self.cw.get([v], None)[0]
            return v

        args = tuple(resolve(a) for a in args)
        kwargs = {k: resolve(v) for k, v in kwargs.items()}
        return args, kwargs

    def _run_and_reply(self, payload, done):
        try:
            reply = self._execute(payload)
        except BaseException as e:  # noqa: BLE001
            if (isinstance(e, KeyboardInterrupt)
                    and payload.get("task_id") in self._cancelled):
                # a soft-cancel interrupt can land anywhere in _execute —
                # including its preamble (e.g. the one-time logging-config
                # GCS fetch), outside the user-call try that normally maps
                # it. cancel_task only injects while THIS task is current,
                # so a KI here is this task's cancellation, not a crash.
                from ant_ray_amd.exceptions import TaskCancelledError

                reply = self._error_reply(payload, TaskCancelledError(
                    "task was cancelled while running"))
            else:
                logger.exception("task execution crashed")
                reply = self._error_reply(payload, e)
        done(reply)

    def _record_event(self, payload, t0, t1, ok: bool):
        """Task-event pipeline (parity: core_worker/task_event_buffer.cc ->
        GcsTaskManager). Fire-and-forget batched notify to the GCS."""
        try:
            gcs = getattr(self.cw, "gcs", None)
            if gcs is None:
                return
            name = payload.get("name") or payload.get("method") or ""
            if not name and payload.get("fn_id"):
                fn = self._fn_cache.get(payload["fn_id"])
                name = getattr(fn, "__name__", "")
            buf = getattr(self, "_event_buf", None)
            if buf is None:
                buf = self._event_buf = []
                self._event_flush_t = 0.0

                async def _periodic_flush():
                    while True:
                        await asyncio.sleep(1.0)
                        if self._event_buf:
                            events, self._event_buf = self._event_buf, []
                            try:
                                await gcs.notify("task_events",
                                                 {"events": events})
                            except Exception:
                                return

                self.cw.io.submit(_periodic_flush())
            buf.append({
                "task_id": payload["task_id"].hex(),
                "type": payload["type"],
                "name": name,
                "actor_id": (self.actor_id.hex() if self.actor_id else None),
                "state": "FINISHED" if ok else "FAILED",
                "start_ts": t0,
                "end_ts": t1,
                "pid": os.getpid(),
                "worker_id": getattr(self.cw, "worker_id", b"").hex()
                if isinstance(getattr(self.cw, "worker_id", None), bytes) else "",
            })
            now = time.monotonic()
            if len(buf) >= 100 or now - self._event_flush_t > 1.0:
                self._event_flush_t = now
                events, self._event_buf = buf, []
                self.cw.io.submit(gcs.notify("task_events", {"events": events}))
        except Exception:
            pass

    def _execute(self, payload) -> dict:
        if not getattr(self, "_logging_cfg_checked", False):
            # one-time re-check: a PRESTARTED worker boots before the
            # driver's ray.init(logging_config=...) kv_put lands; the first
            # task strictly follows init, so this read is race-free
            self._logging_cfg_checked = True
            try:
                from ant_ray_amd._private import logging_config as _lc

                if _lc._applied:
                    raise StopIteration  # boot-time read already applied it
                r = self.cw.io.run(self.cw.gcs.call(
                    "kv_get", {"ns": "_cluster", "key": b"logging_config"},
                    timeout=5), timeout=8)
                if r and r.get("value"):
                    import json as _json

                    from ant_ray_amd._private.logging_config import (
                        LoggingConfig,
                    )

                    LoggingConfig._from_dict(
                        _json.loads(r["value"].decode()))._apply()
            except Exception:
                pass
        ttype = payload["type"]
        task_id = payload["task_id"]
        if task_id in self._cancelled:
            from ant_ray_amd.exceptions import TaskCancelledError

            return self._error_reply(payload, TaskCancelledError(
                "task was cancelled before it started"))
        self.cw.current_task_id = task_id
        _t0 = self._last_t0 = time.time()
        try:
            if ttype == "normal":
                fn = self._resolve_fn(payload)
                if fn is None:
                    return {"status": "need_fn"}
                args, kwargs = self._deserialize_args(payload)
                result = fn(*args, **kwargs)
                reply = self._reply_results(payload, result)
                mc = int(payload.get("max_calls") or 0)
                if mc:
                    self._normal_calls = getattr(self, "_normal_calls", 0) + 1
                    if self._normal_calls >= mc:
                        reply["recycle"] = True
                        import threading as _t

                        _t.Timer(0.2, self.request_exit).start()
                return reply
            if ttype == "actor_create":
                import cloudpickle

                cls = cloudpickle.loads(payload["cls"])
                args, kwargs = self._deserialize_args(payload)
                self.max_concurrency = int(payload.get("max_concurrency", 1))
                if self.max_concurrency > 1:
                    self._pool = concurrent.futures.ThreadPoolExecutor(
                        max_workers=self.max_concurrency
                    )
                # named concurrency groups (parity: reference
                # concurrency_group_manager.cc — per-group executors so a
                # saturated group never blocks another's methods)
                self._groups = {}
                self._group_sems = {}
                for gname, gmax in (payload.get("concurrency_groups")
                                    or {}).items():
                    self._groups[gname] = concurrent.futures.ThreadPoolExecutor(
                        max_workers=int(gmax),
                        thread_name_prefix=f"cg-{gname}")
                    self._group_sem_sizes = getattr(
                        self, "_group_sem_sizes", {})
                    self._group_sem_sizes[gname] = int(gmax)
                if self._groups and self._pool is None:
                    # default group for un-annotated methods
                    self._pool = concurrent.futures.ThreadPoolExecutor(
                        max_workers=max(self.max_concurrency, 1))
                    self.max_concurrency = max(self.max_concurrency, 2)
                self.actor_id = payload.get("actor_id")
                self.cw.actor_id = self.actor_id
                self.actor_instance = cls(*args, **kwargs)
                has_async = any(
                    inspect.iscoroutinefunction(m)
                    for _, m in inspect.getmembers(cls, inspect.isfunction)
                )
                if has_async:
                    from ant_ray_amd._private.protocol import EventLoopThread

                    self._actor_loop = EventLoopThread(name="actor-async")
                    sem_size = self.max_concurrency if self.max_concurrency > 1 else 1000
                    self._async_sem = None
                    self._async_sem_size = sem_size
                return {"status": "ok", "results": []}
            if ttype == "actor_task":
                if self.actor_instance is None:
                    raise RuntimeError("no actor instance in this worker")
                if payload["method"] == "__adag_loop__":
                    (spec,), _ = self._deserialize_args(payload)
                    return self._reply_results(payload, self._adag_loop(spec))
                method = getattr(self.actor_instance, payload["method"])
                args, kwargs = self._deserialize_args(payload)
                result = method(*args, **kwargs)
                return self._reply_results(payload, result)
            raise ValueError(f"unknown task type {ttype}")
        except BaseException as e:  # noqa: BLE001
            from ant_ray_amd.actor import ActorExit

            if isinstance(e, ActorExit):
                # graceful self-exit (ray.actor.exit_actor): ack this call,
                # mark the death intentional (no restart), then leave once
                # the reply has drained
                self._record_event(payload, _t0, time.time(), ok=True)
                import threading as _t

                def _graceful_exit():
                    # after the reply has drained: mark the death
                    # intentional (no restart), then exit
                    if self.actor_id:
                        try:
                            self.cw.io.submit(self.cw.gcs.call(
                                "kill_actor",
                                {"actor_id": self.actor_id,
                                 "no_restart": True})).result(5)
                        except Exception:
                            pass
                    self.request_exit()

                _t.Timer(0.2, _graceful_exit).start()
                return {"status": "ok", "results": [
                    {"oid": ObjectID.for_return(
                        TaskID(task_id), i).binary(),
                     "inline": serialization.serialize(None).to_bytes(),
                     "meta": serialization.META_PICKLE}
                    for i in range(payload.get("n_returns", 1))]}
            self._record_event(payload, _t0, time.time(), ok=False)
            if (isinstance(e, KeyboardInterrupt)
                    and task_id in self._cancelled):
                from ant_ray_amd.exceptions import TaskCancelledError

                return self._error_reply(payload, TaskCancelledError(
                    "task was cancelled while running"))
            return self._error_reply(payload, e)
        finally:
            self.cw.current_task_id = None

    def _adag_loop(self, spec) -> int:
        """Resident compiled-DAG node loop (parity: the reference's
        do_exec_tasks loop in compiled_dag_node.py). Reads every input
        channel once per iteration, applies the bound method, writes the
        output channel; exits when an upstream channel closes and cascades
        the close downstream. Returns the iteration count."""
        from ant_ray_amd.experimental.channel import (
            ChannelClosedError,
            _project_input,
            _WrappedError,
        )

        ops = spec["ops"]
        methods = {op["key"]: getattr(self.actor_instance, op["method"])
                   for op in ops}
        iters = 0
        try:
            while True:
                values: dict = {}      # node key -> result (same-actor edges)
                chan_cache: dict = {}  # id(chan) -> value this iteration
                for op in ops:
                    args = []
                    for desc in op["ins"]:
                        if desc[0] == "local":
                            args.append(values[desc[1]])
                        elif desc[0] in ("chan", "chan_key"):
                            ch = desc[1]
                            if id(ch) not in chan_cache:
                                chan_cache[id(ch)] = ch.read(unwrap=False)
                            v = chan_cache[id(ch)]
                            if desc[0] == "chan_key" \
                                    and not isinstance(v, _WrappedError):
                                v = _project_input(v, desc[2])
                            args.append(v)
                        else:
                            args.append(desc[1])
                    bad = next((a for a in args
                                if isinstance(a, _WrappedError)), None)
                    if bad is not None:
                        result = bad  # forward upstream errors unchanged
                    else:
                        try:
                            result = methods[op["key"]](*args)
                        except ChannelClosedError:
                            raise
                        except BaseException as e:  # noqa: BLE001
                            result = _WrappedError(e)
                    values[op["key"]] = result
                    if op["out"] is not None:
                        op["out"].write(result)
                iters += 1
        except ChannelClosedError:
            for op in ops:
                if op["out"] is not None:
                    op["out"].close()
            return iters

    def _run_async_actor_task(self, payload, done):
        group = payload.get("concurrency_group")
        task_id = payload.get("task_id")

        async def runner():
            if self._async_sem is None:
                self._async_sem = asyncio.Semaphore(self._async_sem_size)
            sem = self._async_sem
            sizes = getattr(self, "_group_sem_sizes", {})
            if group in sizes:
                sems = getattr(self, "_async_group_sems", None)
                if sems is None:
                    sems = self._async_group_sems = {}
                if group not in sems:
                    sems[group] = asyncio.Semaphore(sizes[group])
                sem = sems[group]
            async with sem:
                try:
                    if task_id in self._cancelled:
                        raise asyncio.CancelledError
                    method = getattr(self.actor_instance, payload["method"])
                    args, kwargs = self._deserialize_args(payload)
                    result = await method(*args, **kwargs)
                    reply = self._reply_results(payload, result)
                except asyncio.CancelledError:
                    from ant_ray_amd.exceptions import TaskCancelledError

                    reply = self._error_reply(payload, TaskCancelledError(
                        "actor task was cancelled"))
                except BaseException as e:  # noqa: BLE001
                    reply = self._error_reply(payload, e)
                finally:
                    if not hasattr(self, "_async_tasks"):
                        self._async_tasks = {}
                    self._async_tasks.pop(task_id, None)
                done(reply)

        t = self._actor_loop.submit(runner())
        if not hasattr(self, "_async_tasks"):
            self._async_tasks = {}
        self._async_tasks[task_id] = t

    # ---------------------------------------------------------------- replies
    def _store_one_return(self, payload, i: int, v) -> dict:
        """Serialize + place return value i; returns the result entry."""
        transport = payload.get("tensor_transport")
        oid = ObjectID.for_return(TaskID(payload["task_id"]), i).binary()
        if transport == "hip_ipc":
            with serialization.gpu_transport_context("hip_ipc") as gctx:
                sobj = serialization.serialize(v)
            if gctx.pinned:
                from ant_ray_amd.experimental.gpu_object_manager import (
                    gpu_object_store,
                )

                gpu_object_store.add(oid, gctx.pinned)
        else:
            sobj = serialization.serialize(v)
        self.cw._register_escapes(sobj)
        if sobj.total_size <= INLINE_OBJECT_MAX:
            entry = {"oid": oid, "inline": sobj.to_bytes(), "meta": sobj.metadata}
            if transport == "hip_ipc":
                # the device tensors stay pinned here; record us as holder
                # so frees reach this process
                entry["holder"] = list(self.cw.addr)
            return entry
        self.cw.store.put_serialized_to_shm(oid, sobj)
        return {"oid": oid, "inline": None, "holder": list(self.cw.addr)}

    def _reply_results(self, payload, result) -> dict:
        self._record_event(payload, getattr(self, '_last_t0', time.time()),
                           time.time(), ok=True)
        if payload.get("streaming"):
            return self._stream_results(payload, result)
        n = payload.get("n_returns", 1)
        if n == 0:
            return {"status": "ok", "results": []}
        values = (result,) if n == 1 else tuple(result)
        if n > 1 and len(values) != n:
            raise ValueError(f"task returned {len(values)} values, expected {n}")
        out = [self._store_one_return(payload, i, v)
               for i, v in enumerate(values)]
        return {"status": "ok", "results": out}

    def _stream_results(self, payload, result) -> dict:
        """num_returns='streaming' (parity: reference streaming generators /
        ObjectRefGenerator, core_worker TaskManager dynamic returns): the
        task body is a generator; each yielded value becomes return object
        i, pushed to the caller the moment it is produced via a
        stream_item notify, so the consumer overlaps with the producer."""
        if not hasattr(result, "__next__"):
            result = iter(
                result if hasattr(result, "__iter__") else (result,))
        caller_addr = tuple(payload["caller_addr"])
        task_id = payload["task_id"]
        count = 0
        for v in result:
            entry = self._store_one_return(payload, count, v)
            count += 1

            async def _send(entry=entry, idx=count - 1):
                conn = await self.cw._get_worker_conn_async_cached(caller_addr)
                await conn.notify("stream_item", {
                    "task_id": task_id, "index": idx, "entry": entry})

            # sent in order on the io loop; failures surface on the final
            # reply path (caller gone -> push reply fails too)
            self.cw.io.submit(_send()).result(60)
        return {"status": "ok", "results": [], "streaming_done": count}

    def _error_reply(self, payload, exc: BaseException) -> dict:
        if isinstance(exc, RayTaskError):
            err = exc
        else:
            name = payload.get("method") or payload.get("name") or "task"
            err = RayTaskError.from_exception(exc, name)
        sobj = serialization.serialize_error(err)
        return {
            "status": "error",
            "error": str(exc),
            "error_payload": sobj.to_bytes(),
            "error_meta": sobj.metadata,
        }
