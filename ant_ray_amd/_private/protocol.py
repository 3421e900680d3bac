"""Wire protocol: length-framed msgpack RPC over asyncio streams.

Role parity: the reference's control plane is gRPC services + a
flatbuffers-over-unix-socket worker<->raylet IPC (reference:
src/ray/rpc/grpc_server.h, src/ray/raylet_ipc_client/raylet_ipc_client.h:51).
We use one uniform transport instead: 4-byte-length-framed msgpack messages
over TCP/unix sockets with asyncio. Python-level RPC over gRPC costs ~100us+
per call in protobuf encode/decode; msgpack framing keeps the per-call
overhead low enough for the tasks/sec targets in BASELINE.md.

Message format (msgpack arrays):
    [0, seq, method, payload]   request
    [1, seq, payload]           reply (ok)
    [2, seq, err_string]        reply (error)
    [3, 0,  method, payload]    one-way notification
"""
from __future__ import annotations

import asyncio
import os
import logging
import struct
import threading
from typing import Any, Awaitable, Callable, Dict, Optional, Tuple

import msgpack

logger = logging.getLogger(__name__)

REQUEST, REPLY, ERROR, NOTIFY, HELLO, HELLO_NAK = 0, 1, 2, 3, 4, 5
_LEN = struct.Struct("<I")

# Wire-protocol version (parity: the reference's proto-schema'd messages
# carry versioned service definitions; here every connection opens with a
# HELLO frame [4, 0, {"v": version, "token": token}] and the server
# rejects mismatched versions or bad auth tokens with HELLO_NAK + close —
# token auth parity: reference src/ray/rpc/authentication/).
PROTOCOL_VERSION = 1


def auth_token():
    """Cluster auth token (None = auth disabled). Set RAY_AUTH_TOKEN on
    the head; spawned components inherit it through the environment."""
    return os.environ.get("RAY_AUTH_TOKEN") or None

# Handlers receive (connection, method, payload) and return the reply payload.
Handler = Callable[["Connection", str, Any], Awaitable[Any]]


def pack(msg) -> bytes:
    return msgpack.packb(msg, use_bin_type=True)


def unpack(data: bytes):
    return msgpack.unpackb(data, raw=False, strict_map_key=False)


# Chaos injection (parity: src/ray/common/asio/asio_chaos.cc —
# RAY_testing_asio_delay_us="method=min:max,method2=...,*=min:max" injects a
# uniform-random handler delay; used by the fault-injection test suite).
_chaos_spec = None


def _parse_chaos():
    global _chaos_spec
    raw = os.environ.get("RAY_testing_asio_delay_us", "")
    spec = {}
    for part in raw.split(","):
        if "=" not in part:
            continue
        name, rng = part.split("=", 1)
        try:
            lo, hi = rng.split(":")
            spec[name.strip()] = (int(lo), int(hi))
        except ValueError:
            continue
    _chaos_spec = spec


async def _chaos_delay(method: str):
    if _chaos_spec is None:
        _parse_chaos()
    if not _chaos_spec:
        return
    rng = _chaos_spec.get(method) or _chaos_spec.get("*")
    if rng:
        import random

        await asyncio.sleep(random.uniform(rng[0], rng[1]) / 1e6)


class RpcError(Exception):
    pass


class ConnectionLost(Exception):
    pass


class Connection:
    """A bidirectional framed-msgpack channel. Both sides can issue requests."""

    def __init__(
        self,
        reader: asyncio.StreamReader,
        writer: asyncio.StreamWriter,
        handler: Optional[Handler] = None,
        name: str = "",
    ):
        self.reader = reader
        self.writer = writer
        self.handler = handler
        self.name = name
        self._seq = 0
        self._pending: Dict[int, asyncio.Future] = {}
        self._closed = False
        self._recv_task: Optional[asyncio.Task] = None
        self.on_close: Optional[Callable[["Connection"], None]] = None
        # opaque slot for the server to attach session state
        self.session: Dict[str, Any] = {}
        self._send_lock = asyncio.Lock()

    def start(self):
        self._recv_task = asyncio.get_running_loop().create_task(self._recv_loop())
        return self

    @property
    def closed(self):
        return self._closed

    async def _recv_loop(self):
        try:
            while True:
                hdr = await self.reader.readexactly(4)
                (n,) = _LEN.unpack(hdr)
                data = await self.reader.readexactly(n)
                msg = unpack(data)
                kind = msg[0]
                if kind == REQUEST:
                    asyncio.get_running_loop().create_task(
                        self._handle_request(msg[1], msg[2], msg[3])
                    )
                elif kind == REPLY:
                    fut = self._pending.pop(msg[1], None)
                    if fut is not None and not fut.done():
                        fut.set_result(msg[2])
                elif kind == ERROR:
                    fut = self._pending.pop(msg[1], None)
                    if fut is not None and not fut.done():
                        fut.set_exception(RpcError(msg[2]))
                elif kind == NOTIFY:
                    asyncio.get_running_loop().create_task(
                        self._handle_notify(msg[2], msg[3])
                    )
                elif kind == HELLO:
                    info = msg[2] or {}
                    if info.get("v") != PROTOCOL_VERSION:
                        await self.send([HELLO_NAK, 0,
                                         f"protocol version mismatch: peer "
                                         f"{info.get('v')} != "
                                         f"{PROTOCOL_VERSION}"])
                        break
                    tok = auth_token()
                    if tok is not None and info.get("token") != tok:
                        await self.send([HELLO_NAK, 0,
                                         "authentication failed: bad or "
                                         "missing RAY_AUTH_TOKEN"])
                        break
                    self.session["hello"] = info
                elif kind == HELLO_NAK:
                    self._nak_reason = msg[2]
                    logger.error("connection %s rejected by peer: %s",
                                 self.name, msg[2])
                    break
        except (
            asyncio.IncompleteReadError,
            ConnectionResetError,
            BrokenPipeError,
            OSError,
        ):
            pass
        except Exception:
            logger.exception("connection %s recv loop error", self.name)
        finally:
            await self._do_close()

    async def _handle_request(self, seq, method, payload):
        try:
            await _chaos_delay(method)
            result = await self.handler(self, method, payload)
            await self.send([REPLY, seq, result])
        except Exception as e:  # noqa: BLE001 - forwarded to caller
            logger.debug("rpc handler %s error: %s", method, e, exc_info=True)
            try:
                await self.send([ERROR, seq, f"{type(e).__name__}: {e}"])
            except Exception:
                pass

    async def _handle_notify(self, method, payload):
        try:
            await self.handler(self, method, payload)
        except Exception:
            logger.exception("notify handler %s failed", method)

    async def send(self, msg):
        data = pack(msg)
        async with self._send_lock:
            if self._closed:
                raise ConnectionLost(self.name)
            self.writer.write(_LEN.pack(len(data)) + data)
            await self.writer.drain()

    async def call(self, method: str, payload: Any = None, timeout: float = None):
        self._seq += 1
        seq = self._seq
        fut = asyncio.get_running_loop().create_future()
        self._pending[seq] = fut
        await self.send([REQUEST, seq, method, payload])
        if timeout is not None:
            return await asyncio.wait_for(fut, timeout)
        return await fut

    async def notify(self, method: str, payload: Any = None):
        await self.send([NOTIFY, 0, method, payload])

    async def _do_close(self):
        if self._closed:
            return
        self._closed = True
        reason = getattr(self, "_nak_reason", None)
        for fut in self._pending.values():
            if not fut.done():
                fut.set_exception(RpcError(reason) if reason
                                  else ConnectionLost(self.name))
                # abandoned awaiters (cancelled mid-call) are expected on a
                # lost conn; mark retrieved so GC doesn't log a warning
                fut.exception()
        self._pending.clear()
        try:
            self.writer.close()
        except Exception:
            pass
        if self.on_close:
            try:
                self.on_close(self)
            except Exception:
                logger.exception("on_close callback failed")

    async def close(self):
        await self._do_close()
        if self._recv_task:
            self._recv_task.cancel()


async def connect(
    addr: Tuple[str, int] | str, handler: Handler = None, name: str = ""
) -> Connection:
    """addr: (host, port) for TCP or a string path for a unix socket."""
    if isinstance(addr, str):
        reader, writer = await asyncio.open_unix_connection(addr)
    else:
        reader, writer = await asyncio.open_connection(addr[0], addr[1])
        writer.get_extra_info("socket").setsockopt(
            __import__("socket").IPPROTO_TCP, __import__("socket").TCP_NODELAY, 1
        )
    conn = Connection(reader, writer, handler, name=name)
    conn.start()
    await conn.send([HELLO, 0, {"v": PROTOCOL_VERSION,
                                "token": auth_token()}])
    return conn


async def serve(
    handler: Handler,
    host: str = "127.0.0.1",
    port: int = 0,
    unix_path: str = None,
    on_connect=None,
):
    """Start a server; returns (server, bound_port)."""
    conns = set()

    async def _on_client(reader, writer):
        try:
            sock = writer.get_extra_info("socket")
            if sock is not None and not unix_path:
                import socket as _s

                sock.setsockopt(_s.IPPROTO_TCP, _s.TCP_NODELAY, 1)
        except Exception:
            pass
        conn = Connection(reader, writer, handler, name="server-conn")
        conns.add(conn)
        conn.on_close = lambda c: conns.discard(c)
        if on_connect:
            on_connect(conn)
        conn.start()

    if unix_path:
        server = await asyncio.start_unix_server(_on_client, path=unix_path)
        return server, unix_path
    server = await asyncio.start_server(_on_client, host, port)
    bound = server.sockets[0].getsockname()[1]
    return server, bound


class EventLoopThread:
    """A dedicated asyncio loop in a daemon thread.

    The driver/worker public API is synchronous; all networking runs here
    (parity with the reference's io_service threads in CoreWorkerProcess).
    """

    def __init__(self, name="antray-io"):
        self.loop = asyncio.new_event_loop()
        self._thread = threading.Thread(target=self._run, name=name, daemon=True)
        self._started = threading.Event()
        self._thread.start()
        self._started.wait()

    def _run(self):
        asyncio.set_event_loop(self.loop)
        self.loop.call_soon(self._started.set)
        self.loop.run_forever()

    def run(self, coro, timeout=None):
        """Run a coroutine on the loop from another thread, synchronously."""
        fut = asyncio.run_coroutine_threadsafe(coro, self.loop)
        return fut.result(timeout)

    def submit(self, coro):
        """Schedule without waiting; returns concurrent.futures.Future."""
        return asyncio.run_coroutine_threadsafe(coro, self.loop)

    def stop(self):
        try:
            def _cancel_then_stop():
                # cancel recv loops etc. BEFORE stopping, so tasks unwind
                # inside the loop instead of warning at GC time
                for t in asyncio.all_tasks(self.loop):
                    t.cancel()
                self.loop.call_soon(self.loop.stop)

            self.loop.call_soon_threadsafe(_cancel_then_stop)
            self._thread.join(timeout=2)
            if not self._thread.is_alive():
                self.loop.close()
        except Exception:
            pass
