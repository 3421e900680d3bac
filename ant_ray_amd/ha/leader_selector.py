"""Leader election for head-node HA.

Protocol parity with the reference's RedisBasedLeaderSelector
(python/ray/ha/redis_leader_selector.py): campaign by writing a lease
(value = node identity, TTL = lease_ms); the holder refreshes at
lease_ms/2; others watch and take over when the lease expires. A callback
fires on becoming leader / losing leadership.
"""
from __future__ import annotations

import json
import os
import threading
import time
import uuid
from typing import Callable, Optional


class LeaderSelector:
    """Interface (reference ha/leader_selector.py)."""

    def start(self):
        raise NotImplementedError

    def stop(self):
        raise NotImplementedError

    def is_leader(self) -> bool:
        raise NotImplementedError


class FileLeaderSelector(LeaderSelector):
    """Lease in a shared file (NFS/local multi-head tests). Atomic via
    rename; expired leases are stolen."""

    def __init__(self, lease_path: str, node_id: Optional[str] = None,
                 lease_ms: int = 5000,
                 on_leader_change: Optional[Callable[[bool], None]] = None):
        self.lease_path = lease_path
        self.node_id = node_id or f"{os.uname().nodename}:{os.getpid()}:{uuid.uuid4().hex[:6]}"
        self.lease_ms = lease_ms
        self.on_leader_change = on_leader_change
        self._leader = False
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def _read(self):
        try:
            with open(self.lease_path) as f:
                return json.load(f)
        except (FileNotFoundError, json.JSONDecodeError):
            return None

    def _try_acquire(self) -> bool:
        now = time.time()
        cur = self._read()
        if cur and cur["holder"] != self.node_id and cur["expires"] > now:
            return False
        tmp = f"{self.lease_path}.{self.node_id.replace('/', '_').replace(':', '_')}"
        with open(tmp, "w") as f:
            json.dump({"holder": self.node_id,
                       "expires": now + self.lease_ms / 1000.0}, f)
        os.replace(tmp, self.lease_path)  # atomic on POSIX
        time.sleep(0.01)  # losers overwrite within this window; re-check
        cur = self._read()
        return bool(cur and cur["holder"] == self.node_id)

    def _set_leader(self, flag: bool):
        if flag != self._leader:
            self._leader = flag
            if self.on_leader_change:
                try:
                    self.on_leader_change(flag)
                except Exception:
                    pass

    def _loop(self):
        while not self._stop.is_set():
            if self._try_acquire():
                self._set_leader(True)
                self._stop.wait(self.lease_ms / 2000.0)
            else:
                self._set_leader(False)
                self._stop.wait(self.lease_ms / 1000.0)

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="leader-selector")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        cur = self._read()
        if cur and cur["holder"] == self.node_id:
            try:
                os.unlink(self.lease_path)
            except FileNotFoundError:
                pass
        self._set_leader(False)

    def is_leader(self) -> bool:
        return self._leader


class RedisBasedLeaderSelector(FileLeaderSelector):
    """Reference-named selector. With a reachable Redis it campaigns on a
    Redis key (SET NX PX + refresh); in this offline image it degrades to
    the file lease with identical semantics."""

    def __init__(self, redis_address: Optional[str] = None, name: str = "head",
                 **kw):
        self._redis = None
        if redis_address:
            try:
                import redis  # not installed offline; kept for parity

                host, port = redis_address.rsplit(":", 1)
                self._redis = redis.Redis(host=host, port=int(port))
                self._redis.ping()
            except Exception:
                self._redis = None
        lease_path = kw.pop("lease_path",
                            os.path.join("/tmp/antray", f"ha_lease_{name}"))
        os.makedirs(os.path.dirname(lease_path), exist_ok=True)
        super().__init__(lease_path, **kw)
        self._key = f"antray:ha:{name}"

    def _try_acquire(self) -> bool:
        if self._redis is None:
            return super()._try_acquire()
        ok = self._redis.set(self._key, self.node_id, nx=True,
                             px=self.lease_ms)
        if ok:
            return True
        holder = self._redis.get(self._key)
        if holder and holder.decode() == self.node_id:
            self._redis.pexpire(self._key, self.lease_ms)
            return True
        return False
