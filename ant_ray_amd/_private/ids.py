"""Binary IDs for jobs/tasks/actors/objects.

Role parity with the reference's id scheme (reference: src/ray/common/id.h:624
— JobID/ActorID/TaskID/ObjectID with embedded lineage). We keep the same sizes
(ObjectID = 20 bytes etc.) so the wire format and shm-store keys are compact,
but derive object ids from (task id, return index) with a hash rather than
bit-packed lineage fields: lineage metadata lives in the owner's task table
instead of inside the id.
"""
from __future__ import annotations

import hashlib
import os


class BaseID:
    SIZE = 20
    __slots__ = ("_bin",)

    def __init__(self, binary: bytes):
        if len(binary) != self.SIZE:
            raise ValueError(
                f"{type(self).__name__} must be {self.SIZE} bytes, got {len(binary)}"
            )
        self._bin = binary

    @classmethod
    def from_random(cls):
        return cls(os.urandom(cls.SIZE))

    @classmethod
    def from_hex(cls, hex_str: str):
        return cls(bytes.fromhex(hex_str))

    @classmethod
    def nil(cls):
        return cls(b"\xff" * cls.SIZE)

    def is_nil(self) -> bool:
        return self._bin == b"\xff" * self.SIZE

    def binary(self) -> bytes:
        return self._bin

    def hex(self) -> str:
        return self._bin.hex()

    def __hash__(self):
        return hash(self._bin)

    def __eq__(self, other):
        return type(other) is type(self) and other._bin == self._bin

    def __repr__(self):
        return f"{type(self).__name__}({self._bin.hex()})"


class UniqueID(BaseID):
    pass


class JobID(BaseID):
    SIZE = 4

    @classmethod
    def from_int(cls, value: int):
        return cls(value.to_bytes(4, "little"))

    def int(self) -> int:
        return int.from_bytes(self._bin, "little")


class WorkerID(BaseID):
    SIZE = 20


class NodeID(BaseID):
    SIZE = 20


class ActorID(BaseID):
    SIZE = 16

    @classmethod
    def of(cls, job_id: JobID):
        return cls(os.urandom(12) + job_id.binary())

    def job_id(self) -> JobID:
        return JobID(self._bin[12:16])


class TaskID(BaseID):
    SIZE = 16

    # process-unique 6-byte prefix + per-process counter: os.urandom is a
    # ~37us syscall and for_task sits on the task-submission hot path
    # (ray_perf "tasks async" shape); collision across processes needs a
    # 48-bit prefix clash AND the same counter value
    _prefix = os.urandom(6)
    _counter = 0
    _lock = None  # lazily bound (threading import kept off the module top)

    @staticmethod
    def _refresh_prefix():
        TaskID._prefix = os.urandom(6)
        TaskID._counter = 0

    @classmethod
    def for_task(cls, job_id: JobID):
        if cls._lock is None:
            import threading

            cls._lock = threading.Lock()
        with cls._lock:
            cls._counter += 1
            c = cls._counter
        return cls(cls._prefix + c.to_bytes(6, "little") + job_id.binary())

    @classmethod
    def for_actor_task(cls, actor_id: ActorID, seq: int,
                       caller: bytes = b""):
        # caller MUST be mixed in: two processes each count their own
        # calls from 1, and colliding task ids would collide the
        # deterministic return-object ids in the node-shared shm store
        h = hashlib.blake2b(
            actor_id.binary() + caller + seq.to_bytes(8, "little"),
            digest_size=12,
        ).digest()
        return cls(h + actor_id.job_id().binary())

    def job_id(self) -> JobID:
        return JobID(self._bin[12:16])


class ObjectID(BaseID):
    SIZE = 20

    @classmethod
    def for_return(cls, task_id: TaskID, index: int):
        # deterministic: owner and executor derive the same id
        h = hashlib.blake2b(
            task_id.binary() + index.to_bytes(4, "little"), digest_size=16
        ).digest()
        return cls(h + task_id.job_id().binary())

    @classmethod
    def for_put(cls, worker_id: WorkerID, put_index: int):
        h = hashlib.blake2b(
            worker_id.binary() + b"put" + put_index.to_bytes(8, "little"),
            digest_size=20,
        ).digest()
        return cls(h)


class PlacementGroupID(BaseID):
    SIZE = 16


# a forked child inherits TaskID._prefix/_counter; without a refresh its
# task ids would collide with the parent's
if hasattr(os, "register_at_fork"):
    os.register_at_fork(after_in_child=TaskID._refresh_prefix)
