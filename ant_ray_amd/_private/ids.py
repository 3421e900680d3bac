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

    @classmethod
    def for_task(cls, job_id: JobID):
        return cls(os.urandom(12) + job_id.binary())

    @classmethod
    def for_actor_task(cls, actor_id: ActorID, seq: int):
        h = hashlib.blake2b(
            actor_id.binary() + seq.to_bytes(8, "little"), digest_size=12
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
