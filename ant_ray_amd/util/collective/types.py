"""Types for ray.util.collective parity (reference
python/ray/util/collective/types.py:34 — Backend enum has NCCL+GLOO only;
on MI355X the "nccl" backend IS RCCL via torch.distributed)."""
from __future__ import annotations

from enum import Enum


class Backend(str, Enum):
    NCCL = "nccl"   # RCCL on ROCm
    RCCL = "nccl"   # alias: same backend
    GLOO = "gloo"

    @classmethod
    def parse(cls, v):
        if isinstance(v, Backend):
            return v
        v = str(v).lower()
        if v in ("nccl", "rccl"):
            return cls.NCCL
        if v == "gloo":
            return cls.GLOO
        raise ValueError(f"unsupported collective backend {v}")


class ReduceOp(Enum):
    SUM = "sum"
    PRODUCT = "product"
    MIN = "min"
    MAX = "max"


def torch_reduce_op(op: ReduceOp):
    import torch.distributed as dist

    return {
        ReduceOp.SUM: dist.ReduceOp.SUM,
        ReduceOp.PRODUCT: dist.ReduceOp.PRODUCT,
        ReduceOp.MIN: dist.ReduceOp.MIN,
        ReduceOp.MAX: dist.ReduceOp.MAX,
    }[op]
