from ant_ray_amd.util.collective.collective import (  # noqa: F401
    allgather,
    allgather_multigpu,
    allreduce,
    allreduce_multigpu,
    barrier,
    broadcast,
    create_collective_group,
    destroy_collective_group,
    get_rank,
    get_collective_group_size,
    init_collective_group,
    is_group_initialized,
    recv,
    reduce,
    reducescatter,
    send,
)
from ant_ray_amd.util.collective.types import Backend, ReduceOp  # noqa: F401
