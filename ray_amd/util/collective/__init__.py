from .collective import (  # noqa: F401
    allgather,
    allreduce,
    barrier,
    broadcast,
    create_collective_group,
    destroy_collective_group,
    get_rank,
    get_collective_group_size,
    init_collective_group,
    recv,
    reduce,
    reducescatter,
    send,
)
from .types import Backend, ReduceOp  # noqa: F401
