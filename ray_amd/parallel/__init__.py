"""ray_amd.parallel — parallelism substrates beyond DP.

SURVEY.md §5.7: the reference implements none of ring-attention /
Ulysses / TP itself (delegated to vLLM/DeepSpeed); ray_amd provides
them natively on the RCCL ring (xGMI is 7 point-to-point links per GPU,
so ring attention's one-neighbor K/V pass maps onto a single dedicated
link while block attention computes).
"""
from .sequence import ring_attention, ulysses_all_to_all  # noqa: F401
from .tensor import (  # noqa: F401
    ColumnParallelLinear,
    RowParallelLinear,
)
