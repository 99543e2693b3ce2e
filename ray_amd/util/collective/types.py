"""Collective types (reference: python/ray/util/collective/types.py)."""
from __future__ import annotations

from enum import Enum


class Backend:
    RCCL = "rccl"          # torch.distributed "nccl" == RCCL on ROCm
    NCCL = "rccl"          # alias: reference API name maps to RCCL here
    TORCH_GLOO = "torch_gloo"
    GLOO = "torch_gloo"


class ReduceOp(Enum):
    SUM = "sum"
    PRODUCT = "product"
    MIN = "min"
    MAX = "max"


def torch_reduce_op(op: "ReduceOp"):
    import torch.distributed as dist

    return {
        ReduceOp.SUM: dist.ReduceOp.SUM,
        ReduceOp.PRODUCT: dist.ReduceOp.PRODUCT,
        ReduceOp.MIN: dist.ReduceOp.MIN,
        ReduceOp.MAX: dist.ReduceOp.MAX,
    }[op]
