"""Tensor parallelism: column/row-parallel linear layers over a
process group (reference delegates TP to vLLM — SURVEY.md §2.10; here
it is native, for use over RCCL/xGMI groups)."""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn


def _pg(group):
    if group is not None and hasattr(group, "pg"):
        return group.pg
    return group


def _world(group):
    g = _pg(group)
    return dist.get_world_size() if g is None else g.size()


def _rank(group):
    g = _pg(group)
    return dist.get_rank() if g is None else g.rank()


def _all_reduce(t, group):
    g = _pg(group)
    if g is None:
        dist.all_reduce(t)
    else:
        g.allreduce([t]).wait()
    return t


def _all_gather_cat(t, group, dim):
    g = _pg(group)
    world = _world(group)
    outs = [torch.empty_like(t) for _ in range(world)]
    if g is None:
        dist.all_gather(outs, t.contiguous())
    else:
        g.allgather([outs], [t.contiguous()]).wait()
    return torch.cat(outs, dim=dim)


class ColumnParallelLinear(nn.Module):
    """Y = X @ W^T with W row-sharded over ranks (output features
    split). gather_output concatenates shards (allgather)."""

    def __init__(self, in_features: int, out_features: int, *, group=None,
                 bias: bool = False, gather_output: bool = True,
                 dtype=torch.float32):
        super().__init__()
        self.group = group
        world = _world(group)
        assert out_features % world == 0
        self.out_per_rank = out_features // world
        self.weight = nn.Parameter(
            torch.empty(self.out_per_rank, in_features, dtype=dtype)
        )
        nn.init.normal_(self.weight, std=0.02)
        self.bias = (
            nn.Parameter(torch.zeros(self.out_per_rank, dtype=dtype))
            if bias
            else None
        )
        self.gather_output = gather_output

    def forward(self, x):
        y = torch.nn.functional.linear(x, self.weight, self.bias)
        if self.gather_output and _world(self.group) > 1:
            y = _all_gather_cat(y, self.group, dim=-1)
        return y

    @staticmethod
    def shard_from(full_weight: torch.Tensor, group=None) -> torch.Tensor:
        world, rank = _world(group), _rank(group)
        return full_weight.chunk(world, dim=0)[rank].clone()


class RowParallelLinear(nn.Module):
    """Y = X @ W^T with W column-sharded (input features split); each
    rank consumes its input shard and the partial outputs are
    all-reduced."""

    def __init__(self, in_features: int, out_features: int, *, group=None,
                 bias: bool = False, input_is_parallel: bool = True,
                 dtype=torch.float32):
        super().__init__()
        self.group = group
        world = _world(group)
        assert in_features % world == 0
        self.in_per_rank = in_features // world
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_per_rank, dtype=dtype)
        )
        nn.init.normal_(self.weight, std=0.02)
        self.bias = (
            nn.Parameter(torch.zeros(out_features, dtype=dtype))
            if bias
            else None
        )
        self.input_is_parallel = input_is_parallel

    def forward(self, x):
        if not self.input_is_parallel:
            rank = _rank(self.group)
            x = x.chunk(_world(self.group), dim=-1)[rank]
        y = torch.nn.functional.linear(x, self.weight)
        if _world(self.group) > 1:
            y = _all_reduce(y, self.group)
        if self.bias is not None:
            y = y + self.bias
        return y

    @staticmethod
    def shard_from(full_weight: torch.Tensor, group=None) -> torch.Tensor:
        world, rank = _world(group), _rank(group)
        return full_weight.chunk(world, dim=1)[rank].clone()
