"""Tensor-parallel building blocks (Megatron-style f/g pattern) over RCCL.

ColumnParallelLinear shards the OUTPUT features, RowParallelLinear the
INPUT features; a Column(no-gather) -> elementwise -> Row pair costs ONE
all-reduce per direction — the right shape for xGMI, where each extra
collective pays the per-link ring cost (SURVEY.md §5 'Distributed
communication backend').

The flagship bench remains pure DP (one 8B replica fits a 288 GB MI355X
with room); these blocks are the TP substrate the launcher can form from
the injected env (parallel/groups.py) for models that outgrow one GPU.
Numerics are cross-checked against unsharded references over gloo in
tests/test_tp_gloo.py.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn


def _group_size(group) -> int:
    if not dist.is_initialized():
        return 1
    return dist.get_world_size(group)


class _CopyToTP(torch.autograd.Function):
    """f: identity forward; all-reduce the gradient (input is replicated
    across the TP group, so its grads sum)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, gx):
        if _group_size(ctx.group) > 1:
            gx = gx.contiguous()
            dist.all_reduce(gx, group=ctx.group)
        return gx, None


class _ReduceFromTP(torch.autograd.Function):
    """g: all-reduce forward (sum partials); identity gradient."""

    @staticmethod
    def forward(ctx, x, group):
        if _group_size(group) > 1:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, gx):
        return gx, None


class _GatherFromTP(torch.autograd.Function):
    """all-gather shards along the last dim; backward takes this rank's
    slice (and no reduction — each shard's grad flows to its owner)."""

    @staticmethod
    def forward(ctx, x, group):
        n = _group_size(group)
        ctx.group = group
        ctx.n = n
        if n == 1:
            return x
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(n)]
        dist.all_gather(parts, x, group=group)
        return torch.cat(parts, dim=-1)

    @staticmethod
    def backward(ctx, gx):
        if ctx.n == 1:
            return gx, None
        shard = gx.shape[-1] // ctx.n
        r = dist.get_rank(ctx.group)
        return gx[..., r * shard:(r + 1) * shard].contiguous(), None


class ColumnParallelLinear(nn.Module):
    """Y = X @ W^T with W row-sharded over TP: each rank holds
    out_features/tp rows and computes its slice of Y."""

    def __init__(self, in_features: int, out_features: int, group=None,
                 gather_output: bool = False, dtype=None,
                 copy_input: bool = True):
        super().__init__()
        self.group = group
        n = _group_size(group)
        assert out_features % n == 0, (out_features, n)
        self.out_per_rank = out_features // n
        self.gather_output = gather_output
        # copy_input=False under sequence parallelism: the seq all-gather
        # seam already sums the input gradient across the group, so the
        # f-op's backward all-reduce would double-count (Megatron drops it
        # the same way when sequence_parallel is on).
        self.copy_input = copy_input
        self.weight = nn.Parameter(
            torch.empty(self.out_per_rank, in_features, dtype=dtype))
        self.weight.tp_sharded = True  # distinct shard per tp rank

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.copy_input:
            x = _CopyToTP.apply(x, self.group)
        y = nn.functional.linear(x, self.weight)
        if self.gather_output:
            y = _GatherFromTP.apply(y, self.group)
        return y


class RowParallelLinear(nn.Module):
    """Y = X @ W^T with W column-sharded over TP: input arrives SHARDED on
    the last dim (a ColumnParallelLinear's ungathered output); the partial
    products all-reduce into the full Y."""

    def __init__(self, in_features: int, out_features: int, group=None,
                 dtype=None, reduce_output: bool = True):
        super().__init__()
        self.group = group
        n = _group_size(group)
        assert in_features % n == 0, (in_features, n)
        self.in_per_rank = in_features // n
        # reduce_output=False under sequence parallelism: the partial sums
        # flow into a seq reduce-scatter instead of the all-reduce.
        self.reduce_output = reduce_output
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_per_rank, dtype=dtype))
        self.weight.tp_sharded = True  # distinct shard per tp rank

    def forward(self, x_shard: torch.Tensor) -> torch.Tensor:
        partial = nn.functional.linear(x_shard, self.weight)
        if not self.reduce_output:
            return partial
        return _ReduceFromTP.apply(partial, self.group)


def shard_from(full_weight: torch.Tensor, dim: int, group) -> torch.Tensor:
    """This rank's slice of an unsharded weight (for tests/loading)."""
    n = _group_size(group)
    if n == 1:
        return full_weight
    r = dist.get_rank(group)
    return full_weight.chunk(n, dim=dim)[r].contiguous()
