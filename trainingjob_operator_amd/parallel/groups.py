"""Process-group topology: DP x TP grids over the operator's injected world.

The launched ranks form their groups from the injected env alone
(BASELINE.json north star: "the launched ranks form DP/TP/PP groups over
xGMI"). TP ranks are CONTIGUOUS in the global rank order, so on an
8xMI355X node a TP group maps to adjacent GPUs and its all-reduces stay on
direct xGMI links (every pair is 1 hop — 7 p2p links per GPU — but
contiguity keeps NUMA/host affinity aligned too).

    rank = dp_rank * tp_size + tp_rank
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch.distributed as dist


@dataclass
class ParallelTopology:
    world_size: int
    rank: int
    dp_size: int
    tp_size: int
    dp_rank: int
    tp_rank: int
    dp_group: Optional[object]   # ranks with the same tp_rank
    tp_group: Optional[object]   # ranks with the same dp_rank


def build_topology(tp_size: int = 1) -> ParallelTopology:
    """Split the initialized world into a DP x TP grid.

    Every rank must call this with the same tp_size (new_group is
    collective: all ranks create every subgroup, in the same order).
    """
    if not dist.is_initialized():
        return ParallelTopology(1, 0, 1, max(tp_size, 1), 0, 0, None, None)
    world = dist.get_world_size()
    rank = dist.get_rank()
    if world % tp_size != 0:
        raise ValueError(f"world size {world} not divisible by tp={tp_size}")
    dp_size = world // tp_size
    tp_rank = rank % tp_size
    dp_rank = rank // tp_size

    tp_group = None
    dp_group = None
    if tp_size > 1:
        for d in range(dp_size):
            ranks = list(range(d * tp_size, (d + 1) * tp_size))
            g = dist.new_group(ranks)
            if d == dp_rank:
                tp_group = g
    if dp_size > 1 and tp_size > 1:
        for t in range(tp_size):
            ranks = list(range(t, world, tp_size))
            g = dist.new_group(ranks)
            if t == tp_rank:
                dp_group = g
    # tp_size == 1: dp_group None means the default (whole-world) group
    return ParallelTopology(world, rank, dp_size, tp_size, dp_rank, tp_rank,
                            dp_group, tp_group)


@dataclass
class GridTopology:
    """General DP x PP x TP grid (EP reuses one of the axes — usually EP
    == DP for MoE — so three axes suffice). Rank layout, innermost last:

        rank = ((dp_rank * pp_size) + pp_rank) * tp_size + tp_rank

    tp ranks stay contiguous (adjacent GPUs over xGMI); a pp stage's
    neighbors are +/- tp_size apart (still one xGMI hop on an 8-GPU node).
    """
    world_size: int
    rank: int
    dp_size: int
    pp_size: int
    tp_size: int
    dp_rank: int
    pp_rank: int
    tp_rank: int
    dp_group: Optional[object]
    pp_group: Optional[object]
    tp_group: Optional[object]


def build_grid(tp_size: int = 1, pp_size: int = 1) -> GridTopology:
    """Split the world into dp x pp x tp; every rank must call with the
    same sizes (new_group is collective, created in a fixed global order).
    """
    if not dist.is_initialized():
        return GridTopology(1, 0, 1, max(pp_size, 1), max(tp_size, 1),
                            0, 0, 0, None, None, None)
    world = dist.get_world_size()
    rank = dist.get_rank()
    if world % (tp_size * pp_size) != 0:
        raise ValueError(
            f"world {world} not divisible by tp*pp={tp_size * pp_size}")
    dp_size = world // (tp_size * pp_size)
    tp_rank = rank % tp_size
    pp_rank = (rank // tp_size) % pp_size
    dp_rank = rank // (tp_size * pp_size)

    def _mk(axis_groups):
        mine = None
        for ranks in axis_groups:
            g = dist.new_group(ranks)
            if rank in ranks:
                mine = g
        return mine

    tp_group = pp_group = dp_group = None
    if tp_size > 1:
        tp_group = _mk([[(d * pp_size + p) * tp_size + t
                         for t in range(tp_size)]
                        for d in range(dp_size) for p in range(pp_size)])
    if pp_size > 1:
        pp_group = _mk([[(d * pp_size + p) * tp_size + t
                         for p in range(pp_size)]
                        for d in range(dp_size) for t in range(tp_size)])
    if dp_size > 1 and (tp_size > 1 or pp_size > 1):
        dp_group = _mk([[(d * pp_size + p) * tp_size + t
                         for d in range(dp_size)]
                        for p in range(pp_size) for t in range(tp_size)])
    return GridTopology(world, rank, dp_size, pp_size, tp_size,
                        dp_rank, pp_rank, tp_rank,
                        dp_group, pp_group, tp_group)
