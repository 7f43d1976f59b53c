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


@dataclass
class MoEGridTopology:
    """4-axis MoE pipeline grid: world = edp x plane x pp x tp,
    rank = (((edp_r*plane + pl_r)*pp + p)*tp + t).

    The dp_* fields name the EXPERT PLANE (token-dispatch + dense-grad
    axis inside one data replica group) so PPTrainer's MoE path consumes
    this like a GridTopology; edp_* adds true data parallelism ON TOP:
    expert shards replicate across edp and their grads all-reduce there.
    """
    world_size: int
    rank: int
    edp_size: int
    edp_rank: int
    edp_group: Optional[object]       # same (pl, p, t), varying edp
    dp_size: int                      # == plane size
    dp_rank: int
    dp_group: Optional[object]        # dispatch plane: same (edp, p, t)
    pp_size: int
    pp_rank: int
    pp_group: Optional[object]        # same (edp, pl, t)
    tp_size: int
    tp_rank: int
    tp_group: Optional[object]        # same (edp, pl, p)
    dense_dp_group: Optional[object]  # same (p, t): all edp x plane
    pp_global_ranks: Optional[list] = None
    data_replicas: int = 1
    data_rank: int = 0


def build_moe_grid(plane_size: int = 1, pp_size: int = 1,
                   tp_size: int = 1) -> MoEGridTopology:
    """Split the world into edp x plane x pp x tp (edp = what's left).
    Every rank must call with the same sizes (new_group is collective)."""
    if not dist.is_initialized():
        return MoEGridTopology(1, 0, 1, 0, None, max(plane_size, 1), 0,
                               None, max(pp_size, 1), 0, None,
                               max(tp_size, 1), 0, None, None)
    world = dist.get_world_size()
    rank = dist.get_rank()
    pl, pp, tp = max(plane_size, 1), max(pp_size, 1), max(tp_size, 1)
    if world % (pl * pp * tp) != 0:
        raise ValueError(
            f"world {world} not divisible by plane*pp*tp={pl * pp * tp}")
    edp = world // (pl * pp * tp)
    t = rank % tp
    p = (rank // tp) % pp
    pl_r = (rank // (tp * pp)) % pl
    e = rank // (tp * pp * pl)

    def at(e_, pl_, p_, t_):
        return ((e_ * pl + pl_) * pp + p_) * tp + t_

    def mk(all_ranks_fn, axis_lens, my_key):
        """Create one group per key (all ranks, fixed order)."""
        import itertools
        mine = None
        for key in itertools.product(*(range(n) for n in axis_lens)):
            ranks = all_ranks_fn(key)
            if len(ranks) == world:
                return None if key == my_key else mine
            g = dist.new_group(ranks)
            if key == my_key:
                mine = g
        return mine

    edp_group = dp_group = pp_group = tp_group = dense_dp_group = None
    if edp > 1:
        edp_group = mk(lambda k: [at(x, *k) for x in range(edp)],
                       (pl, pp, tp), (pl_r, p, t))
    if pl > 1:
        dp_group = mk(lambda k: [at(k[0], x, k[1], k[2])
                                 for x in range(pl)],
                      (edp, pp, tp), (e, p, t))
    if pp > 1:
        pp_group = mk(lambda k: [at(k[0], k[1], x, k[2])
                                 for x in range(pp)],
                      (edp, pl, tp), (e, pl_r, t))
    if tp > 1:
        tp_group = mk(lambda k: [at(k[0], k[1], k[2], x)
                                 for x in range(tp)],
                      (edp, pl, pp), (e, pl_r, p))
    if edp * pl > 1:
        dense_dp_group = mk(
            lambda k: [at(x, y, k[0], k[1]) for x in range(edp)
                       for y in range(pl)], (pp, tp), (p, t))
    return MoEGridTopology(
        world, rank, edp, e, edp_group, pl, pl_r, dp_group,
        pp, p, pp_group, tp, t, tp_group, dense_dp_group,
        pp_global_ranks=[at(e, pl_r, x, t) for x in range(pp)],
        data_replicas=edp * pl, data_rank=e * pl + pl_r)
