"""Bucketed data-parallel gradient all-reduce over RCCL / xGMI.

Not a wrap of torch DDP: gradients already land in the contiguous flat
buffer (parallel/flat.py), so DP reduction is all-reduce over contiguous
slices ("buckets") of that buffer, fired as soon as every param in a bucket
has produced its grad — overlapping communication with the rest of backward.

xGMI sizing (SURVEY.md §5 'Distributed communication backend'): each MI355X
has 7 p2p links x ~153 GB/s; ring all-reduce is per-link bound, so buckets
are large (default 128 MiB) to amortize per-collective latency while still
giving the scheduler a few chances to overlap. SUM + a 1/world pre-scale in
the fused optimizer replaces ReduceOp.AVG (gloo, used by the CPU tests,
has no AVG).
"""
from __future__ import annotations

from contextlib import contextmanager
from dataclasses import dataclass, field
from typing import List, Optional

import torch.distributed as dist

from .flat import FlatParamStore


@dataclass
class _Bucket:
    index: int
    start: int
    end: int
    param_ids: set = field(default_factory=set)
    pending: set = field(default_factory=set)
    work: Optional[object] = None


class DDPEngine:
    def __init__(self, store: FlatParamStore,
                 process_group: Optional[object] = None,
                 bucket_bytes: int = 128 << 20,
                 world_size: Optional[int] = None):
        self.store = store
        self.group = process_group
        if world_size is not None:
            # explicit DP degree (a DPxTP grid passes its dp_size: with
            # dp == 1 the group is None, which would otherwise read as the
            # whole world)
            self.world_size = world_size
        else:
            self.world_size = dist.get_world_size(process_group) \
                if dist.is_initialized() else 1
        self.require_sync = True
        bucket_elems = max(1, bucket_bytes // 2)  # bf16

        self.buckets: List[_Bucket] = []
        cur: Optional[_Bucket] = None
        for name, p in zip(store.names, store.params):
            off, numel = store.offsets[name]
            if cur is None:
                cur = _Bucket(len(self.buckets), off, off + numel)
            cur.param_ids.add(id(p))
            cur.end = off + numel
            if cur.end - cur.start >= bucket_elems:
                self.buckets.append(cur)
                cur = None
        if cur is not None:
            self.buckets.append(cur)
        self._bucket_of = {}
        for b in self.buckets:
            for pid in b.param_ids:
                self._bucket_of[pid] = b
        self._next_launch = 0
        self._reset_pending()
        store.on_param_grad_ready(self._param_ready)

    def _reset_pending(self):
        for b in self.buckets:
            b.pending = set(b.param_ids)
            b.work = None
        self._next_launch = 0

    # -- hooks ------------------------------------------------------------
    def _param_ready(self, p) -> None:
        if self.world_size <= 1 or not self.require_sync:
            return
        b = self._bucket_of[id(p)]
        b.pending.discard(id(p))
        self._launch_ready_in_order()

    def _launch_ready_in_order(self) -> None:
        """Launch ready buckets strictly in index order so every rank
        enqueues RCCL collectives in the same sequence (a readiness-order
        launch can deadlock NCCL/RCCL if autograd hook timing differs
        across ranks — same discipline as torch DDP)."""
        while (self._next_launch < len(self.buckets)
               and not self.buckets[self._next_launch].pending):
            self._launch(self.buckets[self._next_launch])
            self._next_launch += 1

    def _launch(self, b: _Bucket) -> None:
        flat = self.store.flat_grad[b.start:b.end]
        b.work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.group,
                                 async_op=True)

    # -- step boundary ----------------------------------------------------
    def finish_backward(self) -> None:
        """Flush un-fired buckets (params without grads) and wait for all
        in-flight reductions. Call after loss.backward() on the sync step."""
        if self.world_size > 1 and self.require_sync:
            # flush: any bucket whose params never produced grads counts as
            # ready (its flat region holds zeros), keeping launch order
            for b in self.buckets[self._next_launch:]:
                b.pending.clear()
            self._launch_ready_in_order()
            for b in self.buckets:
                if b.work is not None:
                    b.work.wait()
        self._reset_pending()

    @contextmanager
    def no_sync(self):
        """Gradient-accumulation context: skip reduction this backward."""
        prev = self.require_sync
        self.require_sync = False
        try:
            yield
        finally:
            self.require_sync = prev

    @property
    def grad_pre_scale(self) -> float:
        return 1.0 / self.world_size
