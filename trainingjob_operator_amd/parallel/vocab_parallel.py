"""Vocab-parallel LM head + cross-entropy for TP.

At Llama-3's 128,256-token vocabulary the lm_head is the largest single
matrix (1 GB bf16) and its logits the largest activation (mb x seq x 128k).
Sharding the head over the TP group keeps both at 1/tp per rank; the loss
is computed WITHOUT ever materializing the full logits row:

  lse   = log sum exp over all shards   (one MAX + one SUM all-reduce)
  picked = the target logit, owned by exactly one rank (one SUM all-reduce)
  loss  = lse - picked

Backward is local: d(local_logits) = (softmax_local - onehot_local) * g,
with softmax_local = exp(local - lse) — no further communication.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from .tp import ColumnParallelLinear, _group_size


class _VocabParallelCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, local_logits, targets, vocab_start, vocab_end, group,
                ignore_index):
        lf = local_logits.float()
        local_max = lf.max(dim=-1).values
        if _group_size(group) > 1:
            gmax = local_max.clone()
            dist.all_reduce(gmax, op=dist.ReduceOp.MAX, group=group)
        else:
            gmax = local_max
        sumexp = torch.exp(lf - gmax.unsqueeze(-1)).sum(dim=-1)
        if _group_size(group) > 1:
            dist.all_reduce(sumexp, group=group)
        lse = gmax + torch.log(sumexp)

        in_shard = (targets >= vocab_start) & (targets < vocab_end)
        local_t = torch.where(in_shard, targets - vocab_start,
                              torch.zeros_like(targets))
        picked = lf.gather(-1, local_t.unsqueeze(-1).long()).squeeze(-1)
        picked = torch.where(in_shard, picked, torch.zeros_like(picked))
        if _group_size(group) > 1:
            dist.all_reduce(picked, group=group)

        valid = targets != ignore_index
        loss = torch.where(valid, lse - picked, torch.zeros_like(lse))
        ctx.save_for_backward(local_logits, targets, lse)
        ctx.vocab_start = vocab_start
        ctx.vocab_end = vocab_end
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, gout):
        local_logits, targets, lse = ctx.saved_tensors
        lf = local_logits.float()
        p = torch.exp(lf - lse.unsqueeze(-1))
        in_shard = (targets >= ctx.vocab_start) & (targets < ctx.vocab_end)
        local_t = torch.where(in_shard, targets - ctx.vocab_start,
                              torch.zeros_like(targets)).long()
        onehot_scale = in_shard.float()
        p.scatter_add_(-1, local_t.unsqueeze(-1),
                       -onehot_scale.unsqueeze(-1))
        valid = (targets != ctx.ignore_index).float()
        p = p * (gout.float() * valid).unsqueeze(-1)
        return p.to(local_logits.dtype), None, None, None, None, None


def vocab_parallel_cross_entropy(local_logits, targets, vocab_start,
                                 vocab_end, group=None,
                                 ignore_index: int = -100):
    """Per-token loss [T] over vocab-sharded logits [T, V/tp]."""
    return _VocabParallelCE.apply(local_logits, targets, vocab_start,
                                  vocab_end, group, ignore_index)


class VocabParallelLMHead(nn.Module):
    """Column-parallel head (vocab rows sharded) + fused sharded CE."""

    def __init__(self, hidden_size: int, vocab_size: int, group=None,
                 dtype=None):
        super().__init__()
        self.group = group
        n = _group_size(group)
        assert vocab_size % n == 0
        self.vocab_per_rank = vocab_size // n
        self.proj = ColumnParallelLinear(hidden_size, vocab_size, group,
                                         gather_output=False, dtype=dtype)

    def _bounds(self):
        r = dist.get_rank(self.group) if (
            dist.is_initialized() and _group_size(self.group) > 1) else 0
        start = r * self.vocab_per_rank
        return start, start + self.vocab_per_rank

    def loss(self, hidden, targets, ignore_index: int = -100):
        """hidden [T, H], targets [T] -> per-token loss [T] (fp32)."""
        local_logits = self.proj(hidden)
        start, end = self._bounds()
        return vocab_parallel_cross_entropy(local_logits, targets, start,
                                            end, self.group, ignore_index)
