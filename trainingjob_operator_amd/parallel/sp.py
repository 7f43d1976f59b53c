"""Sequence parallelism (Megatron-SP) on top of TP.

Between the TP regions, activations and the residual stream live
SEQUENCE-SHARDED across the TP group: the RowParallel output reduce-
scatters over the sequence dim instead of all-reducing, and the next
ColumnParallel input all-gathers it back. RS + AG move the same bytes as
the all-reduce they replace — xGMI cost unchanged — while norms, residual
adds and their activations shrink by 1/tp per rank.

Gloo-verified against the unsharded model in tests/test_tp_gloo.py.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from ..models.config import LlamaConfig
from ..ops import fused_cross_entropy, fused_rmsnorm, make_inv_freq
from .tp import _group_size
from .tp_llama import TPAttention, TPMLP


class _GatherSeq(torch.autograd.Function):
    """fwd: all-gather the sequence shards (dim 1); bwd: reduce-scatter the
    gradient back to this rank's shard (sum — the full tensor is consumed
    by every rank's TP shard computation)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        n = _group_size(group)
        ctx.n = n
        if n == 1:
            return x
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(n)]
        dist.all_gather(parts, x, group=group)
        return torch.cat(parts, dim=1)

    @staticmethod
    def backward(ctx, g):
        if ctx.n == 1:
            return g, None
        shards = list(g.contiguous().chunk(ctx.n, dim=1))
        out = torch.empty_like(shards[0])
        dist.reduce_scatter(out, shards, group=ctx.group)
        return out, None


class _GatherSeqReplicated(torch.autograd.Function):
    """Gather for a seam whose DOWNSTREAM is replicated on every rank (the
    final norm -> head -> loss): each rank owns one full copy of the
    downstream function, so the correct gradient of my shard is MY copy's
    slice alone — a reduce-scatter here would double-count (every rank
    computed the whole loss)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        n = _group_size(group)
        ctx.n = n
        if n == 1:
            return x
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(n)]
        dist.all_gather(parts, x, group=group)
        return torch.cat(parts, dim=1)

    @staticmethod
    def backward(ctx, g):
        if ctx.n == 1:
            return g, None
        r = dist.get_rank(ctx.group)
        return g.chunk(ctx.n, dim=1)[r].contiguous(), None


class _ReduceScatterSeq(torch.autograd.Function):
    """fwd: reduce-scatter partial sums over the sequence dim (replaces the
    RowParallel all-reduce); bwd: all-gather the incoming shard grads."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        n = _group_size(group)
        ctx.n = n
        if n == 1:
            return x
        shards = list(x.contiguous().chunk(n, dim=1))
        out = torch.empty_like(shards[0])
        dist.reduce_scatter(out, shards, group=group)
        return out

    @staticmethod
    def backward(ctx, g):
        if ctx.n == 1:
            return g, None
        g = g.contiguous()
        parts = [torch.empty_like(g) for _ in range(ctx.n)]
        dist.all_gather(parts, g, group=ctx.group)
        return torch.cat(parts, dim=1), None


class SPBlock(nn.Module):
    """TP block operating on a sequence-sharded residual stream. The inner
    attention/MLP run in sequence-parallel mode: ColumnParallel skips the
    f-op (the seq all-gather seam already sums input grads) and RowParallel
    emits partial sums for the reduce-scatter."""

    def __init__(self, cfg: LlamaConfig, group=None):
        super().__init__()
        self.cfg = cfg
        self.group = group
        self.attn = TPAttention(cfg, group, sequence_parallel=True)
        self.mlp = TPMLP(cfg, group, sequence_parallel=True)
        self.input_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attn_norm_weight = nn.Parameter(
            torch.ones(cfg.hidden_size))

    def forward(self, x_s, residual_s, inv_freq):
        """x_s/residual_s are sequence shards [B, S/tp, H]."""
        normed_s, residual_s = fused_rmsnorm(
            x_s, self.input_norm_weight, residual_s, self.cfg.norm_eps)
        full = _GatherSeq.apply(normed_s, self.group)
        attn_partial = self.attn(full, inv_freq)
        attn_s = _ReduceScatterSeq.apply(attn_partial, self.group)
        normed_s, residual_s = fused_rmsnorm(
            attn_s, self.post_attn_norm_weight, residual_s,
            self.cfg.norm_eps)
        full = _GatherSeq.apply(normed_s, self.group)
        mlp_partial = self.mlp(full)
        mlp_s = _ReduceScatterSeq.apply(mlp_partial, self.group)
        return mlp_s, residual_s


class SPLlamaModel(nn.Module):
    """Llama with TP + sequence parallelism. The embedding output is
    scattered to this rank's sequence shard; the final norm runs sharded
    and the (replicated) head consumes the gathered hidden."""

    def __init__(self, cfg: LlamaConfig, group=None):
        super().__init__()
        self.cfg = cfg
        self.group = group
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(SPBlock(cfg, group)
                                    for _ in range(cfg.num_layers))
        self.final_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.register_buffer("inv_freq",
                             make_inv_freq(cfg.head_dim, cfg.rope_theta),
                             persistent=False)

    def _my_seq_shard(self, x):
        n = _group_size(self.group)
        if n == 1:
            return x
        r = dist.get_rank(self.group)
        return x.chunk(n, dim=1)[r].contiguous()

    def forward(self, tokens, targets=None):
        S = tokens.shape[1]
        n = _group_size(self.group)
        assert S % max(n, 1) == 0
        x_s = self._my_seq_shard(self.embed(tokens))
        residual_s = None
        for blk in self.blocks:
            x_s, residual_s = blk(x_s, residual_s, self.inv_freq)
        normed_s, _ = fused_rmsnorm(x_s, self.final_norm_weight, residual_s,
                                    self.cfg.norm_eps)
        normed = _GatherSeqReplicated.apply(normed_s, self.group)
        logits = self.lm_head(normed)
        if targets is None:
            return logits
        T = logits.shape[0] * logits.shape[1]
        per_tok = fused_cross_entropy(
            logits.reshape(T, -1).contiguous(), targets.reshape(T))
        n_valid = (targets.reshape(T) != -100).sum().clamp(min=1)
        return per_tok.sum() / n_valid

    @torch.no_grad()
    def shard_from_full(self, full) -> None:
        from .tp_llama import TPLlamaModel
        TPLlamaModel.shard_from_full(self, full)  # same weight layout

    @torch.no_grad()
    def allreduce_sp_grads(self) -> None:
        """Sum the gradients of parameters that live in the sequence-
        sharded region (norm weights, embedding) over the TP group — each
        rank's backward only saw its own sequence positions (Megatron's
        allreduce_sequence_parallel_gradients). Call after backward, before
        the optimizer."""
        if _group_size(self.group) == 1:
            return
        for p in ([self.embed.weight, self.final_norm_weight]
                  + [b.input_norm_weight for b in self.blocks]
                  + [b.post_attn_norm_weight for b in self.blocks]):
            if p.grad is not None:
                dist.all_reduce(p.grad, group=self.group)
