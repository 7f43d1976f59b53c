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


class SPMoEBlock(nn.Module):
    """Sequence-parallel MoE block: the residual stream stays seq-sharded;
    attention gathers/reduce-scatters around a TP attention (like SPBlock);
    the MoE MLP consumes the SHARD directly — each rank routes its own
    S/tp tokens and the EP all-to-all moves them to expert owners, so
    sequence parallelism composes with expert parallelism with no extra
    seam (experts are NOT tensor-sharded here: TP expert shards would need
    every tp rank to see the same tokens, which SP removes by design)."""

    def __init__(self, cfg, group=None, ep_group=None):
        super().__init__()
        from ..models.moe_llama import MoEMLP
        from .tp_llama import TPAttention
        self.cfg = cfg
        self.group = group
        self.attn = TPAttention(cfg, group, sequence_parallel=True)
        ff = cfg.expert_ff or cfg.intermediate_size // 2
        self.moe = MoEMLP(cfg.hidden_size, ff, cfg.n_experts, cfg.top_k,
                          group=ep_group)
        self.input_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attn_norm_weight = nn.Parameter(
            torch.ones(cfg.hidden_size))

    def forward(self, x_s, residual_s, inv_freq):
        normed_s, residual_s = fused_rmsnorm(
            x_s, self.input_norm_weight, residual_s, self.cfg.norm_eps)
        full = _GatherSeq.apply(normed_s, self.group)
        attn_partial = self.attn(full, inv_freq)
        attn_s = _ReduceScatterSeq.apply(attn_partial, self.group)
        normed_s, residual_s = fused_rmsnorm(
            attn_s, self.post_attn_norm_weight, residual_s,
            self.cfg.norm_eps)
        return self.moe(normed_s), residual_s


class _AllReduceMean(torch.autograd.Function):
    """Differentiable group mean for scalar aux losses: forward sums over
    the group and divides by its size; backward passes the (uniform)
    upstream gradient straight through — each rank's local term got
    weight 1/n in every rank's loss and every rank backprops it."""

    @staticmethod
    def forward(ctx, x, group):
        n = _group_size(group)
        if n == 1:
            return x
        out = x.clone()
        dist.all_reduce(out, group=group)
        return out / n
    @staticmethod
    def backward(ctx, g):
        return g, None


class SPMoEModel(nn.Module):
    """MoE-Llama with sequence-parallel attention (seq shards over
    ``group``) and EP-dispatched experts (over ``ep_group``). The "SP with
    MoE stages" composition of the parallelism matrix."""

    def __init__(self, cfg, group=None, ep_group=None):
        super().__init__()
        self.cfg = cfg
        self.group = group
        self.ep_group = ep_group
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(SPMoEBlock(cfg, group, ep_group)
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
        loss = per_tok.sum() / n_valid
        if self.cfg.aux_loss_coef:
            # each rank's routers saw only its seq shard: group-mean the
            # aux term so every rank optimizes the identical total loss
            aux = sum(blk.moe.aux_loss for blk in self.blocks)
            aux = _AllReduceMean.apply(aux / len(self.blocks), self.group)
            loss = loss + self.cfg.aux_loss_coef * aux
        return loss

    @torch.no_grad()
    def shard_from_full(self, full) -> None:
        """Load from an unsharded MoELlamaModel: attention TP-shards over
        ``group``; each owned expert copies whole (ep over ``ep_group``);
        norms/embed/head/router replicate."""
        from .tp import shard_from
        g = self.group
        ep_rank = dist.get_rank(self.ep_group) if (
            dist.is_initialized() and _group_size(self.ep_group) > 1) else 0
        self.embed.weight.copy_(full.embed.weight)
        self.lm_head.weight.copy_(full.lm_head.weight)
        self.final_norm_weight.copy_(full.final_norm_weight)
        q_size = self.cfg.num_heads * self.cfg.head_dim
        kv_size = self.cfg.num_kv_heads * self.cfg.head_dim
        for blk, fblk in zip(self.blocks, full.blocks):
            blk.input_norm_weight.copy_(fblk.input_norm_weight)
            blk.post_attn_norm_weight.copy_(fblk.post_attn_norm_weight)
            qkv = fblk.attn.qkv_proj.weight
            wq, wk, wv = qkv.split([q_size, kv_size, kv_size], dim=0)
            blk.attn.q_proj.weight.copy_(shard_from(wq, 0, g))
            blk.attn.k_proj.weight.copy_(shard_from(wk, 0, g))
            blk.attn.v_proj.weight.copy_(shard_from(wv, 0, g))
            blk.attn.o_proj.weight.copy_(
                shard_from(fblk.attn.o_proj.weight, 1, g))
            blk.moe.router.weight.copy_(fblk.moe.router.weight)
            per = blk.moe.experts_per_rank
            for le, ex in enumerate(blk.moe.experts):
                src = fblk.moe.experts[ep_rank * per + le]
                ex.gate_proj.weight.copy_(src.gate_proj.weight)
                ex.up_proj.weight.copy_(src.up_proj.weight)
                ex.down_proj.weight.copy_(src.down_proj.weight)

    @torch.no_grad()
    def allreduce_sp_grads(self) -> None:
        """Seq-sharded params (norms, embed, router — their backwards saw
        only this rank's positions) sum grads over the tp group."""
        if _group_size(self.group) == 1:
            return
        params = [self.embed.weight, self.final_norm_weight]
        for b in self.blocks:
            params += [b.input_norm_weight, b.post_attn_norm_weight,
                       b.moe.router.weight]
        for p in params:
            if p.grad is not None:
                dist.all_reduce(p.grad, group=self.group)
