"""Tensor-parallel Llama: Megatron-style sharding of the transformer block
over a TP group (parallel/tp.py primitives).

Per block: ONE all-reduce after attention (RowParallel o_proj) and ONE
after the MLP (RowParallel down_proj) per direction — the minimal
collective count for xGMI. Heads are sharded across TP ranks (num_heads
and num_kv_heads must divide by tp), so RoPE/SDPA run on local heads with
no communication. Embedding and lm_head stay replicated (at 8B the head is
1 GB bf16 — replication is cheap against 288 GB HBM; a vocab-parallel
LM head + sharded CE exists separately in parallel/vocab_parallel.py for
memory-bound configs).

`shard_from_full` loads a rank's shards from an unsharded LlamaModel, which
is also how the gloo tests prove exact fwd/bwd parity
(tests/test_tp_gloo.py::test_tp_llama_matches_unsharded).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..models.config import LlamaConfig
from ..models.llama import LlamaModel, _sdpa
from ..ops import apply_rope, fused_cross_entropy, fused_rmsnorm, make_inv_freq, swiglu
from .tp import ColumnParallelLinear, RowParallelLinear, _group_size, shard_from


class TPAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig, group=None,
                 sequence_parallel: bool = False):
        super().__init__()
        self.cfg = cfg
        n = _group_size(group)
        assert cfg.num_heads % n == 0 and cfg.num_kv_heads % n == 0, \
            (cfg.num_heads, cfg.num_kv_heads, n)
        self.n_local_heads = cfg.num_heads // n
        self.n_local_kv = cfg.num_kv_heads // n
        H = cfg.hidden_size
        D = cfg.head_dim
        ci = not sequence_parallel
        self.q_proj = ColumnParallelLinear(H, cfg.num_heads * D, group,
                                           copy_input=ci)
        self.k_proj = ColumnParallelLinear(H, cfg.num_kv_heads * D, group,
                                           copy_input=ci)
        self.v_proj = ColumnParallelLinear(H, cfg.num_kv_heads * D, group,
                                           copy_input=ci)
        self.o_proj = RowParallelLinear(cfg.num_heads * D, H, group,
                                        reduce_output=ci)

    def forward(self, x, inv_freq):
        B, S, _ = x.shape
        D = self.cfg.head_dim
        q = self.q_proj(x).reshape(B, S, self.n_local_heads, D).contiguous()
        k = self.k_proj(x).reshape(B, S, self.n_local_kv, D).contiguous()
        v = self.v_proj(x).reshape(B, S, self.n_local_kv, D)
        q = apply_rope(q, inv_freq, S).transpose(1, 2)
        k = apply_rope(k, inv_freq, S).transpose(1, 2)
        v = v.transpose(1, 2)
        o = _sdpa(q, k, v, enable_gqa=self.n_local_heads != self.n_local_kv)
        o = o.transpose(1, 2).reshape(B, S, self.n_local_heads * D)
        return self.o_proj(o)


class TPMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, group=None,
                 sequence_parallel: bool = False):
        super().__init__()
        ci = not sequence_parallel
        self.gate_proj = ColumnParallelLinear(cfg.hidden_size,
                                              cfg.intermediate_size, group,
                                              copy_input=ci)
        self.up_proj = ColumnParallelLinear(cfg.hidden_size,
                                            cfg.intermediate_size, group,
                                            copy_input=ci)
        self.down_proj = RowParallelLinear(cfg.intermediate_size,
                                           cfg.hidden_size, group,
                                           reduce_output=ci)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x).contiguous(),
                                     self.up_proj(x).contiguous()))


class TPBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig, group=None):
        super().__init__()
        self.cfg = cfg
        self.attn = TPAttention(cfg, group)
        self.mlp = TPMLP(cfg, group)
        self.input_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attn_norm_weight = nn.Parameter(
            torch.ones(cfg.hidden_size))

    def forward(self, x, residual, inv_freq):
        normed, residual = fused_rmsnorm(x, self.input_norm_weight,
                                         residual, self.cfg.norm_eps)
        attn_out = self.attn(normed, inv_freq)
        normed, residual = fused_rmsnorm(attn_out,
                                         self.post_attn_norm_weight,
                                         residual, self.cfg.norm_eps)
        return self.mlp(normed), residual


class TPLlamaModel(nn.Module):
    """vocab_parallel_head=True swaps the replicated lm_head (1 GB bf16 at
    Llama-3's 128k vocab, plus full-vocab logits) for a row-sharded head
    with the fused sharded CE (parallel/vocab_parallel.py): weight AND
    logits shrink to 1/tp per rank."""

    def __init__(self, cfg: LlamaConfig, group=None,
                 vocab_parallel_head: bool = False):
        super().__init__()
        self.cfg = cfg
        self.group = group
        self.vocab_parallel_head = vocab_parallel_head
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(TPBlock(cfg, group)
                                    for _ in range(cfg.num_layers))
        self.final_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        if vocab_parallel_head:
            from .vocab_parallel import VocabParallelLMHead
            self.lm_head = VocabParallelLMHead(cfg.hidden_size,
                                               cfg.vocab_size, group)
        else:
            self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size,
                                     bias=False)
        self.register_buffer("inv_freq",
                             make_inv_freq(cfg.head_dim, cfg.rope_theta),
                             persistent=False)

    def forward(self, tokens, targets=None):
        x = self.embed(tokens)
        residual = None
        for blk in self.blocks:
            x, residual = blk(x, residual, self.inv_freq)
        normed, _ = fused_rmsnorm(x, self.final_norm_weight, residual,
                                  self.cfg.norm_eps)
        if self.vocab_parallel_head:
            B, S, H = normed.shape
            if targets is None:
                from .tp import _GatherFromTP
                return _GatherFromTP.apply(
                    self.lm_head.proj(normed), self.group)
            per_tok = self.lm_head.loss(
                normed.reshape(B * S, H), targets.reshape(B * S))
            n_valid = (targets.reshape(-1) != -100).sum().clamp(min=1)
            return per_tok.sum() / n_valid
        logits = self.lm_head(normed)
        if targets is None:
            return logits
        T = logits.shape[0] * logits.shape[1]
        per_tok = fused_cross_entropy(
            logits.reshape(T, -1).contiguous(), targets.reshape(T))
        n_valid = (targets.reshape(T) != -100).sum().clamp(min=1)
        return per_tok.sum() / n_valid

    @torch.no_grad()
    def shard_from_full(self, full: LlamaModel) -> None:
        """Load this rank's shards from an unsharded model: heads are
        contiguous row blocks of q/k/v, gate/up shard rows, o/down shard
        columns; norms/embed/head replicate."""
        g = self.group
        self.embed.weight.copy_(full.embed.weight)
        if getattr(self, "vocab_parallel_head", False):  # SPLlama: absent
            self.lm_head.proj.weight.copy_(
                shard_from(full.lm_head.weight, 0, g))
        else:
            self.lm_head.weight.copy_(full.lm_head.weight)
        self.final_norm_weight.copy_(full.final_norm_weight)
        q_size = self.cfg.num_heads * self.cfg.head_dim
        kv_size = self.cfg.num_kv_heads * self.cfg.head_dim
        for blk, fblk in zip(self.blocks, full.blocks):
            blk.input_norm_weight.copy_(fblk.input_norm_weight)
            blk.post_attn_norm_weight.copy_(fblk.post_attn_norm_weight)
            qkv = fblk.attn.qkv_proj.weight  # [q+2kv, H] fused in the base
            wq, wk, wv = qkv.split([q_size, kv_size, kv_size], dim=0)
            blk.attn.q_proj.weight.copy_(shard_from(wq, 0, g))
            blk.attn.k_proj.weight.copy_(shard_from(wk, 0, g))
            blk.attn.v_proj.weight.copy_(shard_from(wv, 0, g))
            blk.attn.o_proj.weight.copy_(
                shard_from(fblk.attn.o_proj.weight, 1, g))
            gu = fblk.mlp.gate_up_proj.weight  # [2F, H]
            wg, wu = gu.chunk(2, dim=0)
            blk.mlp.gate_proj.weight.copy_(shard_from(wg, 0, g))
            blk.mlp.up_proj.weight.copy_(shard_from(wu, 0, g))
            blk.mlp.down_proj.weight.copy_(
                shard_from(fblk.mlp.down_proj.weight, 1, g))
