"""Mixture-of-Experts Llama variant: dense attention blocks with MoE MLPs
(dropless top-k routing, parallel/ep.py). With an EP group the experts
shard across ranks and tokens travel by all-to-all; with none it is a
single-process MoE. A second managed model family beyond dense Llama —
the operator/launcher treat it identically (it is just the workload)."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from ..ops import fused_cross_entropy, fused_rmsnorm, make_inv_freq
from ..parallel.ep import MoEMLP
from .config import LlamaConfig
from .llama import Attention


@dataclass
class MoELlamaConfig(LlamaConfig):
    n_experts: int = 8
    top_k: int = 2
    expert_ff: Optional[int] = None  # default: intermediate_size // 2
    # Switch-style load-balance aux loss coefficient (0 disables); the
    # per-layer aux terms are averaged and added to the LM loss
    aux_loss_coef: float = 0.01

    @property
    def n_params(self) -> int:
        """Total parameters (the dense formula's MLP term replaced by
        router + n_experts SwiGLU experts)."""
        H, V, L = self.hidden_size, self.vocab_size, self.num_layers
        kv = self.num_kv_heads * self.head_dim
        q = self.num_heads * self.head_dim
        ff = self.expert_ff or self.intermediate_size // 2
        attn = H * q + 2 * H * kv + q * H
        moe = H * self.n_experts + self.n_experts * 3 * H * ff
        per_layer = attn + moe + 2 * H
        embed = V * H * (1 if self.tie_embeddings else 2)
        return L * per_layer + embed + H

    @property
    def active_params(self) -> int:
        """Parameters touched per token (top-k experts instead of all)."""
        ff = self.expert_ff or self.intermediate_size // 2
        inactive = (self.n_experts - self.top_k) * 3 * \
            self.hidden_size * ff * self.num_layers
        return self.n_params - inactive


MOE_TINY = MoELlamaConfig(
    name="moe-tiny", vocab_size=512, hidden_size=64, intermediate_size=128,
    num_layers=2, num_heads=4, num_kv_heads=2, head_dim=16,
    rope_theta=10000.0, n_experts=4, top_k=2)


class MoEBlock(nn.Module):
    def __init__(self, cfg: MoELlamaConfig, ep_group=None, tp_group=None):
        super().__init__()
        self.cfg = cfg
        if tp_group is not None:
            from ..parallel.tp_llama import TPAttention
            self.attn = TPAttention(cfg, tp_group)
        else:
            self.attn = Attention(cfg)
        ff = cfg.expert_ff or cfg.intermediate_size // 2
        self.moe = MoEMLP(cfg.hidden_size, ff, cfg.n_experts, cfg.top_k,
                          group=ep_group, tp_group=tp_group)
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
        return self.moe(normed), residual


class MoELlamaModel(nn.Module):
    """ep_group shards experts; tp_group (EP x TP) additionally tensor-
    shards attention and each expert's matrices. Initialize a sharded
    instance from an unsharded one with shard_from_full."""

    def __init__(self, cfg: MoELlamaConfig, ep_group=None, tp_group=None):
        super().__init__()
        self.cfg = cfg
        self.ep_group = ep_group
        self.tp_group = tp_group
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(MoEBlock(cfg, ep_group, tp_group)
                                    for _ in range(cfg.num_layers))
        self.final_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
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
        logits = self.lm_head(normed)
        if targets is None:
            return logits
        T = logits.shape[0] * logits.shape[1]
        per_tok = fused_cross_entropy(
            logits.reshape(T, -1).contiguous(), targets.reshape(T))
        n_valid = (targets.reshape(T) != -100).sum().clamp(min=1)
        loss = per_tok.sum() / n_valid
        if self.cfg.aux_loss_coef:
            aux = sum(blk.moe.aux_loss for blk in self.blocks)
            loss = loss + self.cfg.aux_loss_coef * aux / len(self.blocks)
        return loss

    @torch.no_grad()
    def shard_from_full(self, full: "MoELlamaModel") -> None:
        """Load this rank's (ep, tp) shards from an unsharded MoE model:
        attention slices exactly like TPLlamaModel.shard_from_full; each
        owned expert (global id = ep_rank * experts_per_rank + local)
        takes gate/up ROW blocks and down COLUMN blocks of the full
        expert; norms/embed/head/router replicate."""
        import torch.distributed as dist
        from ..parallel.tp import _group_size, shard_from
        g = self.tp_group
        ep = self.ep_group
        ep_rank = dist.get_rank(ep) if (
            dist.is_initialized() and _group_size(ep) > 1) else 0
        self.embed.weight.copy_(full.embed.weight)
        self.lm_head.weight.copy_(full.lm_head.weight)
        self.final_norm_weight.copy_(full.final_norm_weight)
        q_size = self.cfg.num_heads * self.cfg.head_dim
        kv_size = self.cfg.num_kv_heads * self.cfg.head_dim
        for blk, fblk in zip(self.blocks, full.blocks):
            blk.input_norm_weight.copy_(fblk.input_norm_weight)
            blk.post_attn_norm_weight.copy_(fblk.post_attn_norm_weight)
            if g is not None:
                qkv = fblk.attn.qkv_proj.weight
                wq, wk, wv = qkv.split([q_size, kv_size, kv_size], dim=0)
                blk.attn.q_proj.weight.copy_(shard_from(wq, 0, g))
                blk.attn.k_proj.weight.copy_(shard_from(wk, 0, g))
                blk.attn.v_proj.weight.copy_(shard_from(wv, 0, g))
                blk.attn.o_proj.weight.copy_(
                    shard_from(fblk.attn.o_proj.weight, 1, g))
            else:
                blk.attn.qkv_proj.weight.copy_(fblk.attn.qkv_proj.weight)
                blk.attn.o_proj.weight.copy_(fblk.attn.o_proj.weight)
            blk.moe.router.weight.copy_(fblk.moe.router.weight)
            per = blk.moe.experts_per_rank
            for le, ex in enumerate(blk.moe.experts):
                src = fblk.moe.experts[ep_rank * per + le]
                if g is not None:
                    ex.gate_proj.weight.copy_(
                        shard_from(src.gate_proj.weight, 0, g))
                    ex.up_proj.weight.copy_(
                        shard_from(src.up_proj.weight, 0, g))
                    ex.down_proj.weight.copy_(
                        shard_from(src.down_proj.weight, 1, g))
                else:
                    ex.gate_proj.weight.copy_(src.gate_proj.weight)
                    ex.up_proj.weight.copy_(src.up_proj.weight)
                    ex.down_proj.weight.copy_(src.down_proj.weight)


# Mixtral-class MoE: 8 experts x top-2, ~47B total / ~13B active params.
# The EP target config: one expert shard per MI355X on an 8-GPU node
# (launcher --ep 8), or EP x TP for headroom.
# single-GPU-profileable mid size (~2.6B params: bf16 + fp32 opt state fit
# comfortably in 288 GB) — the EP/MoE kernel-profiling model
MOE_MID = MoELlamaConfig(
    name="moe-mid", vocab_size=32000, hidden_size=2048,
    intermediate_size=11264, num_layers=8, num_heads=16, num_kv_heads=4,
    head_dim=128, rope_theta=500000.0, n_experts=8, top_k=2,
    expert_ff=5632)

MOE_8X7B = MoELlamaConfig(
    name="moe-8x7b", vocab_size=32000, hidden_size=4096,
    intermediate_size=28672, num_layers=32, num_heads=32, num_kv_heads=8,
    head_dim=128, rope_theta=1_000_000.0, n_experts=8, top_k=2,
    expert_ff=14336)


# launcher/EPTrainer look models up by name like the dense families
from .config import CONFIGS  # noqa: E402

CONFIGS[MOE_TINY.name] = MOE_TINY
CONFIGS[MOE_MID.name] = MOE_MID
CONFIGS[MOE_8X7B.name] = MOE_8X7B
