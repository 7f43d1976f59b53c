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


MOE_TINY = MoELlamaConfig(
    name="moe-tiny", vocab_size=512, hidden_size=64, intermediate_size=128,
    num_layers=2, num_heads=4, num_kv_heads=2, head_dim=16,
    rope_theta=10000.0, n_experts=4, top_k=2)


class MoEBlock(nn.Module):
    def __init__(self, cfg: MoELlamaConfig, ep_group=None):
        super().__init__()
        self.cfg = cfg
        self.attn = Attention(cfg)
        ff = cfg.expert_ff or cfg.intermediate_size // 2
        self.moe = MoEMLP(cfg.hidden_size, ff, cfg.n_experts, cfg.top_k,
                          group=ep_group)
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
    def __init__(self, cfg: MoELlamaConfig, ep_group=None):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(MoEBlock(cfg, ep_group)
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


# launcher/EPTrainer look models up by name like the dense families
from .config import CONFIGS  # noqa: E402

CONFIGS[MOE_TINY.name] = MOE_TINY
