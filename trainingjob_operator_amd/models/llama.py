"""Llama-3 family, MI355X-first.

Design (per /opt/skills/guides/MI355X_MICROARCH.md):
  * bf16 weights/activations; plain projections go through hipBLASLt via
    nn.functional.linear (library GEMMs belong to the library — the fused
    non-GEMM hot ops are hand-written CDNA4 kernels in ops/).
  * residual-add + RMSNorm fused to one HBM round trip (ops.fused_rmsnorm);
    SwiGLU and RoPE fused likewise; loss is a fused streaming CE over the
    [T, 128256] logits.
  * attention = torch SDPA (flash path on ROCm) with grouped-query KV.
  * activations optionally checkpointed per block; with 288 GB HBM3E per
    GPU the default keeps activations resident (checkpointing off) — flip
    it on only for long-sequence configs.

This is worker-side compute the reference operator never had (it launches
opaque containers — SURVEY.md §2.3); it is the workload the operator's
benchmark jobs run and what bench.py measures.
"""
from __future__ import annotations

import math
import os
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


def _sdpa(q, k, v, enable_gqa=False, is_causal=True):
    """SDPA with a backend override knob (AITJ_SDPA_BACKEND =
    flash|efficient|math|native|library|auto). Default (auto): the
    hand-written CDNA4 kernels (ops/attention.py, fwd AND bwd) whenever the
    shape supports them — they measure 1.38x the aotriton fwd+bwd at the
    flagship shape — with the library SDPA for everything else (decode
    shapes, small head_dim)."""
    backend = os.environ.get("AITJ_SDPA_BACKEND", "auto")
    if q.is_cuda and backend in ("auto", "native") and is_causal:
        from ..ops.attention import _supported, flash_attention
        qc = q.contiguous() if q.stride(-1) != 1 else q
        if _supported(qc, k):
            return flash_attention(qc, k, v)
        if backend == "native":
            # explicit native on a GQA v5-only shape: expand heads
            groups = q.shape[1] // k.shape[1]
            if groups > 1:
                k = k.repeat_interleave(groups, dim=1)
                v = v.repeat_interleave(groups, dim=1)
            return flash_attention(qc, k, v)
    if backend in ("auto", "native") or not q.is_cuda:
        return F.scaled_dot_product_attention(q, k, v, is_causal=is_causal,
                                              enable_gqa=enable_gqa)
    from torch.nn.attention import SDPBackend, sdpa_kernel
    mapping = {"flash": SDPBackend.FLASH_ATTENTION,
               "efficient": SDPBackend.EFFICIENT_ATTENTION,
               "math": SDPBackend.MATH}
    with sdpa_kernel([mapping[backend]]):
        return F.scaled_dot_product_attention(q, k, v, is_causal=is_causal,
                                              enable_gqa=enable_gqa)

from ..ops import (apply_rope, fused_cross_entropy, fused_rmsnorm,
                   make_inv_freq, swiglu_packed)
from ..ops.fp8 import fp8_enabled, fp8_linear


def _proj(x, linear):
    """Projection GEMM: fp8 forward (AITJ_FP8_PROJ=1, gfx950 e4m3fn via
    hipBLASLt) or the regular bf16 path."""
    if fp8_enabled() and x.is_cuda:
        return fp8_linear(x, linear)
    return linear(x)
from .config import LlamaConfig


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        H = cfg.hidden_size
        self.q_size = cfg.num_heads * cfg.head_dim
        self.kv_size = cfg.num_kv_heads * cfg.head_dim
        # fused QKV: one hipBLASLt GEMM instead of three (the k/v GEMMs are
        # skinny at GQA ratios and underutilize the chip on their own)
        self.qkv_proj = nn.Linear(H, self.q_size + 2 * self.kv_size,
                                  bias=False)
        self.o_proj = nn.Linear(self.q_size, H, bias=False)

    def forward(self, x: torch.Tensor, inv_freq: torch.Tensor) -> torch.Tensor:
        B, S, H = x.shape
        cfg = self.cfg
        qkv = _proj(x, self.qkv_proj)
        q, k, v = qkv.split([self.q_size, self.kv_size, self.kv_size],
                            dim=-1)
        # q/k go through the RoPE kernel, which wants dense rows
        q = q.reshape(B, S, cfg.num_heads, cfg.head_dim).contiguous()
        k = k.reshape(B, S, cfg.num_kv_heads, cfg.head_dim).contiguous()
        v = v.reshape(B, S, cfg.num_kv_heads, cfg.head_dim)
        q = apply_rope(q, inv_freq, S)
        k = apply_rope(k, inv_freq, S)
        q = q.transpose(1, 2)  # [B, nh, S, D]
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        # grouped-query KV handled inside SDPA (enable_gqa) — avoids
        # materializing the repeated K/V (~100 MB/layer/direction at 8B).
        # AITJ_DISABLE_GQA=1 falls back to explicit repeat (A/B knob).
        if cfg.num_heads != cfg.num_kv_heads and \
                os.environ.get("AITJ_DISABLE_GQA"):
            groups = cfg.num_heads // cfg.num_kv_heads
            k = k.repeat_interleave(groups, dim=1)
            v = v.repeat_interleave(groups, dim=1)
            o = _sdpa(q, k, v)
        else:
            o = _sdpa(q, k, v,
                      enable_gqa=cfg.num_heads != cfg.num_kv_heads)
        o = o.transpose(1, 2).reshape(B, S, cfg.num_heads * cfg.head_dim)
        return _proj(o, self.o_proj)


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        H, FF = cfg.hidden_size, cfg.intermediate_size
        self.ff = FF
        # fused gate+up: one GEMM feeding the fused SwiGLU kernel
        self.gate_up_proj = nn.Linear(H, 2 * FF, bias=False)
        self.down_proj = nn.Linear(FF, H, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return _proj(swiglu_packed(_proj(x, self.gate_up_proj)),
                     self.down_proj)


class Block(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.attn = Attention(cfg)
        self.mlp = MLP(cfg)
        self.input_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attn_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor],
                inv_freq: torch.Tensor):
        """Pre-norm block over a carried residual stream:
        (x = branch output, residual = running stream)."""
        normed, residual = fused_rmsnorm(x, self.input_norm_weight, residual,
                                         self.cfg.norm_eps)
        attn_out = self.attn(normed, inv_freq)
        normed, residual = fused_rmsnorm(attn_out, self.post_attn_norm_weight,
                                         residual, self.cfg.norm_eps)
        mlp_out = self.mlp(normed)
        return mlp_out, residual


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig, checkpoint_activations: bool = False):
        super().__init__()
        self.cfg = cfg
        self.checkpoint_activations = checkpoint_activations
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(Block(cfg) for _ in range(cfg.num_layers))
        self.final_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.embed.weight
        self.register_buffer(
            "inv_freq", make_inv_freq(cfg.head_dim, cfg.rope_theta),
            persistent=False)
        self.reset_parameters()

    def reset_parameters(self):
        std = self.cfg.init_std
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, mean=0.0, std=std)
        for name, p in self.named_parameters():
            if name.endswith("norm_weight"):
                nn.init.ones_(p)
        # scaled init for output projections (depth-aware, GPT-2 style)
        scale = 1.0 / math.sqrt(2 * self.cfg.num_layers)
        for blk in self.blocks:
            nn.init.normal_(blk.attn.o_proj.weight, std=std * scale)
            nn.init.normal_(blk.mlp.down_proj.weight, std=std * scale)

    def forward_hidden(self, tokens: torch.Tensor) -> torch.Tensor:
        """tokens [B, S] -> final normed hidden [B, S, H]."""
        x = self.embed(tokens)
        residual = None
        for blk in self.blocks:
            if self.checkpoint_activations and self.training:
                if residual is None:
                    residual = torch.zeros_like(x)
                x, residual = torch.utils.checkpoint.checkpoint(
                    blk, x, residual, self.inv_freq, use_reentrant=False)
            else:
                x, residual = blk(x, residual, self.inv_freq)
        normed, _ = fused_rmsnorm(x, self.final_norm_weight, residual,
                                  self.cfg.norm_eps)
        return normed

    def forward(self, tokens: torch.Tensor,
                targets: Optional[torch.Tensor] = None):
        """Returns loss (scalar, fp32) when targets given, else logits."""
        hidden = self.forward_hidden(tokens)
        logits = self.lm_head(hidden)
        if targets is None:
            return logits
        T = logits.shape[0] * logits.shape[1]
        per_tok = fused_cross_entropy(
            logits.reshape(T, -1).contiguous(), targets.reshape(T))
        n_valid = (targets.reshape(T) != -100).sum().clamp(min=1)
        return per_tok.sum() / n_valid
