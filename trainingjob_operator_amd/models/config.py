"""Model configurations for the managed benchmark workloads.

The flagship is Llama-3-8B (BASELINE.json config 2: "PyTorch-ROCm DDP
Llama-3-8B, 8 replicas x 1 amd.com/gpu on one MI355X node").
"""
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class LlamaConfig:
    name: str = "llama3-8b"
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    max_seq_len: int = 8192
    tie_embeddings: bool = False
    init_std: float = 0.02

    @property
    def n_params(self) -> int:
        H, F, V, L = (self.hidden_size, self.intermediate_size,
                      self.vocab_size, self.num_layers)
        kv = self.num_kv_heads * self.head_dim
        q = self.num_heads * self.head_dim
        per_layer = H * q + 2 * H * kv + q * H + 3 * H * F + 2 * H
        embed = V * H * (1 if self.tie_embeddings else 2)
        return L * per_layer + embed + H


LLAMA3_8B = LlamaConfig()

# Tiny config for CPU tests and smoke runs.
LLAMA_TINY = LlamaConfig(
    name="llama-tiny", vocab_size=512, hidden_size=64, intermediate_size=128,
    num_layers=2, num_heads=4, num_kv_heads=2, head_dim=16,
    rope_theta=10000.0, max_seq_len=128)

# ~1B config for single-GPU iteration without the full 8B footprint.
LLAMA_1B = LlamaConfig(
    name="llama-1b", vocab_size=128256, hidden_size=2048,
    intermediate_size=8192, num_layers=16, num_heads=32, num_kv_heads=8,
    head_dim=64, rope_theta=500000.0)

# GPU smoke config: exercises every HIP kernel (H % 2048 == 0) in seconds.
LLAMA_SMOKE = LlamaConfig(
    name="llama-smoke", vocab_size=32000, hidden_size=2048,
    intermediate_size=8192, num_layers=2, num_heads=16, num_kv_heads=8,
    head_dim=128, rope_theta=500000.0)

# Llama-3-70B: the canonical multi-GPU target — its ~1.12 TB of training
# state (16 B/param) outgrows one 288 GB MI355X, so it runs PP (and/or TP)
# across the node; sizing.py admission computes the same bound.
LLAMA3_70B = LlamaConfig(
    name="llama3-70b", vocab_size=128256, hidden_size=8192,
    intermediate_size=28672, num_layers=80, num_heads=64, num_kv_heads=8,
    head_dim=128, rope_theta=500000.0)

CONFIGS = {c.name: c for c in (LLAMA3_8B, LLAMA_TINY, LLAMA_1B,
                               LLAMA_SMOKE, LLAMA3_70B)}
