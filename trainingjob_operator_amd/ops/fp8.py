"""Opt-in fp8 (OCP e4m3fn) projection GEMMs for gfx950.

CDNA4's fp8 MFMA rate is 2x bf16 (~5 PF dense); hipBLASLt exposes it via
torch._scaled_mm with per-tensor scales (probed on silicon:
scripts/fp8probe.py — the OCP e4m3fn variant works, MI300X's fnuz
correctly does not). This module quantizes activations and weights to
e4m3fn with per-tensor amax scales for the FORWARD GEMM and keeps the
backward in bf16 (dgrad/wgrad via plain matmul on the saved bf16
tensors) — the usual mixed-fp8 training recipe.

Enable with AITJ_FP8_PROJ=1 (TrainConfig.fp8_projections sets it). The
flagship bench stays bf16 (BASELINE dtype contract); fp8 numbers are
published as a separate config.
"""
from __future__ import annotations

import os

import torch

_E4M3_MAX = 448.0


def fp8_enabled() -> bool:
    return os.environ.get("AITJ_FP8_PROJ", "0") == "1" \
        and torch.cuda.is_available()


def _quantize(t: torch.Tensor):
    """Per-tensor amax scaling to e4m3fn; returns (fp8 tensor, scale)."""
    td = t.detach()    # the Function supplies its own backward; a cached
    amax = td.abs().amax().float().clamp(min=1e-12)   # wq must carry NO
    scale = amax / _E4M3_MAX                          # autograd history
    q = (td.float() / scale).clamp(-_E4M3_MAX, _E4M3_MAX) \
        .to(torch.float8_e4m3fn)
    return q, scale


# weights change once per optimizer step: cache their fp8 image keyed by
# a global version the trainer bumps after each step (per-call weight
# re-quantization measured ~5 GB of extra traffic per layer-set per step)
_version = 0


def bump_version() -> None:
    global _version
    _version += 1


def _weight_q(linear: torch.nn.Module):
    c = getattr(linear, "_fp8_cache", None)
    if c is not None and c[2] == _version:
        return c[0], c[1]
    wq, sw = _quantize(linear.weight)
    linear._fp8_cache = (wq, sw, _version)
    return wq, sw


class _Fp8Linear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor,
                wq: torch.Tensor, sw: torch.Tensor):
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        xq, sx = _quantize(x2)
        out = torch._scaled_mm(xq, wq.t(), scale_a=sx, scale_b=sw,
                               out_dtype=torch.bfloat16)
        ctx.save_for_backward(x2, weight)
        return out.reshape(*shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        x2, weight = ctx.saved_tensors
        d2 = dout.reshape(-1, dout.shape[-1])
        dx = (d2 @ weight).reshape(*dout.shape[:-1], weight.shape[1])
        dw = d2.t() @ x2
        return dx, dw, None, None


def fp8_linear(x: torch.Tensor, linear: torch.nn.Module) -> torch.Tensor:
    """y = x @ W^T with the forward GEMM in e4m3fn (per-tensor scales,
    step-cached fp8 weights), backward in bf16."""
    wq, sw = _weight_q(linear)
    return _Fp8Linear.apply(x, linear.weight, wq, sw)
