"""Native MFMA flash attention (hip/attention.hip): hand-written CDNA4
forward AND backward, the default SDPA path for training.

Forwards:
  * v7 (S % 256 == 0, the production path): 8-wave swapped-operand
    32x32x16 MFMA, in-register softmax, 3-slot all-glds K/V ring with
    ONE raw barrier per tile, native GQA — 299 us vs the aotriton
    library's 425 at the flagship shape.
  * v5 (S % 64 == 0 fallback): 4-wave 16x16x32 kernel with the LDS P
    round-trip; requires equal head counts (GQA expanded by the caller).

Backward (S % 256 == 0): delta + dq + dv + dk kernels (fwd+bwd 2858 us
vs aotriton's 4140 — profiles/attention_kernel_notes.md). The forward
emits O + logsumexp at natural-log scale — exactly what
aten::_scaled_dot_product_flash_attention_backward consumes — so the
aten backward remains a drop-in fallback (AITJ_ATTN_BWD=aten, and the
automatic path for S % 256 != 0).
"""
from __future__ import annotations

import math
import os
from typing import Optional

import torch

from . import native


def _supported(q: torch.Tensor, k: torch.Tensor) -> bool:
    B, H, S, D = q.shape
    HKV = k.shape[1]
    return (q.dtype == torch.bfloat16 and D == 128
            and (S % 256 == 0 or (S % 64 == 0 and HKV == H))
            and q.stride(-1) == 1 and k.stride(-1) == 1
            and H % HKV == 0)


def _run_fwd(q, k, v, scale):
    B, H, S, D = q.shape
    HKV = k.shape[1]
    lib = native.load(require=True)
    out = torch.empty(B, H, S, D, dtype=torch.bfloat16, device=q.device)
    lse = torch.empty(B, H, S, dtype=torch.float32, device=q.device)
    rc = lib.attn_fwd(
        native.stream_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
        out.data_ptr(), lse.data_ptr(),
        q.stride(0), q.stride(1), q.stride(2),
        k.stride(0), k.stride(1), k.stride(2),
        v.stride(0), v.stride(1), v.stride(2),
        B, H, HKV, S, scale)
    native.check_rc(rc, "attn_fwd", f"B={B} H={H} HKV={HKV} S={S}")
    return out, lse


class _NativeFlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        out, lse = _run_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, gout):
        q, k, v, out, lse = ctx.saved_tensors
        B, H, S, D = q.shape
        if S % 256 == 0 and os.environ.get("AITJ_ATTN_BWD") != "aten":
            return (*_native_backward(q, k, v, out, lse, gout, ctx.scale),
                    None)
        # None -> undefined Tensor: selects the dense (non-varlen) path in
        # the aten flash backward (empty tensors select varlen and fail).
        philox = torch.empty(0, dtype=torch.int64, device=q.device)
        dq, dk, dv = torch.ops.aten._scaled_dot_product_flash_attention_backward(
            gout.contiguous(), q, k, v, out, lse,
            None, None, S, S, 0.0, True,
            philox, philox, scale=ctx.scale)
        return dq, dk, dv, None


def _native_backward(q, k, v, out, lse, gout, scale):
    """Hand-written CDNA4 flash backward (delta + dq + dv + dk kernels)."""
    B, H, S, D = q.shape
    HKV = k.shape[1]
    lib = native.load(require=True)
    dout = gout.contiguous()
    delta = torch.empty(B, H, S, dtype=torch.float32, device=q.device)
    dq = torch.empty(B, H, S, D, dtype=torch.bfloat16, device=q.device)
    dk = torch.empty(B, HKV, S, D, dtype=torch.bfloat16, device=q.device)
    dv = torch.empty(B, HKV, S, D, dtype=torch.bfloat16, device=q.device)
    rc = lib.attn_bwd(
        native.stream_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
        out.data_ptr(), dout.data_ptr(), lse.data_ptr(), delta.data_ptr(),
        dq.data_ptr(), dk.data_ptr(), dv.data_ptr(),
        q.stride(0), q.stride(1), q.stride(2),
        k.stride(0), k.stride(1), k.stride(2),
        v.stride(0), v.stride(1), v.stride(2),
        B, H, HKV, S, scale)
    native.check_rc(rc, "attn_bwd", f"B={B} H={H} HKV={HKV} S={S}")
    return dq, dk, dv


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    scale: Optional[float] = None) -> torch.Tensor:
    """Causal flash attention over [B, H, S, D=128] bf16; k/v may carry
    fewer (GQA) heads when S % 256 == 0 (the v6 kernel maps them)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    assert _supported(q, k), (
        f"native flash attention needs bf16, D=128, S%256==0 (or S%64==0 "
        f"with equal head counts), d-contiguous strides; got {q.shape} "
        f"{q.dtype} kv={k.shape}")
    return _NativeFlashAttention.apply(q, k, v, scale)


def flash_attention_fwd_only(q, k, v, scale=None):
    """Forward-only entry returning (out, lse) for tests/benchmarks."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    return _run_fwd(q, k, v, scale)


def flash_attention_fwd_nw8(q, k, v, scale=None):
    """The 8-wave (BM=256) instantiation, kept callable for A/B."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    B, H, S, D = q.shape
    HKV = k.shape[1]
    lib = native.load(require=True)
    out = torch.empty(B, H, S, D, dtype=torch.bfloat16, device=q.device)
    lse = torch.empty(B, H, S, dtype=torch.float32, device=q.device)
    rc = lib.attn_fwd_nw8(
        native.stream_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
        out.data_ptr(), lse.data_ptr(),
        q.stride(0), q.stride(1), q.stride(2),
        k.stride(0), k.stride(1), k.stride(2),
        v.stride(0), v.stride(1), v.stride(2),
        B, H, HKV, S, scale)
    native.check_rc(rc, "attn_fwd_nw8", f"B={B} H={H} S={S}")
    return out, lse


def flash_attention_fwd_v7(q, k, v, scale=None):
    """The 3-deep all-glds pipeline forward (A/B candidate)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    B, H, S, D = q.shape
    HKV = k.shape[1]
    lib = native.load(require=True)
    out = torch.empty(B, H, S, D, dtype=torch.bfloat16, device=q.device)
    lse = torch.empty(B, H, S, dtype=torch.float32, device=q.device)
    rc = lib.attn_fwd_v7(
        native.stream_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
        out.data_ptr(), lse.data_ptr(),
        q.stride(0), q.stride(1), q.stride(2),
        k.stride(0), k.stride(1), k.stride(2),
        v.stride(0), v.stride(1), v.stride(2),
        B, H, HKV, S, scale)
    native.check_rc(rc, "attn_fwd_v7", f"B={B} H={H} S={S}")
    return out, lse


def flash_attention_fwd_v5(q, k, v, scale=None):
    """The 4-wave v5 forward, kept callable for A/B benchmarking."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    B, H, S, D = q.shape
    lib = native.load(require=True)
    out = torch.empty(B, H, S, D, dtype=torch.bfloat16, device=q.device)
    lse = torch.empty(B, H, S, dtype=torch.float32, device=q.device)
    rc = lib.attn_fwd_v5(
        native.stream_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
        out.data_ptr(), lse.data_ptr(),
        q.stride(0), q.stride(1), q.stride(2),
        k.stride(0), k.stride(1), k.stride(2),
        v.stride(0), v.stride(1), v.stride(2),
        B, H, S, scale)
    native.check_rc(rc, "attn_fwd_v5", f"B={B} H={H} S={S}")
    return out, lse
