"""Autograd-wrapped ops: HIP/CDNA4 kernels on GPU, PyTorch reference on CPU.

GPU dispatch is strict: if a tensor is on a HIP device and the in-tree
libhipops.so is missing, these ops raise instead of silently running eager
PyTorch (the native path must be the one that runs on a GPU box).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import native, reference


def _hip():
    return native.load(require=True)


def _ptr(t: Optional[torch.Tensor]):
    return None if t is None else t.data_ptr()


# ---------------------------------------------------------------------------
# Fused residual-add + RMSNorm
# ---------------------------------------------------------------------------

class _FusedRMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor,
                residual: Optional[torch.Tensor], eps: float):
        H = x.shape[-1]
        rows = x.numel() // H
        if x.is_cuda:
            assert x.dtype == torch.bfloat16 and x.is_contiguous()
            lib = _hip()
            y = torch.empty_like(x)
            res_out = torch.empty_like(x) if residual is not None else None
            rrms = torch.empty(rows, dtype=torch.float32, device=x.device)
            rc = lib.rmsnorm_fwd(native.stream_ptr(), _ptr(x),
                                 _ptr(residual), _ptr(weight), _ptr(y),
                                 _ptr(res_out), _ptr(rrms), rows, H, eps)
            native.check_rc(rc, "rmsnorm_fwd", f"H={H}")
        else:
            y, res_out, rrms = reference.rmsnorm_fwd(x, residual, weight, eps)
        saved_res = res_out if res_out is not None else x
        ctx.save_for_backward(saved_res, weight, rrms)
        ctx.has_residual = residual is not None
        return (y, res_out) if residual is not None else (y, x)

    @staticmethod
    def backward(ctx, dy: torch.Tensor, dres: Optional[torch.Tensor]):
        res_out, weight, rrms = ctx.saved_tensors
        H = res_out.shape[-1]
        rows = res_out.numel() // H
        if dy.is_cuda:
            lib = _hip()
            dy = dy.contiguous()
            if dres is not None:
                dres = dres.contiguous()
            dx = torch.empty_like(res_out)
            P = lib.rmsnorm_bwd_partials(rows, H)
            dw_partial = torch.empty(P, H, dtype=torch.float32,
                                     device=dy.device)
            # dres (grad via the residual stream) is fused into dx in-kernel
            rc = lib.rmsnorm_bwd(native.stream_ptr(), _ptr(dy),
                                 _ptr(res_out), _ptr(weight), _ptr(rrms),
                                 _ptr(dres), _ptr(dx), _ptr(dw_partial),
                                 rows, H)
            native.check_rc(rc, "rmsnorm_bwd", f"H={H}")
            dw32 = torch.zeros(H, dtype=torch.float32, device=dy.device)
            rc = lib.rmsnorm_dw_reduce(native.stream_ptr(),
                                       _ptr(dw_partial), P, _ptr(dw32), H)
            native.check_rc(rc, "rmsnorm_dw_reduce", f"H={H} P={P}")
            dw = dw32.to(weight.dtype)
        else:
            dx, dw32 = reference.rmsnorm_bwd(dy, res_out, weight, rrms)
            dw = dw32.to(weight.dtype)
            # res_out = x + residual: both receive dx plus any gradient that
            # arrived through res_out directly.
            if dres is not None:
                dx = dx + dres
        if ctx.has_residual:
            return dx, dw, dx, None
        return dx, dw, None, None


def fused_rmsnorm(x: torch.Tensor, weight: torch.Tensor,
                  residual: Optional[torch.Tensor] = None,
                  eps: float = 1e-5) -> Tuple[torch.Tensor, torch.Tensor]:
    """y = RMSNorm(x + residual) * weight.

    Returns (y, new_residual): new_residual = x + residual (the pre-norm
    stream the next block consumes); when residual is None it is x itself.
    One HBM round-trip on GPU (fused add+norm; reference operator has no
    compute at all — SURVEY.md §2.3).
    """
    return _FusedRMSNorm.apply(x, weight, residual, eps)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------

class _Rope(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, inv_freq: torch.Tensor, seq_len: int):
        ctx.seq_len = seq_len
        ctx.save_for_backward(inv_freq)
        return _rope_run(x, inv_freq, seq_len, 1.0)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        (inv_freq,) = ctx.saved_tensors
        return _rope_run(dy.contiguous(), inv_freq, ctx.seq_len, -1.0), None, None


def _rope_run(x: torch.Tensor, inv_freq: torch.Tensor, seq_len: int,
              sign: float) -> torch.Tensor:
    orig_shape = x.shape
    D = x.shape[-1]
    n_heads = x.shape[-2]
    xt = x.reshape(-1, n_heads, D)
    if x.is_cuda:
        assert x.dtype == torch.bfloat16 and xt.is_contiguous()
        assert inv_freq.device == x.device and \
            inv_freq.dtype == torch.float32, (
            f"inv_freq must be fp32 on {x.device}, got "
            f"{inv_freq.dtype}@{inv_freq.device} (a host pointer would "
            f"memory-fault the GPU)")
        lib = _hip()
        out = torch.empty_like(xt)
        rc = lib.rope(native.stream_ptr(), _ptr(xt), _ptr(out),
                      _ptr(inv_freq), xt.shape[0], n_heads, seq_len, D, sign)
        native.check_rc(rc, "rope", f"D={D}")
    else:
        out = reference.rope_rotate(xt, inv_freq, seq_len, sign)
    return out.reshape(orig_shape)


def apply_rope(x: torch.Tensor, inv_freq: torch.Tensor,
               seq_len: int) -> torch.Tensor:
    """Neox-style rotary embedding over [..., S, n_heads, D] (token dim folded:
    position = flat_token_index % seq_len)."""
    return _Rope.apply(x, inv_freq, seq_len)


def make_inv_freq(head_dim: int, theta: float = 500000.0,
                  device=None) -> torch.Tensor:
    # device=None respects an ambient `with torch.device(...)` context —
    # an explicit "cpu" default left standalone PP stages with a host
    # inv_freq under a CUDA model (GPU memory fault in the rope kernel).
    return 1.0 / (theta ** (torch.arange(0, head_dim, 2,
                                         dtype=torch.float32,
                                         device=device) / head_dim))


# ---------------------------------------------------------------------------
# SwiGLU
# ---------------------------------------------------------------------------

class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g: torch.Tensor, u: torch.Tensor):
        ctx.save_for_backward(g, u)
        if g.is_cuda:
            assert g.dtype == torch.bfloat16 and g.is_contiguous()
            lib = _hip()
            out = torch.empty_like(g)
            rc = lib.swiglu_fwd(native.stream_ptr(), _ptr(g), _ptr(u),
                                _ptr(out), g.numel())
            native.check_rc(rc, "swiglu_fwd", f"n={g.numel()}")
            return out
        return reference.swiglu_fwd(g, u)

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        g, u = ctx.saved_tensors
        if g.is_cuda:
            lib = _hip()
            dout = dout.contiguous()
            dg = torch.empty_like(g)
            du = torch.empty_like(u)
            rc = lib.swiglu_bwd(native.stream_ptr(), _ptr(dout), _ptr(g),
                                _ptr(u), _ptr(dg), _ptr(du), g.numel())
            native.check_rc(rc, "swiglu_bwd", f"n={g.numel()}")
            return dg, du
        return reference.swiglu_bwd(dout, g, u)


def swiglu(g: torch.Tensor, u: torch.Tensor) -> torch.Tensor:
    """silu(g) * u, fused (backward recomputes silu from g)."""
    return _SwiGLU.apply(g, u)


class _SwiGLUPacked(torch.autograd.Function):
    """Consumes the fused gate_up GEMM output [..., 2F] directly (rows are
    [gate | up]); backward emits the packed [.., 2F] gradient so the GEMM
    backward needs no concat."""

    @staticmethod
    def forward(ctx, gu: torch.Tensor):
        F2 = gu.shape[-1]
        F = F2 // 2
        rows = gu.numel() // F2
        ctx.save_for_backward(gu)
        if gu.is_cuda:
            assert gu.dtype == torch.bfloat16 and gu.is_contiguous()
            lib = _hip()
            out = torch.empty(*gu.shape[:-1], F, dtype=gu.dtype,
                              device=gu.device)
            rc = lib.swiglu_packed_fwd(native.stream_ptr(), _ptr(gu),
                                       _ptr(out), rows, F)
            native.check_rc(rc, "swiglu_packed_fwd", f"F={F}")
            return out
        g, u = gu.split([F, F], dim=-1)
        return reference.swiglu_fwd(g.contiguous(), u.contiguous())

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        (gu,) = ctx.saved_tensors
        F = gu.shape[-1] // 2
        rows = gu.numel() // (2 * F)
        if gu.is_cuda:
            lib = _hip()
            dout = dout.contiguous()
            dgu = torch.empty_like(gu)
            rc = lib.swiglu_packed_bwd(native.stream_ptr(), _ptr(dout),
                                       _ptr(gu), _ptr(dgu), rows, F)
            native.check_rc(rc, "swiglu_packed_bwd", f"F={F}")
            return dgu
        g, u = gu.split([F, F], dim=-1)
        dg, du = reference.swiglu_bwd(dout, g.contiguous(), u.contiguous())
        return torch.cat([dg, du], dim=-1)


def swiglu_packed(gu: torch.Tensor) -> torch.Tensor:
    """SwiGLU over packed [..., 2F] rows ([gate | up]) from a fused GEMM."""
    return _SwiGLUPacked.apply(gu)


def decode_attention(q: torch.Tensor, kc: torch.Tensor,
                     vc: torch.Tensor, pos_t: torch.Tensor,
                     scale: float) -> Optional[torch.Tensor]:
    """Fused one-token GQA attention over the KV cache (hip/ops.hip
    attn_decode: split-L flash-decoding partials + log-sum-exp combine,
    2 launches instead of the ~10-kernel einsum chain). The valid length
    is read from pos_t ON DEVICE, so the call replays inside a hipGraph.
    Returns [B, H, 1, D], or None when the shape is unsupported
    (caller falls back to the einsum path). Inference-only."""
    B, H, S, D = q.shape
    n_kv, Lmax = kc.shape[1], kc.shape[2]
    G = H // n_kv if H % n_kv == 0 else 0
    if not (q.is_cuda and q.dtype == torch.bfloat16 and D == 128
            and S == 1 and G >= 1 and kc.is_contiguous()
            and vc.is_contiguous() and pos_t.dtype == torch.int64):
        return None
    lib = _hip()
    q2 = q.reshape(B, H, D).contiguous()
    n_chunk = (Lmax + 63) // 64
    partial = torch.empty(B, n_kv, n_chunk, G, 130, dtype=torch.float32,
                          device=q.device)
    o = torch.empty(B, H, D, dtype=torch.bfloat16, device=q.device)
    rc = lib.attn_decode(native.stream_ptr(), _ptr(q2), _ptr(kc),
                         _ptr(vc), _ptr(pos_t), _ptr(partial), _ptr(o),
                         B, H, n_kv, Lmax, n_chunk, scale)
    native.check_rc(rc, "attn_decode", f"H={H} n_kv={n_kv} Lmax={Lmax}")
    return o[:, :, None, :]


def decode_swiglu(x: torch.Tensor,
                  gate_up_weight: torch.Tensor) -> Optional[torch.Tensor]:
    """silu(x @ Wg^T) * (x @ Wu^T) over the packed [2F, K] gate|up weight
    in ONE kernel (hip/ops.hip gemv_swiglu): same weight bytes as the
    plain decode GEMV, no 2F-wide intermediate, no separate swiglu
    launch. Returns [..., F], or None on unsupported shapes (caller
    falls back to decode_linear + swiglu_packed). Inference-only."""
    K = x.shape[-1]
    rows = x.numel() // K
    F = gate_up_weight.shape[0] // 2
    if not (x.is_cuda and x.dtype == torch.bfloat16
            and gate_up_weight.dtype == torch.bfloat16 and 1 <= rows <= 8
            and K % 512 == 0 and gate_up_weight.stride(-1) == 1
            and gate_up_weight.shape[0] % 2 == 0):
        return None
    lib = _hip()
    x2 = x.reshape(rows, K).contiguous()
    y = torch.empty(rows, F, dtype=torch.bfloat16, device=x.device)
    rc = lib.gemv_swiglu(native.stream_ptr(), _ptr(gate_up_weight),
                         _ptr(x2), _ptr(y), F, K, rows)
    native.check_rc(rc, "gemv_swiglu", f"F={F} K={K} N={rows}")
    return y.reshape(*x.shape[:-1], F)


def decode_linear(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """y = x @ W^T for decode-sized x (<= 8 total rows): the hand-written
    wave-per-row GEMV (hip/ops.hip gemv_bf16) instead of hipBLASLt, which
    leaves most of HBM3E idle on these skinny shapes. Inference-only (no
    autograd); falls back to F.linear off-GPU or on unsupported shapes."""
    K = x.shape[-1]
    rows = x.numel() // K
    if (x.is_cuda and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16 and 1 <= rows <= 8
            and K % 512 == 0 and weight.stride(-1) == 1):
        lib = _hip()
        x2 = x.reshape(rows, K).contiguous()
        M = weight.shape[0]
        y = torch.empty(rows, M, dtype=torch.bfloat16, device=x.device)
        rc = lib.gemv_bf16(native.stream_ptr(), _ptr(weight), _ptr(x2),
                           _ptr(y), M, K, rows)
        native.check_rc(rc, "gemv_bf16", f"M={M} K={K} N={rows}")
        return y.reshape(*x.shape[:-1], M)
    return torch.nn.functional.linear(x, weight)


# ---------------------------------------------------------------------------
# Fused cross-entropy
# ---------------------------------------------------------------------------

class _FusedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, targets: torch.Tensor,
                ignore_index: int):
        T, V = logits.shape
        if logits.is_cuda:
            assert logits.dtype == torch.bfloat16 and logits.is_contiguous()
            assert V % 8 == 0
            lib = _hip()
            t32 = targets.to(torch.int32).contiguous()
            lse = torch.empty(T, dtype=torch.float32, device=logits.device)
            loss = torch.empty(T, dtype=torch.float32, device=logits.device)
            rc = lib.ce_fwd(native.stream_ptr(), _ptr(logits), _ptr(t32),
                            _ptr(lse), _ptr(loss), T, V, ignore_index)
            native.check_rc(rc, "ce_fwd", f"V={V}")
        else:
            loss, lse = reference.ce_fwd(logits, targets, ignore_index)
            t32 = targets.to(torch.int32)
        ctx.save_for_backward(logits, t32, lse)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, gout: torch.Tensor):
        logits, t32, lse = ctx.saved_tensors
        T, V = logits.shape
        gscale = gout.to(torch.float32).contiguous()
        if logits.is_cuda:
            lib = _hip()
            dlogits = torch.empty_like(logits)
            rc = lib.ce_bwd(native.stream_ptr(), _ptr(logits), _ptr(t32),
                            _ptr(lse), _ptr(gscale), _ptr(dlogits), T, V,
                            ctx.ignore_index)
            native.check_rc(rc, "ce_bwd", f"V={V}")
        else:
            dlogits = reference.ce_bwd(logits, t32, lse, gscale,
                                       ctx.ignore_index)
        return dlogits, None, None


def fused_cross_entropy(logits: torch.Tensor, targets: torch.Tensor,
                        ignore_index: int = -100) -> torch.Tensor:
    """Per-token CE loss vector (fp32, [T]); rows with ignore_index get 0.
    Mean it over the valid-token count for the training loss."""
    return _FusedCE.apply(logits, targets, ignore_index)


# ---------------------------------------------------------------------------
# MoE token-dispatch row ops (HIP on GPU, torch indexing on CPU)
# ---------------------------------------------------------------------------

class _GatherRows(torch.autograd.Function):
    """out[i] = x[idx[i]] over bf16 rows. bijective=True promises idx is a
    permutation (backward is a conflict-free scatter); False accumulates
    duplicate rows through an fp32 buffer."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, idx: torch.Tensor, bijective: bool):
        ctx.bijective = bijective
        ctx.n_in = x.shape[0]
        ctx.save_for_backward(idx)
        if x.is_cuda and x.dtype == torch.bfloat16 and \
                x.shape[-1] % 8 == 0:
            lib = _hip()
            x = x.contiguous()
            idx = idx.contiguous()
            out = torch.empty(idx.shape[0], x.shape[1], dtype=x.dtype,
                              device=x.device)
            rc = lib.rows_gather(native.stream_ptr(), _ptr(x), _ptr(idx),
                                 _ptr(out), idx.shape[0], x.shape[1])
            native.check_rc(rc, "rows_gather", f"H={x.shape[1]}")
            return out
        return x[idx]

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        (idx,) = ctx.saved_tensors
        H = dout.shape[-1]
        if dout.is_cuda and dout.dtype == torch.bfloat16 and H % 8 == 0:
            lib = _hip()
            dout = dout.contiguous()
            if ctx.bijective:
                dx = torch.empty(ctx.n_in, H, dtype=dout.dtype,
                                 device=dout.device)
                rc = lib.rows_scatter(native.stream_ptr(), _ptr(dout),
                                      _ptr(idx), _ptr(dx), idx.shape[0], H)
                native.check_rc(rc, "rows_scatter", f"H={H}")
            else:
                acc = torch.zeros(ctx.n_in, H, dtype=torch.float32,
                                  device=dout.device)
                rc = lib.rows_scatter_add_f32(native.stream_ptr(),
                                              _ptr(dout), _ptr(idx),
                                              _ptr(acc), idx.shape[0], H)
                native.check_rc(rc, "rows_scatter_add_f32", f"H={H}")
                dx = acc.to(dout.dtype)
            return dx, None, None
        dx = torch.zeros(ctx.n_in, H, dtype=dout.dtype, device=dout.device)
        dx.index_add_(0, idx, dout)
        return dx, None, None


def gather_rows(x: torch.Tensor, idx: torch.Tensor,
                bijective: bool = False) -> torch.Tensor:
    return _GatherRows.apply(x, idx, bijective)


class _MoECombine(torch.autograd.Function):
    """y[t] = sum_j gates[t, j] * src[inv[t*K + j]] — the MoE un-permute +
    gate + top-k combine in one pass (inv is a permutation of src rows)."""

    @staticmethod
    def forward(ctx, src: torch.Tensor, inv: torch.Tensor,
                gates: torch.Tensor):
        T, K = gates.shape
        H = src.shape[-1]
        ctx.save_for_backward(src, inv, gates)
        if src.is_cuda and src.dtype == torch.bfloat16 and H % 8 == 0:
            lib = _hip()
            src = src.contiguous()
            g32 = gates.reshape(-1).to(torch.float32).contiguous()
            y = torch.empty(T, H, dtype=src.dtype, device=src.device)
            rc = lib.moe_combine(native.stream_ptr(), _ptr(src), _ptr(inv),
                                 _ptr(g32), _ptr(y), T, K, H)
            native.check_rc(rc, "moe_combine", f"H={H} K={K}")
            return y
        pair = src[inv] * gates.reshape(-1, 1).to(src.dtype)
        return pair.reshape(T, K, H).sum(dim=1)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        src, inv, gates = ctx.saved_tensors
        T, K = gates.shape
        H = src.shape[-1]
        if dy.is_cuda and dy.dtype == torch.bfloat16 and H % 8 == 0:
            lib = _hip()
            dy = dy.contiguous()
            g32 = gates.reshape(-1).to(torch.float32).contiguous()
            dsrc = torch.empty_like(src)
            dg32 = torch.empty(T * K, dtype=torch.float32, device=dy.device)
            rc = lib.moe_combine_bwd(native.stream_ptr(), _ptr(src),
                                     _ptr(dy), _ptr(inv), _ptr(g32),
                                     _ptr(dsrc), _ptr(dg32), T * K, K, H)
            native.check_rc(rc, "moe_combine_bwd", f"H={H} K={K}")
            return dsrc, None, dg32.reshape(T, K).to(gates.dtype)
        dpair = dy.unsqueeze(1).expand(T, K, H).reshape(T * K, H)
        dsrc = torch.zeros_like(src)
        dsrc.index_add_(0, inv,
                        dpair * gates.reshape(-1, 1).to(src.dtype))
        dg = (src[inv].float() * dpair.float()).sum(-1).reshape(T, K)
        return dsrc, None, dg.to(gates.dtype)


def moe_combine(src: torch.Tensor, inv: torch.Tensor,
                gates: torch.Tensor) -> torch.Tensor:
    return _MoECombine.apply(src, inv, gates)


class _DispatchRows(torch.autograd.Function):
    """MoE dispatch gather: out[p] = x[tok[p]] where each token appears in
    exactly K pairs. Backward is an atomic-free gather-sum using the
    inverse pair permutation: dx[t] = sum_j dout[inv[t*K+j]] (the
    rows_scatter_add atomic path measured 435 us/call at moe-mid)."""

    @staticmethod
    def forward(ctx, x, tok_idx, inv_pairs, K: int):
        ctx.save_for_backward(inv_pairs)
        ctx.K = K
        ctx.n_in = x.shape[0]
        if x.is_cuda and x.dtype == torch.bfloat16 and x.shape[-1] % 8 == 0:
            lib = _hip()
            x = x.contiguous()
            out = torch.empty(tok_idx.shape[0], x.shape[1], dtype=x.dtype,
                              device=x.device)
            rc = lib.rows_gather(native.stream_ptr(), _ptr(x), _ptr(tok_idx),
                                 _ptr(out), tok_idx.shape[0], x.shape[1])
            native.check_rc(rc, "rows_gather", f"H={x.shape[1]}")
            return out
        return x[tok_idx]

    @staticmethod
    def backward(ctx, dout):
        (inv_pairs,) = ctx.saved_tensors
        K = ctx.K
        H = dout.shape[-1]
        T = ctx.n_in
        if dout.is_cuda and dout.dtype == torch.bfloat16 and H % 8 == 0:
            lib = _hip()
            dout = dout.contiguous()
            dx = torch.empty(T, H, dtype=dout.dtype, device=dout.device)
            rc = lib.moe_combine(native.stream_ptr(), _ptr(dout),
                                 _ptr(inv_pairs), None, _ptr(dx), T, K, H)
            native.check_rc(rc, "moe_combine(dispatch-bwd)", f"H={H}")
            return dx, None, None, None
        dx = dout[inv_pairs.reshape(-1)].reshape(T, K, H).sum(dim=1)
        return dx, None, None, None


def dispatch_rows(x: torch.Tensor, tok_idx: torch.Tensor,
                  inv_pairs: torch.Tensor, K: int) -> torch.Tensor:
    return _DispatchRows.apply(x, tok_idx, inv_pairs, K)
