"""Pure-PyTorch reference implementations of every HIP op.

Used (a) as the CPU execution path (gloo multi-process tests run these),
(b) as the fp32 numerics baseline GPU tests compare the HIP kernels against
(the repo-wide rule: numerics tests for a HIP kernel compare it against a
plain PyTorch fp32 reference of the same op).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def rmsnorm_fwd(x: torch.Tensor, res_in: Optional[torch.Tensor],
                w: torch.Tensor, eps: float) -> Tuple[torch.Tensor, Optional[torch.Tensor], torch.Tensor]:
    """Returns (y, res_out, rrms). Math in fp32, outputs cast back."""
    if res_in is not None:
        res_out = (x.float() + res_in.float()).to(x.dtype)
        xr = res_out.float()
    else:
        res_out = None
        xr = x.float()
    rrms = torch.rsqrt(xr.pow(2).mean(dim=-1, keepdim=True) + eps)
    y = (xr * rrms * w.float()).to(x.dtype)
    return y, res_out, rrms.squeeze(-1)


def rmsnorm_bwd(dy: torch.Tensor, res_out: torch.Tensor, w: torch.Tensor,
                rrms: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (dx, dw) in fp32 math."""
    H = res_out.shape[-1]
    xr = res_out.float()
    r = rrms.unsqueeze(-1).float()
    dxh = dy.float() * w.float()
    dot = (dxh * xr).sum(dim=-1, keepdim=True)
    dx = r * dxh - xr * (r ** 3) * dot / H
    dw = (dy.float() * xr * r).reshape(-1, H).sum(dim=0)
    return dx.to(dy.dtype), dw


def rope_rotate(x: torch.Tensor, inv_freq: torch.Tensor, seq_len: int,
                sign: float = 1.0) -> torch.Tensor:
    """x: [T, n_heads, D]; neox half-rotation; pos = token_index % seq_len."""
    T, n_heads, D = x.shape
    half = D // 2
    pos = (torch.arange(T, device=x.device) % seq_len).float()
    ang = pos[:, None] * inv_freq[None, :].float()  # [T, half]
    cos = torch.cos(ang)[:, None, :]                # [T, 1, half]
    sin = torch.sin(ang)[:, None, :] * sign
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    o1 = x1 * cos - x2 * sin
    o2 = x1 * sin + x2 * cos
    return torch.cat([o1, o2], dim=-1).to(x.dtype)


def swiglu_fwd(g: torch.Tensor, u: torch.Tensor) -> torch.Tensor:
    gf = g.float()
    return (gf * torch.sigmoid(gf) * u.float()).to(g.dtype)


def swiglu_bwd(dout: torch.Tensor, g: torch.Tensor,
               u: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    gf, uf, dof = g.float(), u.float(), dout.float()
    sg = torch.sigmoid(gf)
    silu = gf * sg
    du = dof * silu
    dg = dof * uf * sg * (1 + gf * (1 - sg))
    return dg.to(g.dtype), du.to(u.dtype)


def ce_fwd(logits: torch.Tensor, targets: torch.Tensor,
           ignore_index: int = -100) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (loss_per_row fp32 [T], lse fp32 [T]); ignored rows get 0."""
    lf = logits.float()
    lse = torch.logsumexp(lf, dim=-1)
    valid = targets != ignore_index
    safe_t = targets.clamp_min(0)
    picked = lf.gather(-1, safe_t.unsqueeze(-1).long()).squeeze(-1)
    loss = torch.where(valid, lse - picked, torch.zeros_like(lse))
    return loss, lse


def ce_bwd(logits: torch.Tensor, targets: torch.Tensor, lse: torch.Tensor,
           gscale: torch.Tensor, ignore_index: int = -100) -> torch.Tensor:
    lf = logits.float()
    p = torch.exp(lf - lse.unsqueeze(-1))
    valid = (targets != ignore_index)
    safe_t = targets.clamp_min(0).long()
    p.scatter_add_(-1, safe_t.unsqueeze(-1),
                   -torch.ones_like(lse).unsqueeze(-1))
    p = p * (gscale * valid.float()).unsqueeze(-1)
    return p.to(logits.dtype)


def adamw_step(p32: torch.Tensor, m: torch.Tensor, v: torch.Tensor,
               grad: torch.Tensor, p_bf16: torch.Tensor, lr: float,
               beta1: float, beta2: float, eps: float, weight_decay: float,
               step: int, clip: float = 0.0,
               normsq: Optional[torch.Tensor] = None,
               pre_scale: float = 1.0) -> None:
    """In-place flat AdamW matching the HIP kernel (decoupled weight decay).
    pre_scale folds the DDP 1/world_size average into the update."""
    g = grad.float() * pre_scale
    if clip > 0.0 and normsq is not None:
        norm = normsq.sqrt() * pre_scale
        g = g * (clip / torch.clamp(norm, min=clip))
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    mhat = m / bc1
    vhat = v / bc2
    p32.add_(-lr * (mhat / (vhat.sqrt() + eps) + weight_decay * p32))
    p_bf16.copy_(p32.to(p_bf16.dtype))
