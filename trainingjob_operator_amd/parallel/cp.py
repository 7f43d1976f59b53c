"""Context parallelism: ring attention over sequence-sharded activations.

The last parallel axis in the family (DP/TP/SP/PP/EP/PS are siblings in
this package): every rank holds a FULL replica of the weights but only a
contiguous S/cp block of the sequence. Everything except attention is
token-local (embedding, norms, MLP, loss), so the one distributed op is
attention itself: K/V blocks circulate around the ring while each rank's
Q stays put, partial softmax state merging exactly like flash attention
merges tiles (running max + sum-of-exp + rescaled accumulator). Causality
means rank r only attends blocks j <= r — later blocks still transit the
ring (uniform communication) but skip compute.

The backward is a second ring pass: dK/dV accumulators travel WITH their
K/V blocks (each visiting rank adds its contribution; after cp hops the
block and its gradient are home), while dQ accumulates locally. P is
recomputed per block from the saved logsumexp, flash-style, so activation
memory stays O(Sb) per rank.

xGMI mapping: ring neighbor exchange is exactly one point-to-point link
per hop (7 links x ~153 GB/s per MI355X), the communication pattern ring
attention was designed for; compute of block t overlaps the transit of
block t+1 when the blocks are large enough to cover link latency.

Distinct from sequence parallelism (sp.py): SP sharding spans the TP
group and gathers the FULL sequence back for attention (it saves memory
on norms/residuals only); CP never materializes the full sequence
anywhere — it is how a context longer than one GPU's activation budget
trains at all.

Gloo-verified against the unsharded model (loss + every weight gradient)
in tests/test_cp_gloo.py. Per-block attention math runs in fp32 through
torch ops; wiring the native MFMA flash kernels under the ring is a
future-round optimization (ROADMAP.md).

No reference counterpart: the reference operator has no parallelism at
all (SURVEY.md §2.3 — replica counts are its only notion of scale).
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ..models.config import LlamaConfig
from ..ops import fused_cross_entropy, fused_rmsnorm, make_inv_freq
from .tp import _group_size


def _ring_shift(tensors: List[torch.Tensor], group) -> List[torch.Tensor]:
    """Send every tensor to rank+1, receive from rank-1 (one xGMI hop).
    Even ranks send first, odd ranks receive first — no deadlock at any
    ring size (including 2, where isend/irecv pair up)."""
    n = _group_size(group)
    r = dist.get_rank(group)
    dst = dist.get_global_rank(group, (r + 1) % n) if group else (r + 1) % n
    src = dist.get_global_rank(group, (r - 1) % n) if group else (r - 1) % n
    sends = [t.contiguous() for t in tensors]
    outs = [torch.empty_like(t) for t in sends]
    reqs = []
    if r % 2 == 0:
        for t in sends:
            reqs.append(dist.isend(t, dst, group=group))
        for o in outs:
            reqs.append(dist.irecv(o, src, group=group))
    else:
        for o in outs:
            reqs.append(dist.irecv(o, src, group=group))
        for t in sends:
            reqs.append(dist.isend(t, dst, group=group))
    for q in reqs:
        q.wait()
    return outs


def _block_scores(q32: torch.Tensor, kj: torch.Tensor, scale: float,
                  diagonal: bool) -> torch.Tensor:
    """scale * q @ k^T with the in-block causal mask when this is the
    diagonal block (global causality between blocks is handled by the
    j <= r schedule)."""
    s = torch.matmul(q32, kj.float().transpose(-1, -2)) * scale
    if diagonal:
        Sb = s.shape[-1]
        mask = torch.ones(Sb, Sb, dtype=torch.bool,
                          device=s.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    return s


class _RingAttention(torch.autograd.Function):
    """Causal ring attention over [B, H, Sb, D] blocks (equal kv heads —
    the module expands GQA before the ring)."""

    @staticmethod
    def forward(ctx, q, k, v, group, scale):
        n = _group_size(group)
        r = dist.get_rank(group) if n > 1 else 0
        q32 = q.float()
        B, H, Sb, D = q.shape
        o = torch.zeros(B, H, Sb, D, dtype=torch.float32, device=q.device)
        m = torch.full((B, H, Sb, 1), float("-inf"), device=q.device)
        l = torch.zeros(B, H, Sb, 1, device=q.device)
        kj, vj = k, v
        j = r
        for t in range(n):
            if j <= r:
                s = _block_scores(q32, kj, scale, diagonal=(j == r))
                m_new = torch.maximum(m, s.amax(dim=-1, keepdim=True))
                p = torch.exp(s - m_new)
                alpha = torch.exp(m - m_new)
                l = l * alpha + p.sum(dim=-1, keepdim=True)
                o = o * alpha + torch.matmul(p, vj.float())
                m = m_new
            if t + 1 < n:
                kj, vj = _ring_shift([kj, vj], group)
                j = (j - 1) % n
        out32 = o / l
        lse = m + torch.log(l)
        ctx.save_for_backward(q, k, v, out32, lse)
        ctx.group, ctx.scale = group, scale
        return out32.to(q.dtype)

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out32, lse = ctx.saved_tensors
        group, scale = ctx.group, ctx.scale
        n = _group_size(group)
        r = dist.get_rank(group) if n > 1 else 0
        q32, do32 = q.float(), dout.float()
        delta = (do32 * out32).sum(dim=-1, keepdim=True)
        dq = torch.zeros_like(q32)
        kj, vj = k, v
        dkj = torch.zeros_like(k, dtype=torch.float32)
        dvj = torch.zeros_like(v, dtype=torch.float32)
        j = r
        for t in range(n):
            if j <= r:
                s = _block_scores(q32, kj, scale, diagonal=(j == r))
                p = torch.exp(s - lse)            # recomputed, flash-style
                dvj += torch.matmul(p.transpose(-1, -2), do32)
                dp = torch.matmul(do32, vj.float().transpose(-1, -2))
                ds = p * (dp - delta)
                dq += scale * torch.matmul(ds, kj.float())
                dkj += scale * torch.matmul(ds.transpose(-1, -2), q32)
            # n shifts total: the (k, dk, dv) triplet arrives back home
            if n > 1:
                kj, vj, dkj, dvj = _ring_shift([kj, vj, dkj, dvj], group)
                j = (j - 1) % n
        return (dq.to(q.dtype), dkj.to(k.dtype), dvj.to(v.dtype),
                None, None)


def ring_attention(q, k, v, group, scale: Optional[float] = None):
    """Causal ring attention; q/k/v [B, H, Sb, D] are this rank's
    sequence block with equal head counts."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    return _RingAttention.apply(q, k, v, group, scale)


def _rope_offset(x: torch.Tensor, inv_freq: torch.Tensor,
                 pos0: int) -> torch.Tensor:
    """Differentiable Neox rotation at positions pos0..pos0+Sb-1
    (x [B, Sb, nh, D]); the shard's global offset makes every rank agree
    on absolute positions."""
    B, Sb, nh, D = x.shape
    half = D // 2
    pos = (torch.arange(Sb, device=x.device) + pos0).float()
    ang = pos[:, None] * inv_freq[None, :].float()
    cos = ang.cos()[None, :, None, :]
    sin = ang.sin()[None, :, None, :]
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    return torch.cat([x1 * cos - x2 * sin,
                      x1 * sin + x2 * cos], dim=-1).to(x.dtype)


class CPAttention(nn.Module):
    """Attention over a sequence shard: replicated weights (same layout
    as models.llama.Attention), rope at global positions, GQA expanded
    before the ring (every rank needs full kv heads for its block)."""

    def __init__(self, cfg: LlamaConfig, group=None):
        super().__init__()
        self.cfg = cfg
        self.group = group
        H = cfg.hidden_size
        self.q_size = cfg.num_heads * cfg.head_dim
        self.kv_size = cfg.num_kv_heads * cfg.head_dim
        self.qkv_proj = nn.Linear(H, self.q_size + 2 * self.kv_size,
                                  bias=False)
        self.o_proj = nn.Linear(self.q_size, H, bias=False)

    def forward(self, x_s: torch.Tensor, inv_freq: torch.Tensor,
                pos0: int) -> torch.Tensor:
        B, Sb, _ = x_s.shape
        cfg = self.cfg
        qkv = self.qkv_proj(x_s)
        q, k, v = qkv.split([self.q_size, self.kv_size, self.kv_size],
                            dim=-1)
        q = q.reshape(B, Sb, cfg.num_heads, cfg.head_dim)
        k = k.reshape(B, Sb, cfg.num_kv_heads, cfg.head_dim)
        v = v.reshape(B, Sb, cfg.num_kv_heads, cfg.head_dim)
        q = _rope_offset(q, inv_freq, pos0).transpose(1, 2)
        k = _rope_offset(k, inv_freq, pos0).transpose(1, 2)
        v = v.transpose(1, 2)
        G = cfg.num_heads // cfg.num_kv_heads
        if G > 1:
            k = k.repeat_interleave(G, dim=1)
            v = v.repeat_interleave(G, dim=1)
        o = ring_attention(q, k, v, self.group)
        o = o.transpose(1, 2).reshape(B, Sb, self.q_size)
        return self.o_proj(o)


class CPBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig, group=None):
        super().__init__()
        self.cfg = cfg
        self.attn = CPAttention(cfg, group)
        from ..models.llama import MLP
        self.mlp = MLP(cfg)
        self.input_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attn_norm_weight = nn.Parameter(
            torch.ones(cfg.hidden_size))

    def forward(self, x_s, residual_s, inv_freq, pos0):
        normed, residual_s = fused_rmsnorm(x_s, self.input_norm_weight,
                                           residual_s, self.cfg.norm_eps)
        attn_out = self.attn(normed, inv_freq, pos0)
        normed, residual_s = fused_rmsnorm(attn_out,
                                           self.post_attn_norm_weight,
                                           residual_s, self.cfg.norm_eps)
        return self.mlp(normed), residual_s


class _AllReduceSumLoss(torch.autograd.Function):
    """fwd: sum the per-rank partial losses so every rank reports the
    GLOBAL loss; bwd: identity (d global / d local = 1) — each rank then
    backprops exactly its own partial term, and allreduce_cp_grads sums
    the resulting replica gradients into the global gradient."""

    @staticmethod
    def forward(ctx, x, group):
        y = x.clone()
        dist.all_reduce(y, group=group)
        return y

    @staticmethod
    def backward(ctx, g):
        return g, None


class CPLlamaModel(nn.Module):
    """Llama with every weight replicated and the sequence sharded into
    cp contiguous blocks (rank r owns tokens [r*Sb, (r+1)*Sb)). forward
    takes the FULL batch (every rank slices its own block — the callers
    already feed identical data to the group) and returns the global
    loss on every rank."""

    def __init__(self, cfg: LlamaConfig, group=None):
        super().__init__()
        self.cfg = cfg
        self.group = group
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(CPBlock(cfg, group)
                                    for _ in range(cfg.num_layers))
        self.final_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size,
                                 bias=False)
        self.register_buffer("inv_freq",
                             make_inv_freq(cfg.head_dim, cfg.rope_theta),
                             persistent=False)

    def forward(self, tokens: torch.Tensor,
                targets: Optional[torch.Tensor] = None):
        n = _group_size(self.group)
        r = dist.get_rank(self.group) if n > 1 else 0
        S = tokens.shape[1]
        assert S % max(n, 1) == 0, (S, n)
        Sb = S // max(n, 1)
        pos0 = r * Sb
        tok_s = tokens[:, pos0:pos0 + Sb]
        x_s = self.embed(tok_s)
        residual_s = None
        for blk in self.blocks:
            x_s, residual_s = blk(x_s, residual_s, self.inv_freq, pos0)
        normed_s, _ = fused_rmsnorm(x_s, self.final_norm_weight,
                                    residual_s, self.cfg.norm_eps)
        logits_s = self.lm_head(normed_s)
        if targets is None:
            return logits_s                 # this rank's block of logits
        tgt_s = targets[:, pos0:pos0 + Sb].reshape(-1)
        T = logits_s.shape[0] * logits_s.shape[1]
        per_tok = fused_cross_entropy(
            logits_s.reshape(T, -1).contiguous(), tgt_s)
        n_valid = (targets.reshape(-1) != -100).sum().clamp(min=1)
        local = per_tok.sum() / n_valid     # global denominator
        if n == 1:
            return local
        return _AllReduceSumLoss.apply(local, self.group)

    @torch.no_grad()
    def shard_from_full(self, full) -> None:
        """Weights replicate 1:1 from an unsharded LlamaModel (same
        module names/shapes — only the activations are sharded)."""
        mine = dict(self.named_parameters())
        for name, p in full.named_parameters():
            mine[name].copy_(p)

    @torch.no_grad()
    def allreduce_cp_grads(self) -> None:
        """Sum replica gradients over the CP group (each rank backpropped
        only its own partial loss term). Call after backward, before the
        optimizer — like DP's all-reduce but with SUM semantics because
        the partials already carry the global 1/n_valid."""
        if _group_size(self.group) == 1:
            return
        for p in self.parameters():
            if p.grad is not None:
                dist.all_reduce(p.grad, group=self.group)


# ---------------------------------------------------------------------------
# DP x CP trainer
# ---------------------------------------------------------------------------

class CPTopology:
    """world = dp x cp; cp peers are ADJACENT ranks (ring hops are one
    xGMI link) and share the data batch; dp replicas stream their own."""

    def __init__(self, cp_size: int = 0):
        if not dist.is_initialized():
            self.world, self.rank = 1, 0
            self.cp_size, self.cp_rank, self.cp_group = 1, 0, None
            self.dp_size, self.dp_rank = 1, 0
            return
        world = dist.get_world_size()
        rank = dist.get_rank()
        cp = cp_size or world
        if world % cp != 0:
            raise ValueError(f"world {world} not divisible by cp={cp}")
        self.world, self.rank = world, rank
        self.cp_size, self.dp_size = cp, world // cp
        self.cp_rank, self.dp_rank = rank % cp, rank // cp
        self.cp_group = None
        for d in range(self.dp_size):          # new_group is collective
            g = dist.new_group(list(range(d * cp, (d + 1) * cp)))
            if d == self.dp_rank:
                self.cp_group = g
        if self.cp_size == 1:
            self.cp_group = None


class CPTrainer:
    """Context-parallel trainer: weights replicated, sequence sharded
    cp-ways (ring attention), dp on top for throughput.

    Gradient seam: each rank backprops its own partial of the GLOBAL
    loss (see _AllReduceSumLoss), so the flat grad needs one SUM
    all-reduce over the whole world, pre-scaled by 1/dp for the
    data-parallel mean. After it every rank holds identical global
    gradients, so the optimizer's local grad-norm clip is globally
    correct and checkpoints are a single rank-0 stream (any-world-size
    resume like DP)."""

    def __init__(self, cfg, cp_size: int = 0, device=None):
        from ..launcher.data import make_batches
        from ..models.config import CONFIGS
        from ..optim import FlatAdamW
        from .flat import FlatParamStore

        self.cfg = cfg
        mcfg = CONFIGS[cfg.model]
        self.device = torch.device(device or "cpu")
        self.topo = CPTopology(cp_size)
        assert cfg.seq_len % self.topo.cp_size == 0, \
            (cfg.seq_len, self.topo.cp_size)
        torch.manual_seed(cfg.seed)            # identical init everywhere
        with torch.device(self.device):
            model = CPLlamaModel(mcfg, group=self.topo.cp_group)
        self.model = model.to(torch.bfloat16)
        self.model.inv_freq = make_inv_freq(mcfg.head_dim, mcfg.rope_theta,
                                            device=self.device)
        self.store = FlatParamStore(self.model, device=self.device)
        self.opt = FlatAdamW(self.store, lr=cfg.lr, betas=cfg.betas,
                             weight_decay=cfg.weight_decay,
                             clip_grad_norm=cfg.clip_grad_norm)
        # cp peers consume the SAME stream (each slices its own block)
        self.data = make_batches(cfg, self.device, rank=self.topo.dp_rank)
        self.step_count = 0

    def _sync_grads(self) -> None:
        if self.topo.world == 1:
            return
        fg = self.store.flat_grad
        if self.topo.dp_size > 1:
            fg.mul_(1.0 / self.topo.dp_size)
        dist.all_reduce(fg)                     # SUM over dp x cp

    def train_step(self):
        cfg = self.cfg
        if cfg.warmup_steps or cfg.lr_decay_steps:
            from ..optim import lr_at
            self.opt.lr = lr_at(self.opt.step_count, cfg.lr,
                                cfg.warmup_steps, cfg.lr_decay_steps,
                                cfg.min_lr)
        loss = None
        for _ in range(cfg.grad_accum):
            tokens, targets = next(self.data)
            loss = self.model(tokens, targets)
            (loss / cfg.grad_accum).backward()
        self._sync_grads()
        self.opt.step(grad_pre_scale=1.0)
        self.opt.zero_grad()
        self.step_count += 1
        return loss.detach()
