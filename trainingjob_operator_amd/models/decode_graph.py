"""hipGraph-captured one-token decode step for dense Llama serving.

The eager decode step issues ~650 kernels per token (projections, rope,
cache writes, attention einsums, norms) and the MI355X sits ~80% idle in
launch gaps (profiles/r02_serving_profile.md). This module captures the
whole step in one hipGraph and replays it per token. Everything the step
needs per token lives on the DEVICE so a replay takes no host input:

  * `cur`   — the token ids, copied in before each replay;
  * `pos_t` — the fill position as a device tensor: the RoPE angles and
    the attention mask are computed from it in-graph, the KV write is an
    `index_copy_` with it, and the graph increments it at the end, so
    consecutive replays decode consecutive positions;
  * attention runs over the cache's full `max_len` with an additive
    `-inf` mask beyond `pos_t` (fixed shapes are what make the step
    capturable; the masked tail costs bandwidth, not correctness).

Opt-in via AITJ_DECODE_GRAPH=1 (generate() checks it): hipGraph capture
of a large training step hung on ROCm 7.2 (ROADMAP.md), so the default
stays on the eager path and the graph path is validated explicitly under
a timeout in scripts/graphcheck.py rather than in the driver-run suite.

Reference parity note: the reference operator has no serving path at all
 — this is framework-side capability its users get on
top.
"""
from __future__ import annotations

import torch

from ..ops import decode_attention, decode_linear, fused_rmsnorm, native
from .generate import KVCache, _mlp_cached
from .llama import LlamaModel


def _step_static(model: LlamaModel, cur: torch.Tensor, cache: KVCache,
                 pos_t: torch.Tensor, arange: torch.Tensor) -> torch.Tensor:
    """One decode step with device-resident position: returns logits
    [B, V]. Every shape is independent of the position, so the whole
    call is stream-capturable."""
    import math
    cfg = model.cfg
    B = cur.shape[0]
    x = model.embed(cur)                                   # [B, 1, H]
    mask = None          # built lazily: only the einsum fallback needs it
    residual = None
    for li, blk in enumerate(model.blocks):
        attn = blk.attn
        normed, residual = fused_rmsnorm(x, blk.input_norm_weight,
                                         residual, cfg.norm_eps)
        qkv = decode_linear(normed, attn.qkv_proj.weight)
        # one fused launch: rope q -> contiguous buffer, rope k ->
        # cache[pos], copy v -> cache[pos] (vs 2 rope + 2 index_copy)
        lib = native.load(require=True)
        q = torch.empty(B, cfg.num_heads, cfg.head_dim,
                        dtype=torch.bfloat16, device=qkv.device)
        rc = lib.rope_cache(
            native.stream_ptr(), qkv.contiguous().data_ptr(),
            q.data_ptr(), cache.k[li].data_ptr(), cache.v[li].data_ptr(),
            model.inv_freq.data_ptr(), pos_t.data_ptr(),
            B, cfg.num_heads, cfg.num_kv_heads, cache.max_len,
            cfg.head_dim)
        native.check_rc(rc, "rope_cache", f"D={cfg.head_dim}")
        q = q[:, :, None, :]                               # [B, nh, 1, D]
        kk, vv = cache.k[li], cache.v[li]                  # full max_len
        scale = 1.0 / math.sqrt(cfg.head_dim)
        o = decode_attention(q, kk, vv, pos_t, scale)
        if o is None:                         # einsum fallback (odd GQA)
            if mask is None:
                mask = torch.where(
                    arange[None, None, None, :] > pos_t[0],
                    float("-inf"), 0.0)
            G = cfg.num_heads // cfg.num_kv_heads
            qg = q.reshape(B, cfg.num_kv_heads, G, cfg.head_dim)
            scores = torch.einsum("bkgd,bksd->bkgs", qg.float(),
                                  kk.float())
            scores = scores * scale + mask
            p = torch.softmax(scores, dim=-1)
            o = torch.einsum("bkgs,bksd->bkgd", p, vv.float())
        o = o.reshape(B, 1, cfg.num_heads * cfg.head_dim).to(x.dtype)
        attn_out = decode_linear(o, attn.o_proj.weight)
        normed, residual = fused_rmsnorm(attn_out,
                                         blk.post_attn_norm_weight,
                                         residual, cfg.norm_eps)
        # MoE blocks are capturable because the routed decode path keeps
        # routing on device (parallel/ep.py _decode_forward): topk ->
        # pointer-table GEMVs, fixed shapes, no host syncs
        x = _mlp_cached(blk.mlp, normed) if hasattr(blk, "mlp") \
            else blk.moe(normed)
    normed, _ = fused_rmsnorm(x, model.final_norm_weight, residual,
                              cfg.norm_eps)
    logits = decode_linear(normed, model.lm_head.weight)
    pos_t.add_(1)                                          # in-graph advance
    return logits.reshape(B, -1)


class GraphedDecoder:
    """Captures `_step_static` once and replays it per token.

    Usage (generate() drives this): prefill eagerly, then
        dec = GraphedDecoder(model, cache, batch)
        dec.prime()                     # warmup + capture at cache.len
        logits = dec.step(tokens)       # per decoded token
    The warmup steps write scratch K/V at positions >= cache.len; real
    replays overwrite those same positions, so the cache stays correct.
    """

    def __init__(self, model: LlamaModel, cache: KVCache, batch: int):
        dev = model.embed.weight.device
        self.model, self.cache = model, cache
        self.cur = torch.zeros(batch, dtype=torch.long, device=dev)
        self.pos_t = torch.zeros(1, dtype=torch.long, device=dev)
        self.arange = torch.arange(cache.max_len, device=dev)
        self.graph: torch.cuda.CUDAGraph | None = None
        self.logits: torch.Tensor | None = None

    def _run(self) -> torch.Tensor:
        return _step_static(self.model, self.cur[:, None], self.cache,
                            self.pos_t, self.arange)

    def prime(self) -> None:
        assert self.cache.len + 2 < self.cache.max_len, \
            "graph warmup needs 2 free cache slots"
        self.pos_t.fill_(self.cache.len)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):                 # warmup (eager, side stream)
                self._run()
        torch.cuda.current_stream().wait_stream(side)
        self.pos_t.fill_(self.cache.len)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self.logits = self._run()
        self.graph = g
        # capture records without executing, but reset defensively: the
        # first replay must decode at exactly cache.len
        self.pos_t.fill_(self.cache.len)

    def step(self, tokens: torch.Tensor) -> torch.Tensor:
        """tokens [B] -> logits [B, V]; advances the cache position."""
        self.cur.copy_(tokens)
        self.graph.replay()
        self.cache.len += 1
        return self.logits


class DecodeSession:
    """Persistent serving session: one KV cache + one captured graph
    reused across requests.

    The captured step is position-INDEPENDENT — everything positional is
    read from the device `pos_t` — so a single capture serves every
    request at this batch size: a new request just prefills into the
    same cache buffers (stale rows beyond the fill are masked, never
    read) and resets `pos_t`. Per-request graph capture was ~15% of a
    512-token generation and dominates short requests.
    """

    def __init__(self, model, batch: int, max_len: int):
        from .generate import KVCache
        self.model = model
        self.batch = batch
        self.max_len = max_len
        self.cache = KVCache(model.cfg, batch, max_len,
                             model.embed.weight.device,
                             model.embed.weight.dtype)
        self.dec: GraphedDecoder | None = None

    @torch.no_grad()
    def generate(self, tokens: torch.Tensor, max_new_tokens: int,
                 temperature: float = 0.0, top_k: int = 0,
                 eos_token=None, seed=None) -> torch.Tensor:
        from .generate import _forward_cached, _sample
        B, S0 = tokens.shape
        assert B == self.batch, (B, self.batch)
        assert S0 + max_new_tokens <= self.max_len, "session max_len"
        was_training = self.model.training
        self.model.eval()
        try:
            self.cache.len = 0                  # reuse buffers in place
            last = _forward_cached(self.model, tokens, self.cache)[:, -1]
            if self.dec is None:
                self.dec = GraphedDecoder(self.model, self.cache, B)
                self.dec.prime()                # captures ONCE
            else:
                self.dec.pos_t.fill_(self.cache.len)
            gen = None
            if seed is not None:
                gen = torch.Generator(device="cpu").manual_seed(seed)
            out = tokens
            done = torch.zeros(B, dtype=torch.bool, device=tokens.device)
            for _ in range(max_new_tokens):
                if temperature <= 0.0:
                    nxt = last.argmax(dim=-1)
                else:
                    nxt = _sample(last.cpu(), temperature, top_k,
                                  gen).to(tokens.device)
                if eos_token is not None:
                    nxt = torch.where(done, torch.full_like(nxt,
                                                            eos_token), nxt)
                    done = done | (nxt == eos_token)
                out = torch.cat([out, nxt[:, None]], dim=1)
                if eos_token is not None and bool(done.all()):
                    break
                last = self.dec.step(nxt)
            return out
        finally:
            if was_training:
                self.model.train()
