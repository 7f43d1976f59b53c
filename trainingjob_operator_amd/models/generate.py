"""KV-cached autoregressive generation for the Llama family (dense and
MoE — MoE blocks route each decoded token through their experts).

Training is the flagship workload, but a framework its users can switch
to needs an eval/inference path: greedy or temperature/top-k sampling
with a per-layer KV cache (prefill once, then one-token decode steps).

The decode path reuses the training modules' weights and runs the
hand-written decode stack: wave-per-row GEMV projections
(ops.decode_linear / decode_swiglu), fused flash-decoding attention
over the cache (ops.decode_attention, device-resident length), native
explicit-position RoPE, and — opt-in via AITJ_DECODE_GRAPH=1 — the
hipGraph-captured step in models/decode_graph.py. Prompt-lookup
speculative decoding (generate_lookup) emits greedy-identical output
with fewer forwards. See docs/SERVING.md and
profiles/r02_serving_profile.md for the measured arc.
"""
from __future__ import annotations

import os
from typing import List, Optional

import torch

from ..ops import decode_linear, fused_rmsnorm, swiglu_packed
from .config import LlamaConfig
from .llama import LlamaModel, _sdpa


def build_inference_model(cfg, device):
    """Build any registered config (dense LlamaModel or MoELlamaModel)
    in inference form: bf16 weights, fp32 rope frequencies, eval mode.
    The training-side build_model() deliberately rejects MoE configs
    (those go through EPTrainer); serving and genbench accept both."""
    from ..ops import make_inv_freq
    from .moe_llama import MoELlamaConfig, MoELlamaModel
    if isinstance(cfg, MoELlamaConfig):
        with torch.device(device):
            model = MoELlamaModel(cfg)
        model = model.to(torch.bfloat16)
    else:
        from ..training import build_model
        return build_model(cfg, device).eval()
    model.inv_freq = make_inv_freq(cfg.head_dim, cfg.rope_theta,
                                   device=device)
    return model.eval()


class KVCache:
    """Per-layer [B, n_kv, max_len, D] key/value buffers + fill length."""

    def __init__(self, cfg: LlamaConfig, batch: int, max_len: int,
                 device, dtype):
        shape = (batch, cfg.num_kv_heads, max_len, cfg.head_dim)
        self.k: List[torch.Tensor] = [
            torch.zeros(shape, device=device, dtype=dtype)
            for _ in range(cfg.num_layers)]
        self.v: List[torch.Tensor] = [
            torch.zeros(shape, device=device, dtype=dtype)
            for _ in range(cfg.num_layers)]
        self.max_len = max_len
        self.len = 0


def _rope_tables(inv_freq: torch.Tensor, pos0: int, S: int, device):
    """cos/sin for positions pos0..pos0+S-1, computed ONCE per step (the
    round-1 decode rebuilt them per LAYER: ~10 extra launches x n_layers
    per decoded token)."""
    pos = (torch.arange(S, device=device) + pos0).float()
    ang = pos[:, None] * inv_freq[None, :].float()       # [S, half]
    return ang.cos()[None, :, None, :], ang.sin()[None, :, None, :]


def _rope_at(x: torch.Tensor, cos: torch.Tensor,
             sin: torch.Tensor) -> torch.Tensor:
    """Neox half-rotation with precomputed tables. x: [B, S, n_heads, D]."""
    half = x.shape[-1] // 2
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    return torch.cat([x1 * cos - x2 * sin,
                      x1 * sin + x2 * cos], dim=-1).to(x.dtype)


def _rope_positions(x: torch.Tensor, inv_freq: torch.Tensor, S: int,
                    pos0: int) -> torch.Tensor:
    """RoPE at positions pos0 + (flat % S): the native kernel on GPU
    (one launch vs ~8 torch ops), torch fallback on CPU.
    x: [B, S, n_heads, D] bf16-contiguous for the kernel path."""
    if x.is_cuda and x.dtype == torch.bfloat16 and x.shape[-1] % 16 == 0:
        from ..ops import native
        lib = native.load(require=True)
        xt = x.reshape(-1, x.shape[-2], x.shape[-1]).contiguous()
        out = torch.empty_like(xt)
        rc = lib.rope_at(native.stream_ptr(), xt.data_ptr(),
                         out.data_ptr(), inv_freq.data_ptr(),
                         xt.shape[0], xt.shape[1], S, xt.shape[2], 1.0,
                         pos0)
        native.check_rc(rc, "rope_at", f"D={x.shape[-1]}")
        return out.reshape(x.shape)
    return None


def _decode_attn(q, kk, vv, n_kv: int, ck=None, cv=None, pos_t=None):
    """One-token attention over the cache WITHOUT expanding KV heads.
    With the full cache buffers + a device position tensor, the fused
    flash-decoding kernel (ops.decode_attention, 2 launches) runs;
    otherwise a grouped einsum on views (the library GQA path
    repeat_interleaves the cache -- hundreds of MB per layer at batch
    8). q: [B, H, 1, D]."""
    import math
    B, H, _, D = q.shape
    if ck is not None and pos_t is not None and q.is_cuda:
        from ..ops import decode_attention
        o = decode_attention(q, ck, cv, pos_t, 1.0 / math.sqrt(D))
        if o is not None:
            return o
    G = H // n_kv
    qg = q.reshape(B, n_kv, G, D)
    scores = torch.einsum("bkgd,bksd->bkgs", qg.float(), kk.float())
    scores = scores / math.sqrt(D)
    p = torch.softmax(scores, dim=-1)
    o = torch.einsum("bkgs,bksd->bkgd", p, vv.float())
    return o.reshape(B, H, 1, D).to(q.dtype)


def _attn_cached(attn, x: torch.Tensor, cos: torch.Tensor,
                 sin: torch.Tensor, ck: torch.Tensor, cv: torch.Tensor,
                 pos0: int) -> torch.Tensor:
    cfg = attn.cfg
    B, S, _ = x.shape
    qkv = decode_linear(x, attn.qkv_proj.weight)
    q, k, v = qkv.split([attn.q_size, attn.kv_size, attn.kv_size], dim=-1)
    q = q.reshape(B, S, cfg.num_heads, cfg.head_dim)
    k = k.reshape(B, S, cfg.num_kv_heads, cfg.head_dim)
    v = v.reshape(B, S, cfg.num_kv_heads, cfg.head_dim)
    qr = _rope_positions(q, attn._inv_freq_dec, S, pos0) \
        if hasattr(attn, "_inv_freq_dec") else None
    if qr is not None:
        q = qr.transpose(1, 2)
        k = _rope_positions(k, attn._inv_freq_dec, S,
                            pos0).transpose(1, 2)
    else:
        q = _rope_at(q, cos, sin).transpose(1, 2)        # [B, nh, S, D]
        k = _rope_at(k, cos, sin).transpose(1, 2)
    v = v.transpose(1, 2)
    ck[:, :, pos0:pos0 + S] = k
    cv[:, :, pos0:pos0 + S] = v
    kk = ck[:, :, :pos0 + S]
    vv = cv[:, :, :pos0 + S]
    gqa = cfg.num_heads != cfg.num_kv_heads
    if S > 1:
        if pos0 == 0:
            # prefill: causal among the new tokens
            o = _sdpa(q, kk, vv, enable_gqa=gqa)
        else:
            # multi-token continuation mid-cache (speculative verify):
            # is_causal aligns TOP-LEFT in torch SDPA, which would hide
            # the cache — build the offset (bottom-right) causal mask
            mask = torch.ones(S, pos0 + S, dtype=torch.bool,
                              device=q.device).tril(pos0)
            o = torch.nn.functional.scaled_dot_product_attention(
                q, kk, vv, attn_mask=mask, enable_gqa=gqa)
    else:
        o = _decode_attn(q, kk, vv, cfg.num_kv_heads, ck, cv,
                         getattr(attn, "_pos_dec", None))
    o = o.transpose(1, 2).reshape(B, S, cfg.num_heads * cfg.head_dim)
    return decode_linear(o, attn.o_proj.weight)



def _mlp_cached(mlp, x: torch.Tensor) -> torch.Tensor:
    """SwiGLU MLP through the decode GEMV path: fused GEMV+SwiGLU when
    supported, else decode_linear + swiglu (both fall back to the
    library GEMM above 8 rows, so prefill takes the normal path)."""
    from ..ops import decode_swiglu
    h = decode_swiglu(x, mlp.gate_up_proj.weight)
    if h is None:
        gu = decode_linear(x, mlp.gate_up_proj.weight)
        h = swiglu_packed(gu.contiguous())
    return decode_linear(h, mlp.down_proj.weight)


@torch.no_grad()
def _forward_cached(model: LlamaModel, tokens: torch.Tensor,
                    cache: KVCache) -> torch.Tensor:
    """Run tokens through the model appending to the cache; returns
    logits [B, S, V]."""
    cfg = model.cfg
    pos0 = cache.len
    x = model.embed(tokens)
    use_kernel = tokens.is_cuda and model.inv_freq.dtype == torch.float32
    cos = sin = None
    if not use_kernel:
        cos, sin = _rope_tables(model.inv_freq, pos0, tokens.shape[1],
                                tokens.device)
    residual = None
    pos_dec = None
    if tokens.shape[1] == 1 and tokens.is_cuda:
        # device-side fill position for the fused decode-attention kernel
        pos_dec = torch.full((1,), pos0, dtype=torch.int64,
                             device=tokens.device)
    for li, blk in enumerate(model.blocks):
        if use_kernel:
            blk.attn._inv_freq_dec = model.inv_freq
        elif hasattr(blk.attn, "_inv_freq_dec"):
            del blk.attn._inv_freq_dec
        blk.attn._pos_dec = pos_dec
        normed, residual = fused_rmsnorm(x, blk.input_norm_weight,
                                         residual, cfg.norm_eps)
        attn_out = _attn_cached(blk.attn, normed, cos, sin,
                                cache.k[li], cache.v[li], pos0)
        normed, residual = fused_rmsnorm(attn_out,
                                         blk.post_attn_norm_weight,
                                         residual, cfg.norm_eps)
        # dense blocks carry .mlp; MoE blocks carry .moe (routing works
        # per token, so single-token decode steps route normally)
        x = _mlp_cached(blk.mlp, normed) if hasattr(blk, "mlp") \
            else blk.moe(normed)
    cache.len = pos0 + tokens.shape[1]
    normed, _ = fused_rmsnorm(x, model.final_norm_weight, residual,
                              cfg.norm_eps)
    return decode_linear(normed, model.lm_head.weight)


def _sample(logits: torch.Tensor, temperature: float, top_k: int,
            generator: Optional[torch.Generator]) -> torch.Tensor:
    """logits [B, V] -> next tokens [B]."""
    if temperature <= 0.0:
        return logits.argmax(dim=-1)
    logits = logits.float() / temperature
    if top_k > 0 and top_k < logits.shape[-1]:
        kth = logits.topk(top_k, dim=-1).values[:, -1:]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    probs = torch.softmax(logits, dim=-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)


@torch.no_grad()
def generate(model: LlamaModel, tokens: torch.Tensor,
             max_new_tokens: int, temperature: float = 0.0,
             top_k: int = 0, eos_token: Optional[int] = None,
             seed: Optional[int] = None) -> torch.Tensor:
    """Autoregressive generation: prefill the prompt once, then KV-cached
    one-token decode steps. temperature 0 = greedy. Returns
    [B, prompt + generated]."""
    was_training = model.training
    model.eval()
    try:
        B, S0 = tokens.shape
        cache = KVCache(model.cfg, B, S0 + max_new_tokens,
                        tokens.device, model.embed.weight.dtype)
        gen = None
        if seed is not None:
            gen = torch.Generator(device="cpu").manual_seed(seed)
        last = _forward_cached(model, tokens, cache)[:, -1]
        # opt-in hipGraph decode (see decode_graph.py for why not default)
        dec = None
        def _moe_graphable(moe):
            # must guarantee the ROUTED (sync-free) decode path engages,
            # or capture would hit the grouped path's D2H slice sizes
            ex = moe.experts[0]
            return (moe.group is None and moe.tp_group is None
                    and B * moe.top_k <= 16
                    and type(ex).__name__ == "Expert"
                    and ex.gate_proj.weight.shape[1] % 512 == 0
                    and ex.gate_proj.weight.shape[0] % 512 == 0)

        graph_ok = all(hasattr(b, "mlp") or _moe_graphable(b.moe)
                       for b in model.blocks)
        if (os.environ.get("AITJ_DECODE_GRAPH") == "1" and tokens.is_cuda
                and max_new_tokens > 2 and graph_ok):
            from .decode_graph import GraphedDecoder
            dec = GraphedDecoder(model, cache, B)
            dec.prime()
        out = tokens
        done = torch.zeros(B, dtype=torch.bool, device=tokens.device)
        for _ in range(max_new_tokens):
            if temperature <= 0.0:
                nxt = last.argmax(dim=-1)            # greedy: on device,
            else:                                    # no per-token sync
                nxt = _sample(last.cpu(), temperature, top_k,
                              gen).to(tokens.device)
            if eos_token is not None:
                nxt = torch.where(done, torch.full_like(nxt, eos_token),
                                  nxt)
                done = done | (nxt == eos_token)
            out = torch.cat([out, nxt[:, None]], dim=1)
            if eos_token is not None and bool(done.all()):
                break
            last = dec.step(nxt) if dec is not None else \
                _forward_cached(model, nxt[:, None], cache)[:, -1]
        return out
    finally:
        if was_training:
            model.train()


def _ngram_proposal(ctx: list, k: int) -> list:
    """Prompt-lookup draft: find the latest earlier occurrence of the
    current 3-gram (then 2-gram) suffix and propose the k tokens that
    followed it. Free speculation — no draft model — that pays off on
    repetitive text (code, retrieval contexts) and costs one batched
    verify forward otherwise."""
    L = len(ctx)
    for n in (3, 2):
        if L < n + 1:
            continue
        tail = ctx[L - n:]
        # search latest match of the n-gram strictly before the suffix
        for i in range(L - n - 1, -1, -1):
            if ctx[i:i + n] == tail:
                prop = ctx[i + n:i + n + k]
                if prop:
                    return prop
    return []


@torch.no_grad()
def generate_lookup(model, tokens: torch.Tensor, max_new_tokens: int,
                    lookup_k: int = 8,
                    eos_token: Optional[int] = None) -> torch.Tensor:
    """Greedy generation with prompt-lookup speculative decoding
    (batch 1): propose up to lookup_k continuation tokens from an
    earlier occurrence of the current n-gram, verify them in ONE
    cached forward, accept the matching prefix plus the bonus token
    from the first mismatch, and roll the cache back over the rest.
    Output is IDENTICAL to plain greedy generate() — speculation only
    changes how many forwards it takes (tests/test_generate.py)."""
    assert tokens.shape[0] == 1, "prompt-lookup decode is batch-1"
    if max_new_tokens <= 0:
        return tokens
    was_training = model.training
    model.eval()
    try:
        B, S0 = tokens.shape
        cache = KVCache(model.cfg, B, S0 + max_new_tokens + lookup_k + 1,
                        tokens.device, model.embed.weight.dtype)
        logits = _forward_cached(model, tokens, cache)
        cur = int(logits[0, -1].argmax())
        ctx = tokens[0].tolist() + [cur]
        produced = 1
        while produced < max_new_tokens and                 (eos_token is None or cur != eos_token):
            room = max_new_tokens - produced
            prop = _ngram_proposal(ctx, min(lookup_k, room)) if room > 1                 else []
            step = torch.tensor([[cur] + prop], device=tokens.device)
            pos0 = cache.len
            logits = _forward_cached(model, step, cache)
            nxt = logits[0].argmax(dim=-1).tolist()   # true tokens
            accepted = 0
            for j, pj in enumerate(prop):
                if pj == nxt[j] and accepted + 1 < room:
                    accepted += 1
                else:
                    break
            emitted = nxt[:accepted + 1]              # accepted + bonus
            # roll back cache rows written for rejected draft tokens
            cache.len = pos0 + 1 + accepted
            ctx.extend(emitted)
            produced += len(emitted)
            cur = emitted[-1]
            if eos_token is not None and eos_token in emitted:
                ctx = ctx[:len(ctx) - len(emitted)
                          + emitted.index(eos_token) + 1]
                break
        return torch.tensor([ctx], device=tokens.device)
    finally:
        if was_training:
            model.train()
