"""Checkpoint topology resharding: convert saved training streams between
parallel layouts — full (pure DP) <-> PP stage slices <-> TP shards <->
EP expert shards —
carrying parameters AND AdamW moments (p32/m/v transform identically to
their parameters, since the optimizer flats mirror the param flat layout).

The reference operator has no training state at all (zero
compute code); this is the MI355X stack's answer to "the elastic policy
changed the topology, resume anyway": DP resizes need nothing (flat state
is replicated — checkpoint.py), PP/TP changes run through here.

Vocab-parallel-head TP streams (lm_head.proj.weight shards) are not yet
convertible — the name-set assertions fail loudly rather than guessing.

Streams are the exact files Checkpointer writes (ckpt_stepNNNNNNNN.pt with
names/offsets/shapes metadata). Directory conventions match the launcher:
full -> DIR/, PP -> DIR/stage{i}/, TP -> DIR/tp{r}/, EP -> DIR/ep{r}/.
"""
from __future__ import annotations

import os
from typing import Dict, List

import torch

from ..models.config import CONFIGS, LlamaConfig
from ..parallel.flat import _aligned
from ..parallel.pp import partition_layers
from .checkpoint import CKPT_PREFIX, Checkpointer

FLATS = ("param", "p32", "m", "v")   # bf16 weights + fp32 moments


# -- stream <-> named -----------------------------------------------------

def load_stream(path: str) -> dict:
    return torch.load(path, map_location="cpu", weights_only=False)


def stream_to_named(state: dict) -> Dict[str, Dict[str, torch.Tensor]]:
    """One checkpoint file -> {param_name: {param, p32, m, v}} (views
    reshaped to the param shape)."""
    named = {}
    flats = {"param": state["flat_param"], "p32": state["opt"]["p32"],
             "m": state["opt"]["m"], "v": state["opt"]["v"]}
    for name in state["names"]:
        off, numel = state["offsets"][name]
        shape = tuple(state["shapes"][name])
        named[name] = {k: f[off:off + numel].view(shape)
                       for k, f in flats.items()}
    return named


def _store_order(model) -> List[str]:
    """FlatParamStore layout order = reversed named_parameters."""
    names = [n for n, _ in model.named_parameters()]
    names.reverse()
    return names


def _is_moe(cfg: LlamaConfig) -> bool:
    from ..models.moe_llama import MoELlamaConfig
    return isinstance(cfg, MoELlamaConfig)


def _model_order(cfg: LlamaConfig, kind: str) -> List[str]:
    """Flat layout order for a target model, built on the meta device
    (no allocation)."""
    with torch.device("meta"):
        if kind == "full" and _is_moe(cfg):
            from ..models.moe_llama import MoELlamaModel
            model = MoELlamaModel(cfg)
        elif kind == "full":
            from ..models.llama import LlamaModel
            model = LlamaModel(cfg)
        elif kind == "tp":
            assert not _is_moe(cfg), "TP of MoE models is not supported"
            from ..parallel.tp_llama import TPLlamaModel
            model = TPLlamaModel(cfg)
        else:
            raise ValueError(kind)
    return _store_order(model)


def _stage_order(cfg: LlamaConfig, stage: int, n_stages: int) -> List[str]:
    """Flat layout order for one pipeline stage (registration order
    differs from the full model's, so filtering the full order would
    misalign load_flat_param's raw flat copy)."""
    from ..parallel.pp import LlamaStage
    return _store_order(
        LlamaStage.from_config(cfg, stage, n_stages, device="meta"))


def named_to_stream(named: Dict[str, Dict[str, torch.Tensor]],
                    order: List[str], meta: dict) -> dict:
    """Pack named tensors back into flat buffers in the given name order
    (must be the target FlatParamStore order) -> a Checkpointer-loadable
    state dict."""
    offsets, shapes = {}, {}
    off = 0
    for n in order:
        t = named[n]["param"]
        offsets[n] = (off, t.numel())
        shapes[n] = tuple(t.shape)
        off += _aligned(t.numel())
    total = _aligned(off)
    flats = {"param": torch.zeros(total, dtype=torch.bfloat16),
             "p32": torch.zeros(total, dtype=torch.float32),
             "m": torch.zeros(total, dtype=torch.float32),
             "v": torch.zeros(total, dtype=torch.float32)}
    for n in order:
        o, numel = offsets[n]
        for k in FLATS:
            flats[k][o:o + numel].copy_(
                named[n][k].reshape(-1).to(flats[k].dtype))
    return {
        "step": meta["step"],
        "flat_param": flats["param"],
        "opt": {"p32": flats["p32"], "m": flats["m"], "v": flats["v"],
                "step": meta["opt"]["step"]},
        "train_config": dict(meta.get("train_config") or {}),
        "names": list(order),
        "offsets": offsets,
        "shapes": shapes,
        "time": meta.get("time", 0.0),
    }


# -- PP: stage slices <-> full --------------------------------------------

def merge_pp(stage_named: List[Dict], cfg: LlamaConfig) -> Dict:
    """Per-stage named dicts (stage order) -> full-model named dict:
    local block indices remap to global layer numbers."""
    n_stages = len(stage_named)
    parts = partition_layers(cfg.num_layers, n_stages)
    full = {}
    for s, named in enumerate(stage_named):
        globs = list(parts[s])
        for name, t in named.items():
            if name.startswith("blocks."):
                loc, tail = name[len("blocks."):].split(".", 1)
                full[f"blocks.{globs[int(loc)]}.{tail}"] = t
            else:
                full[name] = t       # embed / final_norm / lm_head
    return full


def split_pp(full_named: Dict, cfg: LlamaConfig,
             n_stages: int) -> List[Dict]:
    """Inverse of merge_pp."""
    parts = partition_layers(cfg.num_layers, n_stages)
    out = []
    for s in range(n_stages):
        named = {}
        for loc, glob in enumerate(parts[s]):
            for name, t in full_named.items():
                pre = f"blocks.{glob}."
                if name.startswith(pre):
                    named[f"blocks.{loc}.{name[len(pre):]}"] = t
        if s == 0:
            named["embed.weight"] = full_named["embed.weight"]
        if s == n_stages - 1:
            named["final_norm_weight"] = full_named["final_norm_weight"]
            named["lm_head.weight"] = full_named["lm_head.weight"]
        out.append(named)
    return out


# -- TP: shards <-> full ---------------------------------------------------

def _cat(shards: List[Dict], name: str, dim: int) -> Dict:
    return {k: torch.cat([s[name][k] for s in shards], dim=dim)
            for k in FLATS}


def merge_tp(shard_named: List[Dict], cfg: LlamaConfig) -> Dict:
    """Per-tp-rank named dicts (rank order) -> full-model named dict.
    Mirrors TPLlamaModel.shard_from_full exactly: q/k/v/gate/up are
    contiguous ROW blocks, o/down contiguous COLUMN blocks, norms/embed/
    lm_head replicated; the full model re-fuses qkv and gate_up."""
    full = {}
    r0 = shard_named[0]
    for name in r0:
        if ".attn.q_proj." in name:
            base = name.split(".attn.")[0]
            q = _cat(shard_named, f"{base}.attn.q_proj.weight", 0)
            k = _cat(shard_named, f"{base}.attn.k_proj.weight", 0)
            v = _cat(shard_named, f"{base}.attn.v_proj.weight", 0)
            full[f"{base}.attn.qkv_proj.weight"] = {
                kk: torch.cat([q[kk], k[kk], v[kk]], dim=0) for kk in FLATS}
        elif ".attn.k_proj." in name or ".attn.v_proj." in name:
            continue
        elif ".mlp.gate_proj." in name:
            base = name.split(".mlp.")[0]
            g = _cat(shard_named, f"{base}.mlp.gate_proj.weight", 0)
            u = _cat(shard_named, f"{base}.mlp.up_proj.weight", 0)
            full[f"{base}.mlp.gate_up_proj.weight"] = {
                kk: torch.cat([g[kk], u[kk]], dim=0) for kk in FLATS}
        elif ".mlp.up_proj." in name:
            continue
        elif ".attn.o_proj." in name or ".mlp.down_proj." in name:
            full[name] = _cat(shard_named, name, 1)
        else:
            full[name] = r0[name]    # replicated
    return full


def split_tp(full_named: Dict, cfg: LlamaConfig, tp: int) -> List[Dict]:
    """Inverse of merge_tp (the exact shard_from_full slicing)."""
    q_size = cfg.num_heads * cfg.head_dim
    kv = cfg.num_kv_heads * cfg.head_dim

    def rows(entry, r, n):
        return {k: t.chunk(n, dim=0)[r] for k, t in entry.items()}

    def cols(entry, r, n):
        return {k: t.chunk(n, dim=1)[r] for k, t in entry.items()}

    out = []
    for r in range(tp):
        named = {}
        for name, entry in full_named.items():
            if name.endswith(".attn.qkv_proj.weight"):
                base = name[:-len(".qkv_proj.weight")]
                split = {k: t.split([q_size, kv, kv], dim=0)
                         for k, t in entry.items()}
                for i, part in enumerate(("q", "k", "v")):
                    e = {k: split[k][i] for k in FLATS}
                    named[f"{base}.{part}_proj.weight"] = rows(e, r, tp)
            elif name.endswith(".mlp.gate_up_proj.weight"):
                base = name[:-len(".gate_up_proj.weight")]
                split = {k: t.chunk(2, dim=0) for k, t in entry.items()}
                for i, part in enumerate(("gate", "up")):
                    e = {k: split[k][i] for k in FLATS}
                    named[f"{base}.{part}_proj.weight"] = rows(e, r, tp)
            elif name.endswith(".attn.o_proj.weight") or \
                    name.endswith(".mlp.down_proj.weight"):
                named[name] = cols(entry, r, tp)
            else:
                named[name] = entry  # replicated
        out.append(named)
    return out


# -- EP: expert shards <-> full -------------------------------------------

def _expert_gid(name: str) -> int:
    return int(name.split(".experts.")[1].split(".")[0])


def _rename_expert(name: str, new_id: int) -> str:
    pre, rest = name.split(".experts.")
    _, tail = rest.split(".", 1)
    return f"{pre}.experts.{new_id}.{tail}"


def merge_ep(shard_named: List[Dict], cfg) -> Dict:
    """Per-ep-rank named dicts (rank order) -> full-model named dict:
    local expert ids remap to global (rank r owns the contiguous block
    [r*E/ep, (r+1)*E/ep) — parallel/ep.py MoEMLP); non-expert params are
    replicated, taken from rank 0."""
    ep = len(shard_named)
    per = cfg.n_experts // ep
    full = {}
    for name, t in shard_named[0].items():
        if ".experts." not in name:
            full[name] = t
    for r, named in enumerate(shard_named):
        for name, t in named.items():
            if ".experts." in name:
                full[_rename_expert(name, r * per + _expert_gid(name))] = t
    return full


def split_ep(full_named: Dict, cfg, ep: int) -> List[Dict]:
    """Inverse of merge_ep."""
    per = cfg.n_experts // ep
    out = []
    for r in range(ep):
        named = {}
        for name, t in full_named.items():
            if ".experts." not in name:
                named[name] = t
                continue
            gid = _expert_gid(name)
            if r * per <= gid < (r + 1) * per:
                named[_rename_expert(name, gid - r * per)] = t
        out.append(named)
    return out


# -- ZeRO-1: per-rank moment shards <-> full --------------------------------

def zero_chunk_bounds(total: int, dp: int) -> List:
    """The exact shard math training.py uses: the reduce-scatter layout
    (equal ALIGN-floored chunks, tail on the last rank) when it divides,
    else the legacy aligned ceil-div chunks."""
    from ..parallel.flat import ALIGN
    chunk_lo = (total // dp) // ALIGN * ALIGN
    if chunk_lo > 0:
        return [(i * chunk_lo,
                 (i + 1) * chunk_lo if i < dp - 1 else total)
                for i in range(dp)]
    chunk = _aligned(-(-total // dp))
    return [(min(i * chunk, total), min((i + 1) * chunk, total))
            for i in range(dp)]


def merge_zero(states: List[dict]) -> dict:
    """Per-rank ZeRO-1 streams (rank order) -> one full-layout state.
    Every stream carries the FULL bf16 flat_param; p32/m/v are that
    rank's chunk and concatenate back to full."""
    full = dict(states[0])
    total = states[0]["flat_param"].numel()
    # bounds from the ACTUAL shard lengths (works for both the RS layout
    # and the legacy ceil-div layout — checkpoints carry the truth)
    lens = [st["opt"]["p32"].numel() for st in states]
    offs = [0]
    for n in lens:
        offs.append(offs[-1] + n)
    assert offs[-1] == total, f"zero shards sum {offs[-1]} != {total}"
    bounds = list(zip(offs[:-1], offs[1:]))
    opt = {"step": states[0]["opt"]["step"]}
    for key in ("p32", "m", "v"):
        buf = torch.zeros(total, dtype=torch.float32)
        for st, (s, e) in zip(states, bounds):
            shard = st["opt"][key]
            assert shard.numel() == e - s,                 f"zero shard {key} length {shard.numel()} != {e - s}"
            buf[s:e].copy_(shard)
        opt[key] = buf
    full["opt"] = opt
    return full


def split_zero(state: dict, dp: int) -> List[dict]:
    """Inverse of merge_zero: slice the full moments into dp chunks."""
    total = state["flat_param"].numel()
    out = []
    for s, e in zero_chunk_bounds(total, dp):
        st = dict(state)
        st["opt"] = {"step": state["opt"]["step"],
                     "p32": state["opt"]["p32"][s:e].clone(),
                     "m": state["opt"]["m"][s:e].clone(),
                     "v": state["opt"]["v"][s:e].clone()}
        out.append(st)
    return out


# -- directory-level conversion -------------------------------------------

def _latest_state(directory: str) -> dict:
    ck = Checkpointer(directory)
    path = ck.latest()
    if path is None:
        raise FileNotFoundError(f"no {CKPT_PREFIX}* in {directory}")
    return load_stream(path)


def _write_state(directory: str, state: dict) -> str:
    os.makedirs(directory, exist_ok=True)
    path = os.path.join(directory,
                        f"{CKPT_PREFIX}{state['step']:08d}.pt")
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)
    return path


def _tp_order(cfg: LlamaConfig, shard_named: Dict) -> List[str]:
    order = _model_order(cfg, "tp")
    assert set(order) == set(shard_named), \
        "tp stream names do not match the TP model"
    return order


def reshard(model: str, in_dir: str, out_dir: str,
            src: str, dst: str) -> List[str]:
    """Convert the latest checkpoint in `in_dir` (layout `src`) to layout
    `dst` under `out_dir`. Layouts: "full", "pp=N", "tp=N", "ep=N",
    "zero=N" (ZeRO-1 moment shards). Returns the written paths."""
    cfg = CONFIGS[model]

    def parse(lay):
        if lay == "full":
            return ("full", 1)
        kind, n = lay.split("=")
        return (kind, int(n))

    skind, sn = parse(src)
    dkind, dn = parse(dst)

    # ZeRO-1 conversions operate on raw streams (the flat layout is the
    # full model's already; only the optimizer moments are sharded)
    if skind == "zero":
        states = [_latest_state(os.path.join(in_dir, f"zero{r}"))
                  for r in range(sn)]
        merged = merge_zero(states)
        if dkind == "full":
            return [_write_state(out_dir, merged)]
        if dkind == "zero":
            return [_write_state(os.path.join(out_dir, f"zero{r}"), st)
                    for r, st in enumerate(split_zero(merged, dn))]
        st = merged
        full, meta = stream_to_named(st), st
    elif skind == "full" and dkind == "zero":
        st = _latest_state(in_dir)
        return [_write_state(os.path.join(out_dir, f"zero{r}"), sh)
                for r, sh in enumerate(split_zero(st, dn))]
    elif skind == "full":
        st = _latest_state(in_dir)
        full, meta = stream_to_named(st), st
    elif skind == "pp":
        states = [_latest_state(os.path.join(in_dir, f"stage{i}"))
                  for i in range(sn)]
        full = merge_pp([stream_to_named(s) for s in states], cfg)
        meta = states[0]
    elif skind == "tp":
        states = [_latest_state(os.path.join(in_dir, f"tp{r}"))
                  for r in range(sn)]
        full = merge_tp([stream_to_named(s) for s in states], cfg)
        meta = states[0]
    elif skind == "ep":
        states = [_latest_state(os.path.join(in_dir, f"ep{r}"))
                  for r in range(sn)]
        full = merge_ep([stream_to_named(s) for s in states], cfg)
        meta = states[0]
    else:
        raise ValueError(src)

    # write target layout
    written = []
    if dkind == "full":
        full_order = _model_order(cfg, "full")
        assert set(full_order) == set(full)
        written.append(_write_state(
            out_dir, named_to_stream(full, full_order, meta)))
    elif dkind == "pp":
        for i, named in enumerate(split_pp(full, cfg, dn)):
            order = _stage_order(cfg, i, dn)
            assert set(order) == set(named)
            written.append(_write_state(
                os.path.join(out_dir, f"stage{i}"),
                named_to_stream(named, order, meta)))
    elif dkind == "tp":
        shards = split_tp(full, cfg, dn)
        order = _tp_order(cfg, shards[0])
        for r, named in enumerate(shards):
            written.append(_write_state(
                os.path.join(out_dir, f"tp{r}"),
                named_to_stream(named, order, meta)))
    elif dkind == "ep":
        full_order = _model_order(cfg, "full")
        per = cfg.n_experts // dn
        for r, named in enumerate(split_ep(full, cfg, dn)):
            base = r * per
            order = []
            for n in full_order:
                if ".experts." not in n:
                    order.append(n)
                elif base <= _expert_gid(n) < base + per:
                    order.append(_rename_expert(n, _expert_gid(n) - base))
            assert set(order) == set(named)
            written.append(_write_state(
                os.path.join(out_dir, f"ep{r}"),
                named_to_stream(named, order, meta)))
    else:
        raise ValueError(dst)
    return written
