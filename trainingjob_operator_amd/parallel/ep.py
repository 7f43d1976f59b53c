"""Expert parallelism: a dropless top-k MoE MLP with token dispatch over
all-to-all.

On RCCL the dispatch is ONE all_to_all_single each way — the collective the
briefing's xGMI topology favors (7 direct p2p links per MI355X; all-to-all
rides every link instead of serializing on a ring). Gloo (CPU tests) has no
alltoall, so a pairwise isend/irecv emulation with identical semantics
backs the same autograd seam.

Experts are sharded over the EP group (contiguous blocks: rank r owns
experts [r*E/ep, (r+1)*E/ep)); routing is exact/dropless — variable split
sizes are exchanged first, no capacity factor, no token dropping.
gloo-verified against a single-process MoE in tests/test_ep_gloo.py.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ..ops import swiglu
from .tp import _group_size


def _backend_has_alltoall(group) -> bool:
    if not dist.is_initialized():
        return False
    return dist.get_backend(group) in ("nccl",)


def _a2a_raw(inp: torch.Tensor, in_splits: List[int], out_splits: List[int],
             group) -> torch.Tensor:
    """all_to_all_single with uneven splits; P2P emulation where the
    backend lacks alltoall (gloo)."""
    n = _group_size(group)
    out = torch.empty(sum(out_splits), *inp.shape[1:], dtype=inp.dtype,
                      device=inp.device)
    if n == 1:
        out.copy_(inp)
        return out
    if _backend_has_alltoall(group):
        dist.all_to_all_single(out, inp.contiguous(),
                               output_split_sizes=out_splits,
                               input_split_sizes=in_splits, group=group)
        return out
    rank = dist.get_rank(group)

    def _glob(r: int) -> int:
        # p2p dst/src are GLOBAL ranks even when a group is passed
        return dist.get_global_rank(group, r) if group is not None else r

    in_chunks = list(inp.split(in_splits, dim=0))
    out_chunks = list(out.split(out_splits, dim=0))
    reqs = []
    for peer in range(n):
        if peer == rank:
            out_chunks[peer].copy_(in_chunks[peer])
            continue
        if in_splits[peer] > 0:
            reqs.append(dist.isend(in_chunks[peer].contiguous(),
                                   dst=_glob(peer), group=group))
        if out_splits[peer] > 0:
            reqs.append(dist.irecv(out_chunks[peer], src=_glob(peer),
                                   group=group))
    for r in reqs:
        r.wait()
    return out


class _AllToAll(torch.autograd.Function):
    """Differentiable token exchange: backward is the reverse exchange."""

    @staticmethod
    def forward(ctx, x, in_splits, out_splits, group):
        ctx.in_splits = in_splits
        ctx.out_splits = out_splits
        ctx.group = group
        return _a2a_raw(x, in_splits, out_splits, group)

    @staticmethod
    def backward(ctx, g):
        return (_a2a_raw(g.contiguous(), ctx.out_splits, ctx.in_splits,
                         ctx.group), None, None, None)


class Expert(nn.Module):
    """One SwiGLU expert (same shape family as the dense MLP)."""

    def __init__(self, hidden: int, ff: int):
        super().__init__()
        self.gate_proj = nn.Linear(hidden, ff, bias=False)
        self.up_proj = nn.Linear(hidden, ff, bias=False)
        self.down_proj = nn.Linear(ff, hidden, bias=False)

    def forward(self, x):
        if not torch.is_grad_enabled() and x.is_cuda and x.shape[0] <= 8:
            # decode-sized slices (<= batch*top_k rows) stream the expert
            # weights through the wave-per-row GEMV instead of hipBLASLt
            # (decode_linear falls back itself on unsupported shapes)
            from ..ops import decode_linear
            g = decode_linear(x, self.gate_proj.weight)
            u = decode_linear(x, self.up_proj.weight)
            return decode_linear(swiglu(g.contiguous(), u.contiguous()),
                                 self.down_proj.weight)
        return self.down_proj(swiglu(self.gate_proj(x).contiguous(),
                                     self.up_proj(x).contiguous()))


class TPExpert(nn.Module):
    """One SwiGLU expert with its weights TENSOR-sharded over the TP group
    (EP x TP composition: experts split across EP ranks, each expert's
    matrices split across TP ranks). Input tokens arrive replicated on
    every tp peer (each peer ran its own ep-plane all-to-all on identical
    activations); copy_input=False because the single f-op at the MoE
    entry already sums the partial input grads — a per-expert f would
    double-count. The Row all-reduce re-assembles full outputs."""

    def __init__(self, hidden: int, ff: int, tp_group=None):
        super().__init__()
        from .tp import ColumnParallelLinear, RowParallelLinear
        self.gate_proj = ColumnParallelLinear(hidden, ff, tp_group,
                                              copy_input=False)
        self.up_proj = ColumnParallelLinear(hidden, ff, tp_group,
                                            copy_input=False)
        self.down_proj = RowParallelLinear(ff, hidden, tp_group,
                                           reduce_output=True)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x).contiguous(),
                                     self.up_proj(x).contiguous()))


class MoEMLP(nn.Module):
    """Dropless top-k mixture-of-experts MLP, experts sharded over the EP
    group. With ep_size == 1 it is a plain (single-process) MoE — the
    reference the EP tests compare against. With a tp_group the experts
    are additionally tensor-sharded (TPExpert)."""

    def __init__(self, hidden: int, ff: int, n_experts: int, top_k: int = 2,
                 group=None, tp_group=None):
        super().__init__()
        self.group = group
        self.tp_group = tp_group
        self.n_experts = n_experts
        self.top_k = top_k
        ep = _group_size(group)
        assert n_experts % ep == 0, (n_experts, ep)
        self.experts_per_rank = n_experts // ep
        self.router = nn.Linear(hidden, n_experts, bias=False)
        if tp_group is not None and _group_size(tp_group) > 1:
            self.experts = nn.ModuleList(
                TPExpert(hidden, ff, tp_group)
                for _ in range(self.experts_per_rank))
        else:
            self.experts = nn.ModuleList(
                Expert(hidden, ff) for _ in range(self.experts_per_rank))

    def _expert_owner_splits(self, counts: torch.Tensor) -> List[int]:
        """counts per expert [E] -> tokens destined per EP rank."""
        per_rank = counts.view(-1, self.experts_per_rank).sum(dim=1)
        return [int(c) for c in per_rank]

    def _decode_tables(self, device):
        """Device-resident expert weight pointer tables for the routed
        decode GEMVs (built lazily, invalidated if the weights move)."""
        key = self.experts[0].gate_proj.weight.data_ptr()
        cached = getattr(self, "_ptr_tables", None)
        if cached is not None and cached[0] == key:
            return cached[1:]
        tg = torch.tensor([e.gate_proj.weight.data_ptr()
                           for e in self.experts], dtype=torch.int64,
                          device=device)
        tu = torch.tensor([e.up_proj.weight.data_ptr()
                           for e in self.experts], dtype=torch.int64,
                          device=device)
        td = torch.tensor([e.down_proj.weight.data_ptr()
                           for e in self.experts], dtype=torch.int64,
                          device=device)
        self._ptr_tables = (key, tg, tu, td)
        return tg, tu, td

    def _decode_forward(self, xt: torch.Tensor):
        """Sync-free decode path (single process, <= 8 routed pairs):
        router topk feeds DEVICE index tensors straight into the routed
        GEMV kernels via expert pointer tables — no bincount/argsort/
        all-to-all and no D2H slice sizes. Weight bytes scale with the
        pair count, so the grouped path stays better at prefill sizes.
        Returns None when unsupported (caller falls back)."""
        H = xt.shape[-1]
        ex = self.experts[0]
        ff = ex.gate_proj.weight.shape[0]
        if not (xt.is_cuda and xt.dtype == torch.bfloat16
                and H % 512 == 0 and ff % 512 == 0
                and type(ex).__name__ == "Expert"):
            return None
        from ..ops import native
        lib = native.load(require=True)
        T = xt.shape[0]
        logits = nn.functional.linear(xt.float(),
                                      self.router.weight.float())
        probs = torch.softmax(logits, dim=-1)
        gates, idx = probs.topk(self.top_k, dim=-1)
        gates = gates / gates.sum(dim=-1, keepdim=True)
        self.aux_loss = xt.new_zeros(())        # inference: unused
        tg, tu, td = self._decode_tables(xt.device)
        N = T * self.top_k
        idxf = idx.reshape(-1).to(torch.int64)
        xc = xt.contiguous()
        y = torch.empty(N, ff, dtype=torch.bfloat16, device=xt.device)
        rc = lib.gemv_moe_swiglu(native.stream_ptr(), tg.data_ptr(),
                                 tu.data_ptr(), idxf.data_ptr(),
                                 xc.data_ptr(), y.data_ptr(),
                                 ff, H, N, self.top_k)
        native.check_rc(rc, "gemv_moe_swiglu", f"ff={ff} H={H} N={N}")
        z = torch.empty(N, H, dtype=torch.bfloat16, device=xt.device)
        rc = lib.gemv_moe(native.stream_ptr(), td.data_ptr(),
                          idxf.data_ptr(), y.data_ptr(), z.data_ptr(),
                          H, ff, N, 1)
        native.check_rc(rc, "gemv_moe", f"H={H} ff={ff} N={N}")
        out = (z.reshape(T, self.top_k, H).float()
               * gates[..., None]).sum(dim=1)
        return out.to(xt.dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        orig_shape = x.shape
        H = orig_shape[-1]
        xt = x.reshape(-1, H)
        if (not torch.is_grad_enabled() and self.tp_group is None
                and _group_size(self.group) == 1
                and xt.shape[0] * self.top_k <= 16):
            fast = self._decode_forward(xt)
            if fast is not None:
                return fast.reshape(orig_shape)
        if self.tp_group is not None and _group_size(self.tp_group) > 1:
            # the ONE f-op for the whole MoE under EP x TP: forward
            # identity, backward sums the tp peers' partial input grads
            # (each peer only backprops its own expert shards)
            from .tp import _CopyToTP
            xt = _CopyToTP.apply(xt, self.tp_group)
        T = xt.shape[0]

        # routing in fp32 regardless of model dtype (gate numerics);
        # grads flow back into the (possibly bf16) router weight
        logits = nn.functional.linear(xt.float(), self.router.weight.float())
        probs = torch.softmax(logits, dim=-1)
        gates, idx = probs.topk(self.top_k, dim=-1)          # [T, k]
        gates = gates / gates.sum(dim=-1, keepdim=True)

        # flatten (token, k) pairs and sort by expert id — experts are
        # contiguous per rank, so this is also sorted by destination rank
        # AND by local expert within each destination
        flat_e = idx.reshape(-1)                              # [T*k]
        flat_t = (torch.arange(T, device=x.device)
                  .repeat_interleave(self.top_k))
        order = torch.argsort(flat_e, stable=True)
        inv = torch.empty_like(order)
        inv[order] = torch.arange(order.numel(), device=order.device)
        from ..ops import dispatch_rows, gather_rows, moe_combine
        send_tokens = dispatch_rows(xt, flat_t[order], inv, self.top_k)
        counts = torch.bincount(flat_e, minlength=self.n_experts)
        in_splits = self._expert_owner_splits(counts)

        # Switch-style load-balance auxiliary loss over THIS rank's
        # tokens: E * sum_e f_e * P_e, where f_e is the routed fraction
        # (non-differentiable counts) and P_e the mean router prob
        # (differentiable) — minimized at 1.0 by a uniform router.
        # Consumed by MoELlamaModel (aux_loss_coef); harmless elsewhere.
        f = counts.float() / max(T * self.top_k, 1)
        self.aux_loss = self.n_experts * (f.detach() * probs.mean(0)).sum()

        # exchange split sizes: every rank needs how much each peer sends it
        ep = _group_size(self.group)
        if ep > 1:
            # per-expert counts from every source rank (device tensors so
            # the same code runs over gloo/cpu and rccl/gpu)
            ecounts = [torch.zeros(self.n_experts, dtype=torch.long,
                                   device=x.device) for _ in range(ep)]
            dist.all_gather(ecounts, counts.to(torch.long), group=self.group)
            r = dist.get_rank(self.group)
            out_splits = [
                int(c.view(ep, self.experts_per_rank)[r].sum())
                for c in ecounts]
            e0 = r * self.experts_per_rank
            per_src = torch.stack(
                [c[e0:e0 + self.experts_per_rank] for c in ecounts])
        else:
            out_splits = in_splits
            per_src = counts.view(1, -1)

        recv = _AllToAll.apply(send_tokens, in_splits, out_splits,
                               self.group)

        # recv layout: source-rank major, each source's block sorted by
        # local expert. Permute to expert-major (contiguous run per
        # expert) with ONE device-side stable argsort over per-row expert
        # labels, apply each expert to its slice, permute back. (Round 1
        # built per-(expert, src) index ranges in a Python loop — dozens
        # of tiny kernels and a HOST SYNC per iteration from the int()
        # casts; this form has a single host sync for the slice sizes.)
        le_ids = torch.arange(self.experts_per_rank, device=recv.device)
        block_le = le_ids.repeat(per_src.shape[0])     # src-major blocks
        le_labels = torch.repeat_interleave(block_le, per_src.reshape(-1))
        perm = torch.argsort(le_labels, stable=True)
        grouped = gather_rows(recv, perm, bijective=True)
        seg_sizes = per_src.sum(dim=0).tolist()        # one host sync
        seg_out = []
        off = 0
        for le, size in enumerate(seg_sizes):
            if size:
                seg_out.append(self.experts[le](grouped[off:off + size]))
            off += size
        expert_out = (torch.cat(seg_out) if seg_out
                      else grouped[:0])
        inv_perm = torch.empty_like(perm)
        inv_perm[perm] = torch.arange(perm.numel(), device=perm.device)
        outputs = gather_rows(expert_out, inv_perm, bijective=True)

        back = _AllToAll.apply(outputs, out_splits, in_splits, self.group)

        # undo the sort, apply gates, combine top-k — one fused pass
        # (gather + gate + sum and their backwards: ops/hip rows kernels)
        y = moe_combine(back, inv, gates)
        return y.reshape(orig_shape)

    @torch.no_grad()
    def shard_from_full(self, full: "MoEMLP") -> None:
        """Take this rank's expert block (and the replicated router) from a
        single-process MoE with the full expert list."""
        self.router.weight.copy_(full.router.weight)
        r = dist.get_rank(self.group) if (
            dist.is_initialized() and _group_size(self.group) > 1) else 0
        base = r * self.experts_per_rank
        for i, ex in enumerate(self.experts):
            src = full.experts[base + i]
            ex.gate_proj.weight.copy_(src.gate_proj.weight)
            ex.up_proj.weight.copy_(src.up_proj.weight)
            ex.down_proj.weight.copy_(src.down_proj.weight)


@dataclass
class EPTopology:
    """world = edp x ep x tp, rank = ((edp_r*ep + ep_r)*tp + tp_r) —
    tp innermost (adjacent GPUs over direct xGMI), then EP planes (token
    all-to-all among same-tp ranks), then expert-dp.

    Groups:
      ep_group      token-dispatch plane: same (edp_r, tp_r)
      edp_group     same expert shard across edp replicas: same (ep_r, tp_r)
      tp_group      tensor shards of one model instance: same (edp_r, ep_r)
      dense_dp_group  all data replicas of a dense/attn shard: same tp_r
    None means the default (whole-world) group."""
    world: int
    rank: int
    ep_size: int
    ep_rank: int
    ep_group: Optional[object]
    edp_size: int
    edp_rank: int
    edp_group: Optional[object]
    tp_size: int = 1
    tp_rank: int = 0
    tp_group: Optional[object] = None
    dense_dp_group: Optional[object] = None
    data_rank: int = 0        # index among the edp*ep data replicas
    data_replicas: int = 1


def build_ep_topology(ep_size: int = 0, tp_size: int = 1) -> EPTopology:
    """ep_size == 0 means 'everything left after tp is one EP group'.
    Every rank must call with the same sizes (new_group is collective)."""
    if not dist.is_initialized():
        return EPTopology(1, 0, max(ep_size, 1), 0, None, 1, 0, None,
                          tp_size=max(tp_size, 1))
    world = dist.get_world_size()
    rank = dist.get_rank()
    tp = max(tp_size, 1)
    if world % tp != 0:
        raise ValueError(f"world {world} not divisible by tp={tp}")
    ep = ep_size or world // tp
    if (world // tp) % ep != 0:
        raise ValueError(f"world {world} not divisible by ep*tp={ep * tp}")
    edp = world // (ep * tp)
    tp_r = rank % tp
    ep_r = (rank // tp) % ep
    edp_r = rank // (ep * tp)

    def _mk(groups_ranks, mine_key):
        mine = None
        for key, ranks in groups_ranks:
            if len(ranks) == world:
                return None if key == mine_key else mine
            grp = dist.new_group(ranks)
            if key == mine_key:
                mine = grp
        return mine

    ep_group = edp_group = tp_group = dense_dp_group = None
    if ep > 1:
        ep_group = _mk([((d, t), [(d * ep + e) * tp + t
                                  for e in range(ep)])
                        for d in range(edp) for t in range(tp)],
                       (edp_r, tp_r))
    if edp > 1:
        edp_group = _mk([((e, t), [(d * ep + e) * tp + t
                                   for d in range(edp)])
                         for e in range(ep) for t in range(tp)],
                        (ep_r, tp_r))
    if tp > 1:
        tp_group = _mk([((d, e), [(d * ep + e) * tp + t
                                  for t in range(tp)])
                        for d in range(edp) for e in range(ep)],
                       (edp_r, ep_r))
    if edp * ep > 1:
        dense_dp_group = _mk([(t, [(d * ep + e) * tp + t
                                   for d in range(edp)
                                   for e in range(ep)])
                              for t in range(tp)], tp_r)
    return EPTopology(world, rank, ep, ep_r, ep_group,
                      edp, edp_r, edp_group,
                      tp_size=tp, tp_rank=tp_r, tp_group=tp_group,
                      dense_dp_group=dense_dp_group,
                      data_rank=edp_r * ep + ep_r,
                      data_replicas=edp * ep)


def solo_group():
    """A 1-member group per rank (collective: every rank creates every
    solo group). Passing it as ep_group yields a truly UNSHARDED model
    inside an initialized world, where group=None would mean world."""
    if not dist.is_initialized():
        return None
    mine = None
    for r in range(dist.get_world_size()):
        g = dist.new_group([r])
        if r == dist.get_rank():
            mine = g
    return mine


def diversify_experts(model, seed: int, ep_rank: int = 0) -> None:
    """Re-draw each expert's weights from a generator keyed by its GLOBAL
    expert id, so (a) shards differ across ep ranks, (b) edp peers of the
    same shard match bit-for-bit, and (c) a single-process full model
    (ep_rank=0 owning every expert) reproduces every shard exactly."""
    with torch.no_grad():
        for li, blk in enumerate(model.blocks):
            moe = blk.moe
            for le, ex in enumerate(moe.experts):
                geid = ep_rank * moe.experts_per_rank + le
                g = torch.Generator().manual_seed(
                    seed * 1_000_003 + li * 1009 + geid)
                for w in (ex.gate_proj.weight, ex.up_proj.weight,
                          ex.down_proj.weight):
                    # nn.Linear default init: kaiming_uniform(a=sqrt(5))
                    # == U(-1/sqrt(fan_in), 1/sqrt(fan_in))
                    bound = 1.0 / (w.shape[1] ** 0.5)
                    w.copy_(torch.empty(
                        w.shape, dtype=torch.float32).uniform_(
                        -bound, bound, generator=g).to(w.dtype))


class EPTrainer:
    """DP x EP (x TP) trainer for the MoE-Llama family: every (edp, ep)
    coordinate is a data-parallel worker over its own batch stream (tp
    peers share data); each EP plane shards the experts and exchanges
    tokens by all-to-all; with tp_size > 1 attention and each expert's
    matrices are additionally tensor-sharded (TPExpert).

    Gradient seams (loss = mean over the global batch of edp*ep data
    replicas):
      * whole flat grad pre-scaled by 1/(edp*ep),
      * expert spans all-reduced over the expert-dp group (the edp peers
        holding the same shard) — a no-op when edp == 1,
      * every other span (dense replicated + attention tp-shards)
        all-reduced over the dense-dp group (all data replicas at my
        tp_rank; the whole world when tp == 1).
    The global grad-norm clip counts every shard exactly once:
      * expert normsq summed over the ep plane AND the tp group,
      * attention-shard normsq summed over the tp group,
      * replicated normsq counted locally (identical on every rank).
    """

    def __init__(self, cfg, ep_size: int = 0, device=None,
                 tp_size: int = 1):
        from ..models.config import CONFIGS
        from ..models.moe_llama import MoELlamaModel
        from ..optim import FlatAdamW
        from ..parallel.flat import FlatParamStore, classify_spans
        from ..launcher.data import make_batches

        self.cfg = cfg
        mcfg = CONFIGS[cfg.model]
        self.device = torch.device(device or "cpu")
        self.topo = build_ep_topology(ep_size, tp_size)
        torch.manual_seed(cfg.seed)       # identical dense init everywhere
        if self.topo.tp_size > 1:
            # build the unsharded model once and slice both tp and ep
            # shards from it (exact shard_from_full semantics)
            with torch.device(self.device):
                full = MoELlamaModel(mcfg, ep_group=solo_group())
            diversify_experts(full, cfg.seed, ep_rank=0)  # all experts
            with torch.device(self.device):
                model = MoELlamaModel(mcfg, ep_group=self.topo.ep_group,
                                      tp_group=self.topo.tp_group)
            model.shard_from_full(full)
            del full
        else:
            with torch.device(self.device):
                model = MoELlamaModel(mcfg, ep_group=self.topo.ep_group)
            diversify_experts(model, cfg.seed, self.topo.ep_rank)
        self.model = model.to(torch.bfloat16)
        from ..ops import make_inv_freq
        self.model.inv_freq = make_inv_freq(mcfg.head_dim, mcfg.rope_theta,
                                            device=self.device)
        self.store = FlatParamStore(self.model, device=self.device)
        # optimizer clips nothing; EPTrainer applies the topology-aware clip
        self.opt = FlatAdamW(self.store, lr=cfg.lr, betas=cfg.betas,
                             weight_decay=cfg.weight_decay,
                             clip_grad_norm=0.0)
        self.expert_spans, rest = classify_spans(
            self.store, lambda n: ".experts." in n)
        if self.topo.tp_size > 1:
            named = dict(self.model.named_parameters())
            shard_rest = {n for n in named
                          if ".experts." not in n
                          and getattr(named[n], "tp_sharded", False)}
            self.attn_shard_spans, _ = classify_spans(
                self.store, lambda n: n in shard_rest)
            _, self.replicated_spans = classify_spans(
                self.store,
                lambda n: n in shard_rest or ".experts." in n)
        else:
            self.attn_shard_spans = []
            self.replicated_spans = rest
        self.dense_spans = rest   # everything non-expert: one sync class
        self.data = make_batches(cfg, self.device,
                                rank=self.topo.data_rank)
        self.step_count = 0

    def _reduce_grads(self) -> None:
        t = self.topo
        if t.data_replicas == 1:
            return
        fg = self.store.flat_grad
        fg.mul_(1.0 / t.data_replicas)
        for s, e in self.dense_spans:
            dist.all_reduce(fg[s:e], group=t.dense_dp_group)
        if t.edp_size > 1:
            for s, e in self.expert_spans:
                dist.all_reduce(fg[s:e], group=t.edp_group)

    def _clip_grads(self) -> None:
        clip = self.cfg.clip_grad_norm
        if not clip or clip <= 0:
            return
        t = self.topo
        fg = self.store.flat_grad
        expert_nsq = fg.new_zeros((), dtype=torch.float32)
        for s, e in self.expert_spans:
            expert_nsq += fg[s:e].float().pow(2).sum()
        if t.ep_size > 1 and t.world > 1:
            dist.all_reduce(expert_nsq, group=t.ep_group)
        if t.tp_size > 1:
            dist.all_reduce(expert_nsq, group=t.tp_group)
            shard_nsq = fg.new_zeros((), dtype=torch.float32)
            for s, e in self.attn_shard_spans:
                shard_nsq += fg[s:e].float().pow(2).sum()
            dist.all_reduce(shard_nsq, group=t.tp_group)
            rep_nsq = fg.new_zeros((), dtype=torch.float32)
            for s, e in self.replicated_spans:
                rep_nsq += fg[s:e].float().pow(2).sum()
            gnorm = (expert_nsq + shard_nsq + rep_nsq).sqrt()
        else:
            dense_nsq = fg.new_zeros((), dtype=torch.float32)
            for s, e in self.dense_spans:
                dense_nsq += fg[s:e].float().pow(2).sum()
            gnorm = (dense_nsq + expert_nsq).sqrt()
        if float(gnorm) > clip:
            fg.mul_(clip / float(gnorm))

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
        self._reduce_grads()
        self._clip_grads()
        self.opt.step(grad_pre_scale=1.0)
        self.opt.zero_grad()
        self.step_count += 1
        return loss.detach()
