"""Training engine: model build, synthetic data, fused step loop.

This is the workload the operator's AITrainingJob pods run (launcher ->
Trainer) and what bench.py measures standalone. DDP is the in-house
bucketed engine over RCCL/xGMI (parallel/ddp.py), the optimizer the fused
flat AdamW (optim.py).
"""
from __future__ import annotations

import os
from contextlib import nullcontext
from dataclasses import dataclass
from typing import Iterator, Optional, Tuple

import torch

from .models.config import CONFIGS, LlamaConfig
from .models.llama import LlamaModel
from .optim import FlatAdamW
from .ops import make_inv_freq
from .parallel.ddp import DDPEngine
from .parallel.dist_ctx import DistContext
from .parallel.flat import FlatParamStore


@dataclass
class TrainConfig:
    model: str = "llama3-8b"
    micro_batch: int = 1
    grad_accum: int = 4
    seq_len: int = 4096
    lr: float = 3e-4
    weight_decay: float = 0.1
    betas: Tuple[float, float] = (0.9, 0.95)
    clip_grad_norm: float = 1.0
    checkpoint_activations: bool = False
    bucket_bytes: int = 128 << 20
    seed: int = 1234
    # capture the whole optimizer step (grad_accum micro-batches + fused
    # AdamW) as ONE hipGraph: removes the per-kernel launch gaps that
    # measured ~18% of step wall time on MI355X
    use_graphs: bool = False
    # tensor-parallel degree (world splits into dp x tp; tp ranks are
    # contiguous -> adjacent GPUs over xGMI). 1 = pure DP (the flagship
    # 8B config — one replica fits a 288 GB MI355X).
    tp_size: int = 1
    # vocab-parallel LM head + sharded CE under TP: head weight and
    # logits shrink to 1/tp per rank (parallel/vocab_parallel.py)
    vocab_parallel: bool = False
    # Megatron-style sequence parallelism on top of TP (tp_size > 1):
    # norms/residual run on seq shards with all-gather/reduce-scatter
    # seams instead of the f/g all-reduces (parallel/sp.py)
    sequence_parallel: bool = False
    # ZeRO-1: shard the fp32 master weights + AdamW moments across the dp
    # group (12 bytes/param -> 12/dp); each rank updates its slice, then
    # the updated bf16 params are re-assembled across the group
    zero1: bool = False
    # fp8 (e4m3fn) projection GEMMs — forward only, per-tensor scales,
    # backward stays bf16. OPT-IN: the flagship bench dtype stays bf16;
    # publish fp8 numbers as a separate config.
    fp8_projections: bool = False
    # bf16 first/second AdamW moments: ~29% less optimizer HBM traffic and
    # half the checkpoint moment bytes. Default OFF until loss-curve parity
    # is validated for the target run (tests cover short-horizon parity).
    adamw_bf16_moments: bool = False
    # ZeRO-1 comm pattern: reduce-scatter grads + all-gather params over
    # RCCL (halves grad traffic vs all-reduce; per-link-bound on xGMI).
    # Falls back to all-reduce + per-slice broadcast on gloo (no
    # reduce_scatter_tensor there) and for tiny flat buffers.
    zero1_rs: bool = True
    # real data: path to a flat token-id binary (launcher/data.py);
    # "" = deterministic synthetic stream
    data_path: str = ""
    data_dtype: str = "uint16"
    # LR schedule: linear warmup to lr, then cosine decay to min_lr over
    # lr_decay_steps (0 = constant lr after warmup). NOTE: not applied
    # inside a captured hipGraph (the kernel arg is baked at capture).
    warmup_steps: int = 0
    lr_decay_steps: int = 0
    min_lr: float = 0.0

    @property
    def model_config(self) -> LlamaConfig:
        return CONFIGS[self.model]

    def tokens_per_step_per_rank(self) -> int:
        return self.micro_batch * self.grad_accum * self.seq_len


def build_model(cfg: LlamaConfig, device: torch.device,
                checkpoint_activations: bool = False) -> LlamaModel:
    from .models.moe_llama import MoELlamaConfig
    if isinstance(cfg, MoELlamaConfig):
        raise ValueError(
            f"{cfg.name!r} is a MoE config; build MoELlamaModel directly "
            "(EPTrainer / PPTrainer do)")
    with torch.device(device):
        model = LlamaModel(cfg, checkpoint_activations=checkpoint_activations)
    model = model.to(torch.bfloat16)
    # RoPE frequencies must stay fp32 (precision of angles at long context)
    model.inv_freq = make_inv_freq(cfg.head_dim, cfg.rope_theta,
                                   device=device)
    return model


def synthetic_batches(cfg: TrainConfig, device: torch.device,
                      rank: int = 0) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
    """Deterministic-per-rank synthetic LM batches (no network for real data)."""
    g = torch.Generator(device="cpu").manual_seed(cfg.seed * 1000 + rank)
    V = cfg.model_config.vocab_size
    while True:
        tokens = torch.randint(0, V, (cfg.micro_batch, cfg.seq_len + 1),
                               generator=g)
        tokens = tokens.to(device, non_blocking=True)
        yield tokens[:, :-1].contiguous(), tokens[:, 1:].contiguous()


class Trainer:
    def __init__(self, cfg: TrainConfig, ctx: Optional[DistContext] = None,
                 device: Optional[torch.device] = None):
        if cfg.fp8_projections:
            os.environ["AITJ_FP8_PROJ"] = "1"
        from .models.moe_llama import MoELlamaConfig
        if isinstance(cfg.model_config, MoELlamaConfig):
            raise ValueError(
                f"{cfg.model!r} is a MoE config: train it with EPTrainer "
                "(launcher --ep) or PPTrainer (--pp); the dense Trainer "
                "would silently build a dense model from it")
        self.cfg = cfg
        self.ctx = ctx or DistContext()
        if device is None:
            device = torch.device(
                f"cuda:{self.ctx.local_rank}" if torch.cuda.is_available()
                else "cpu")
        self.device = device
        torch.manual_seed(cfg.seed)  # identical init on every rank
        from .parallel.groups import build_topology
        self.topo = build_topology(cfg.tp_size)
        if cfg.tp_size > 1:
            kw = {}
            if cfg.sequence_parallel:
                assert not cfg.vocab_parallel, \
                    "vocab_parallel with sequence_parallel is roadmap"
                from .parallel.sp import SPLlamaModel as _TPModel
            else:
                from .parallel.tp_llama import TPLlamaModel as _TPModel
                kw["vocab_parallel_head"] = cfg.vocab_parallel
            full = build_model(cfg.model_config, device,
                               cfg.checkpoint_activations)
            self.model = _TPModel(cfg.model_config,
                                  group=self.topo.tp_group, **kw).to(
                full.embed.weight.dtype).to(device)
            self.model.inv_freq = full.inv_freq
            self.model.shard_from_full(full)
            del full
        else:
            self.model = build_model(cfg.model_config, device,
                                     cfg.checkpoint_activations)
        self.store = FlatParamStore(self.model, device=device)
        self.ddp = DDPEngine(self.store, process_group=self.topo.dp_group,
                             bucket_bytes=cfg.bucket_bytes,
                             world_size=self.topo.dp_size)
        # Under TP the flat buffer mixes per-rank shards with replicated
        # params, so the optimizer's LOCAL-norm clip would scale tp peers
        # differently and silently diverge the replicated params. The
        # trainer applies a TP-aware global clip instead (_tp_clip).
        tp_aware_clip = self.topo.tp_size > 1 and cfg.clip_grad_norm > 0
        self._zero_shards = None
        self._zero_chunk = 0       # equal reduce-scatter chunk (elements)
        self._zero_rs = False
        shard = None
        if cfg.zero1 and self.topo.dp_size > 1:
            from .parallel.flat import ALIGN, _aligned
            dp, r = self.topo.dp_size, self.topo.dp_rank
            chunk_lo = (self.store.total // dp) // ALIGN * ALIGN
            # the RS layout leaves non-shard grad regions UNreduced, so it
            # cannot combine with the TP-aware clip (which reads the full
            # flat grad); zero1+tp falls back to the broadcast layout
            if cfg.zero1_rs and chunk_lo > 0 and self.topo.tp_size == 1:
                # equal chunks + tail on the last rank: the equal region
                # [0, dp*chunk) goes through ONE reduce_scatter_tensor /
                # all_gather_into_tensor pair on RCCL; the small tail
                # (< dp*64 elements) is all-reduced / broadcast
                self._zero_chunk = chunk_lo
                self._zero_rs = True
                self._zero_shards = [
                    (i * chunk_lo,
                     (i + 1) * chunk_lo if i < dp - 1 else self.store.total)
                    for i in range(dp)]
            else:
                chunk = _aligned(-(-self.store.total // dp))
                self._zero_shards = [
                    (min(i * chunk, self.store.total),
                     min((i + 1) * chunk, self.store.total))
                    for i in range(dp)]
            shard = self._zero_shards[r]
        self.opt = FlatAdamW(self.store, lr=cfg.lr, betas=cfg.betas,
                             weight_decay=cfg.weight_decay,
                             clip_grad_norm=(0.0 if tp_aware_clip
                                             else cfg.clip_grad_norm),
                             shard=shard,
                             shard_norm_group=(self.topo.dp_group
                                               if self._zero_rs else None),
                             bf16_moments=cfg.adamw_bf16_moments)
        self._tp_spans = None
        if tp_aware_clip:
            from .parallel.flat import classify_spans
            named = dict(self.model.named_parameters())
            self._tp_spans = classify_spans(
                self.store,
                lambda n: getattr(named[n], "tp_sharded", False))
        # tp peers train on the SAME data (they hold shards of one replica)
        from .launcher.data import make_batches
        self.data = make_batches(cfg, device, self.topo.dp_rank)
        self.step_count = 0
        self._graph = None
        self._static_batches = None
        self._static_loss = None

    # -- eager step body (also what the graph captures) -------------------
    def _step_body(self, batches, in_graph: bool) -> torch.Tensor:
        cfg = self.cfg
        loss = None
        use_rs = self._zero_rs and not in_graph
        for micro, (tokens, targets) in enumerate(batches):
            sync = micro == cfg.grad_accum - 1 and not use_rs
            with (nullcontext() if sync else self.ddp.no_sync()):
                loss = self.model(tokens, targets)
                # scale so accumulated grads average over micro-batches
                (loss / cfg.grad_accum).backward()
        self.ddp.finish_backward()
        if use_rs:
            self._zero1_grad_sync()
        if not in_graph and (cfg.warmup_steps or cfg.lr_decay_steps):
            from .optim import lr_at
            self.opt.lr = lr_at(self.opt.step_count, cfg.lr,
                                cfg.warmup_steps, cfg.lr_decay_steps,
                                cfg.min_lr)
        if cfg.sequence_parallel and self.topo.tp_size > 1:
            # seq-sharded params (norms/embed) saw only this rank's
            # positions: sum their grads over the tp group BEFORE clip
            self.model.allreduce_sp_grads()
        if self._tp_spans is not None:
            self._tp_clip()
            self.opt.step(grad_pre_scale=1.0, in_graph_capture=in_graph)
        else:
            self.opt.step(grad_pre_scale=self.ddp.grad_pre_scale,
                          in_graph_capture=in_graph)
        if self._zero_shards is not None:
            self._zero_allgather_params()
        if self.cfg.fp8_projections:
            from .ops import fp8
            fp8.bump_version()     # weights changed: refresh fp8 images
        self.opt.zero_grad()
        return loss

    def _zero1_grad_sync(self) -> None:
        """ZeRO-1 gradient seam over RCCL: ONE reduce_scatter_tensor for
        the equal region (each rank receives only its shard's SUM — half
        the xGMI traffic of an all-reduce) + a small all-reduced tail.
        gloo has no reduce_scatter_tensor: plain all-reduce fallback."""
        import torch.distributed as dist
        fg = self.store.flat_grad
        g = self.topo.dp_group
        if dist.get_backend(g) != "nccl":
            dist.all_reduce(fg, group=g)
            return
        dp = self.topo.dp_size
        c = self._zero_chunk
        out = torch.empty(c, dtype=fg.dtype, device=fg.device)
        dist.reduce_scatter_tensor(out, fg[:dp * c], group=g)
        s0 = self.topo.dp_rank * c
        fg[s0:s0 + c].copy_(out)
        if self.store.total > dp * c:
            dist.all_reduce(fg[dp * c:], group=g)

    def _zero_allgather_params(self) -> None:
        """ZeRO-1 re-assembly. RS layout + RCCL: ONE
        all_gather_into_tensor over the equal region + tail broadcast
        from the last rank; otherwise per-slice broadcasts (gloo-safe,
        uneven tails trivial)."""
        import torch.distributed as dist
        g = self.topo.dp_group
        fp = self.store.flat_param
        if self._zero_rs and dist.get_backend(g) == "nccl":
            dp = self.topo.dp_size
            c = self._zero_chunk
            s0 = self.topo.dp_rank * c
            local = fp[s0:s0 + c].contiguous()
            dist.all_gather_into_tensor(fp[:dp * c], local, group=g)
            if self.store.total > dp * c:
                src = (dist.get_global_rank(g, dp - 1)
                       if g is not None else dp - 1)
                dist.broadcast(fp[dp * c:], src=src, group=g)
            return
        for r, (s, e) in enumerate(self._zero_shards):
            if e <= s:
                continue
            src = dist.get_global_rank(g, r) if g is not None else r
            dist.broadcast(fp[s:e], src=src, group=g)

    def _tp_clip(self) -> None:
        """Global grad-norm clip under TP: sharded-param normsq summed
        over the tp group plus replicated normsq counted once. The DP
        pre-scale is folded into the flat grad first so every tp peer
        applies the IDENTICAL clip factor (bit-aligned replicated params).
        Not hipGraph-capturable (host sync on the norm) — TP is not the
        graphed flagship path."""
        import torch.distributed as dist
        fg = self.store.flat_grad
        if self.ddp.grad_pre_scale != 1.0:
            fg.mul_(self.ddp.grad_pre_scale)
        sharded, replicated = self._tp_spans
        nsq_sh = fg.new_zeros((), dtype=torch.float32)
        for s, e in sharded:
            nsq_sh += fg[s:e].float().pow(2).sum()
        if self.topo.tp_group is not None:
            dist.all_reduce(nsq_sh, group=self.topo.tp_group)
        nsq_rep = fg.new_zeros((), dtype=torch.float32)
        for s, e in replicated:
            nsq_rep += fg[s:e].float().pow(2).sum()
        gnorm = float((nsq_sh + nsq_rep).sqrt())
        clip = self.cfg.clip_grad_norm
        if gnorm > clip:
            fg.mul_(clip / gnorm)

    def _capture_graph(self) -> None:
        """Warm up twice on a side stream, then record one full step."""
        cfg = self.cfg
        self._static_batches = [tuple(t.clone() for t in next(self.data))
                                for _ in range(cfg.grad_accum)]
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                self.opt.bump_step()
                self.opt.update_bias_correction()
                self._step_body(self._static_batches, in_graph=True)
                self.step_count += 1
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._static_loss = self._step_body(self._static_batches,
                                                in_graph=True)

    def train_step(self) -> torch.Tensor:
        """One optimizer step = grad_accum micro-batches; returns the last
        micro-batch loss (device tensor, not synced)."""
        cfg = self.cfg
        if cfg.use_graphs and self.device.type == "cuda":
            if self._graph is None:
                self._capture_graph()
            for i in range(cfg.grad_accum):
                tokens, targets = next(self.data)
                self._static_batches[i][0].copy_(tokens, non_blocking=True)
                self._static_batches[i][1].copy_(targets, non_blocking=True)
            self.opt.bump_step()
            self.opt.update_bias_correction()
            self._graph.replay()
            self.step_count += 1
            return self._static_loss.detach()
        batches = [next(self.data) for _ in range(cfg.grad_accum)]
        loss = self._step_body(batches, in_graph=False)
        self.step_count += 1
        return loss.detach()
