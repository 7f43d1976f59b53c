"""Pipeline parallelism: Llama stage partitioning + GPipe/1F1B schedules
over torch.distributed P2P (RCCL send/recv on GPU, gloo in CPU tests).

The launcher forms pipelines from the operator's injected env
(BASELINE.json north star); PPTrainer composes with DP (replicated
pipelines), TP (TP-sharded stages) and EP (MoE stages whose experts
shard across the stage plane) — docs/PARALLELISM.md. The flagship 8B
bench stays pure DP (one replica fits a 288 GB MI355X); PP is for models
that outgrow one GPU, e.g. llama3-70b. On one 8xMI355X node every stage
boundary is a direct xGMI link, so the P2P activations ride
point-to-point bandwidth (~153 GB/s/link) without touching collectives.

Pipeline boundary payload: the pre-norm Llama block carries TWO tensors
(branch output x and the running residual stream — models/llama.py Block),
so each hop sends/recvs the pair.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ..models.config import LlamaConfig
from ..models.llama import Block, LlamaModel
from ..ops import fused_cross_entropy, fused_rmsnorm


def partition_layers(num_layers: int, n_stages: int) -> List[range]:
    """Contiguous, balanced layer ranges per stage."""
    base = num_layers // n_stages
    extra = num_layers % n_stages
    out = []
    start = 0
    for s in range(n_stages):
        n = base + (1 if s < extra else 0)
        out.append(range(start, start + n))
        start += n
    return out


class LlamaStage(nn.Module):
    """One pipeline stage of a LlamaModel: first stage owns the embedding,
    last owns final norm + lm_head + loss. Built as views onto a full
    model's modules so checkpoints/state are shared with the unsharded
    layout (each PP rank instantiates only its slice in production via
    `from_config`)."""

    def __init__(self, cfg: LlamaConfig, layers: List[Block],
                 embed: Optional[nn.Embedding],
                 final_norm_weight: Optional[nn.Parameter],
                 lm_head: Optional[nn.Linear], inv_freq: torch.Tensor):
        super().__init__()
        self.cfg = cfg
        self.checkpoint_activations = False
        self.embed = embed
        self.blocks = nn.ModuleList(layers)
        self.lm_head = lm_head
        if final_norm_weight is not None:
            self.final_norm_weight = final_norm_weight
        else:
            self.final_norm_weight = None
        self.register_buffer("inv_freq", inv_freq, persistent=False)

    @property
    def is_first(self) -> bool:
        return self.embed is not None

    @property
    def is_last(self) -> bool:
        return self.lm_head is not None

    @classmethod
    def from_model(cls, model: LlamaModel, stage: int,
                   n_stages: int) -> "LlamaStage":
        parts = partition_layers(model.cfg.num_layers, n_stages)
        return cls(
            model.cfg,
            [model.blocks[i] for i in parts[stage]],
            model.embed if stage == 0 else None,
            model.final_norm_weight if stage == n_stages - 1 else None,
            model.lm_head if stage == n_stages - 1 else None,
            model.inv_freq,
        )

    @classmethod
    def from_config(cls, cfg: LlamaConfig, stage: int, n_stages: int,
                    device=None) -> "LlamaStage":
        """Standalone stage (only this slice's parameters exist)."""
        with torch.device(device or "cpu"):
            model = LlamaModel(cfg)
        return cls.from_model(model, stage, n_stages)

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor],
                targets: Optional[torch.Tensor] = None):
        """First stage: x is the token ids. Last stage (with targets):
        returns the per-token loss vector. Middle: returns (x, residual)."""
        if self.is_first:
            x = self.embed(x)
            residual = None
        for blk in self.blocks:
            if self.checkpoint_activations and self.training:
                if residual is None:
                    residual = torch.zeros_like(x)
                x, residual = torch.utils.checkpoint.checkpoint(
                    blk, x, residual, self.inv_freq, use_reentrant=False)
            else:
                x, residual = blk(x, residual, self.inv_freq)
        if not self.is_last:
            return x, residual
        normed, _ = fused_rmsnorm(x, self.final_norm_weight, residual,
                                  self.cfg.norm_eps)
        logits = self.lm_head(normed)
        if targets is None:
            return logits
        T = logits.shape[0] * logits.shape[1]
        per_tok = fused_cross_entropy(
            logits.reshape(T, -1).contiguous(), targets.reshape(T))
        n_valid = (targets.reshape(T) != -100).sum().clamp(min=1)
        return per_tok.sum() / n_valid

    def pop_aux(self):
        """Auxiliary loss of the micro-batch that JUST ran forward, for
        stages whose extra losses cannot ride the pipeline (MoEStage);
        dense stages have none."""
        return None


class MoEStage(LlamaStage):
    """Pipeline stage of a MoELlamaModel (PP x EP): blocks are MoEBlocks
    whose MoEMLPs dispatch tokens over the stage's EP plane (the ranks
    holding the other expert shards of the SAME stage). The Switch aux
    term of this stage's routers backprops LOCALLY via pop_aux — only the
    last stage owns the LM loss, so middle stages feed their aux gradient
    into the schedule's backward alongside the received activation
    grads."""

    total_layers: int = 0
    aux_coef: float = 0.0

    @classmethod
    def from_moe_model(cls, model, stage: int,
                       n_stages: int) -> "MoEStage":
        st = cls.from_model(model, stage, n_stages)
        st.total_layers = model.cfg.num_layers
        st.aux_coef = model.cfg.aux_loss_coef
        return st

    def _stage_aux(self):
        aux = sum(blk.moe.aux_loss for blk in self.blocks)
        return self.aux_coef * aux / max(self.total_layers, 1)

    def forward(self, x, residual, targets=None):
        out = super().forward(x, residual, targets)
        if self.is_last and targets is not None and self.aux_coef:
            out = out + self._stage_aux()   # fold into the stage loss
        return out

    def pop_aux(self):
        if self.is_last or not self.aux_coef:
            return None
        return self._stage_aux()


class SPStage(LlamaStage):
    """Pipeline stage of an SPLlamaModel (PP x TP with sequence
    parallelism): blocks run on seq shards, so the P2P boundary payload
    shrinks to S/tp per hop — each tp peer's pipeline ships only its own
    shard. First stage scatters the embedding output; the last stage
    gathers through the replicated-seam op before the (replicated) head
    so its backward splits grads correctly."""

    group = None   # tp group, set by PPTrainer after slicing

    def _shard(self, x):
        from .tp import _group_size
        n = _group_size(self.group)
        if n == 1:
            return x
        import torch.distributed as d
        return x.chunk(n, dim=1)[d.get_rank(self.group)].contiguous()

    def forward(self, x, residual, targets=None):
        from ..ops import fused_cross_entropy as _ce
        from .sp import _GatherSeqReplicated
        if self.is_first:
            x = self._shard(self.embed(x))
            residual = None
        for blk in self.blocks:
            x, residual = blk(x, residual, self.inv_freq)
        if not self.is_last:
            return x, residual
        normed, _ = fused_rmsnorm(x, self.final_norm_weight, residual,
                                  self.cfg.norm_eps)
        normed = _GatherSeqReplicated.apply(normed, self.group)
        logits = self.lm_head(normed)
        if targets is None:
            return logits
        T = logits.shape[0] * logits.shape[1]
        per_tok = _ce(logits.reshape(T, -1).contiguous(),
                      targets.reshape(T))
        n_valid = (targets.reshape(T) != -100).sum().clamp(min=1)
        return per_tok.sum() / n_valid

    @torch.no_grad()
    def allreduce_sp_grads(self) -> None:
        """Seq-sharded params OF THIS STAGE (norms, embed if owned) saw
        only this rank's positions: sum grads over the tp group (mirrors
        SPLlamaModel.allreduce_sp_grads)."""
        from .tp import _group_size
        if _group_size(self.group) == 1:
            return
        params = ([b.input_norm_weight for b in self.blocks]
                  + [b.post_attn_norm_weight for b in self.blocks])
        if self.embed is not None:
            params.append(self.embed.weight)
        if self.final_norm_weight is not None:
            params.append(self.final_norm_weight)
        for p in params:
            if p.grad is not None:
                dist.all_reduce(p.grad, group=self.group)


class GPipeSchedule:
    """Fill-drain (GPipe) schedule: all micro-batch forwards, then all
    backwards in reverse — simple, correct, and bubble-bounded by
    (stages-1)/micro_batches. OneFOneBSchedule below bounds the live
    activations and is the PPTrainer default; GPipe remains as the
    simpler reference schedule (--pp-schedule gpipe)."""

    def __init__(self, stage: LlamaStage, stage_idx: int, n_stages: int,
                 group=None, device=None,
                 pp_ranks: Optional[List[int]] = None):
        self.stage = stage
        self.idx = stage_idx
        self.n = n_stages
        self.group = group
        self.device = device or "cpu"
        # GLOBAL ranks of this pipeline, ordered by stage (p2p dst/src are
        # global ranks even with a subgroup). Default: pure-PP world where
        # rank == stage.
        ranks = pp_ranks if pp_ranks is not None else list(range(n_stages))
        self.prev_rank = ranks[stage_idx - 1] if stage_idx > 0 else None
        self.next_rank = ranks[stage_idx + 1] \
            if stage_idx < n_stages - 1 else None

    # -- P2P helpers ------------------------------------------------------
    def _send(self, t: torch.Tensor, dst: int):
        dist.send(t.detach().contiguous(), dst=dst, group=self.group)

    def _recv(self, shape, dtype) -> torch.Tensor:
        t = torch.empty(*shape, dtype=dtype, device=self.device)
        dist.recv(t, src=self.prev_rank, group=self.group)
        return t

    # -- one optimizer-step's worth of micro-batches ----------------------
    def step(self, micro_batches, hidden_shape,
             act_dtype=torch.float32) -> Optional[torch.Tensor]:
        """micro_batches: list of (tokens, targets) on EVERY rank (only the
        ranks that need them use them: tokens at stage 0, targets at the
        last). hidden_shape = (mb, seq, hidden). Returns the mean loss on
        the last stage (None elsewhere). Gradients accumulate into the
        stage parameters; the caller runs its optimizer."""
        saved = []
        n_micro = len(micro_batches)
        loss_total = None

        for tokens, targets in micro_batches:
            if self.is_first_stage:
                out = self.stage(tokens, None,
                                 targets if self.is_last_stage else None)
                x_in = res_in = None
            else:
                x_in = self._recv(hidden_shape, act_dtype).requires_grad_()
                res_in = self._recv(hidden_shape, act_dtype).requires_grad_()
                out = self.stage(x_in, res_in,
                                 targets if self.is_last_stage else None)
            if self.is_last_stage:
                loss = out / n_micro
                loss_total = loss.detach() if loss_total is None \
                    else loss_total + loss.detach()
                saved.append((x_in, res_in, loss, None, None))
            else:
                x_out, res_out = out
                aux = self.stage.pop_aux()
                self._send(x_out, self.next_rank)
                self._send(res_out, self.next_rank)
                saved.append((x_in, res_in, x_out, res_out,
                              None if aux is None else aux / n_micro))

        for x_in, res_in, a, b, aux in reversed(saved):
            if self.is_last_stage:
                a.backward()  # loss already scaled by 1/n_micro
            else:
                dx = self._recv_grad(a)
                dres = self._recv_grad(b)
                if aux is None:
                    torch.autograd.backward((a, b), (dx, dres))
                else:
                    # the stage-local aux loss (MoE router balance) rides
                    # the same backward pass as the pipeline grads
                    torch.autograd.backward(
                        (a, b, aux), (dx, dres, torch.ones_like(aux)))
            if not self.is_first_stage:
                self._send(x_in.grad, self.prev_rank)
                self._send(res_in.grad, self.prev_rank)

        return loss_total  # sum of (loss/n_micro) == mean over micro-batches

    def _recv_grad(self, like: torch.Tensor) -> torch.Tensor:
        t = torch.empty_like(like)
        dist.recv(t, src=self.next_rank, group=self.group)
        return t

    @property
    def is_first_stage(self) -> bool:
        return self.idx == 0

    @property
    def is_last_stage(self) -> bool:
        return self.idx == self.n - 1


class OneFOneBSchedule(GPipeSchedule):
    """1F1B: after a (stages - stage_idx - 1)-deep warmup, every forward is
    immediately followed by the OLDEST outstanding backward, so at most
    warmup+1 micro-batches of activations are ever live (GPipe keeps all
    of them). Sends are async (isend) so the canonical 1F1B orderings
    cannot rendezvous-deadlock; receives stay blocking."""

    def __init__(self, *args, **kw):
        super().__init__(*args, **kw)
        self.peak_live = 0  # observability/tests: max outstanding micros

    def _send(self, t: torch.Tensor, dst: int):  # async override
        w = dist.isend(t.detach().contiguous(), dst=dst, group=self.group)
        self._send_works.append(w)

    def step(self, micro_batches, hidden_shape,
             act_dtype=torch.float32) -> Optional[torch.Tensor]:
        self._send_works = []
        self.peak_live = 0
        n_micro = len(micro_batches)
        warmup = min(self.n - self.idx - 1, n_micro)
        outstanding = []
        loss_total = None

        def forward(mb):
            tokens, targets = mb
            if self.is_first_stage:
                out = self.stage(tokens, None,
                                 targets if self.is_last_stage else None)
                x_in = res_in = None
            else:
                x_in = self._recv(hidden_shape, act_dtype).requires_grad_()
                res_in = self._recv(hidden_shape, act_dtype).requires_grad_()
                out = self.stage(x_in, res_in,
                                 targets if self.is_last_stage else None)
            if self.is_last_stage:
                outstanding.append((x_in, res_in, out / n_micro, None,
                                    None))
            else:
                x_out, res_out = out
                aux = self.stage.pop_aux()
                self._send(x_out, self.next_rank)
                self._send(res_out, self.next_rank)
                outstanding.append((x_in, res_in, x_out, res_out,
                                    None if aux is None
                                    else aux / n_micro))
            self.peak_live = max(self.peak_live, len(outstanding))

        def backward():
            nonlocal loss_total
            x_in, res_in, a, b, aux = outstanding.pop(0)
            if self.is_last_stage:
                loss_total = a.detach() if loss_total is None \
                    else loss_total + a.detach()
                a.backward()
            else:
                dx = self._recv_grad(a)
                dres = self._recv_grad(b)
                if aux is None:
                    torch.autograd.backward((a, b), (dx, dres))
                else:
                    torch.autograd.backward(
                        (a, b, aux), (dx, dres, torch.ones_like(aux)))
            if not self.is_first_stage:
                self._send(x_in.grad, self.prev_rank)
                self._send(res_in.grad, self.prev_rank)

        for i in range(warmup):
            forward(micro_batches[i])
        for i in range(warmup, n_micro):
            forward(micro_batches[i])
            backward()
        while outstanding:
            backward()
        for w in self._send_works:
            w.wait()
        return loss_total


class PPTrainer:
    """Pipeline trainer: one LlamaStage per rank, per-stage flat param/grad
    store + fused AdamW, a 1F1B (or GPipe) schedule per optimizer step.
    Composes with the launcher's env bootstrap exactly like the DP Trainer
    (training.py).

    Pure PP: pass stage_idx/n_stages (rank == stage, world == pp).
    DP x PP: pass a GridTopology from parallel/groups.build_grid(pp_size=N)
    — each dp replica runs its own pipeline over distinct data, and each
    stage all-reduces its flat grad across its dp peers (one bucket: the
    whole stage slice) before the optimizer step, exactly the DDP seam of
    ddp.py collapsed to a single flat buffer."""

    def __init__(self, cfg, stage_idx: Optional[int] = None,
                 n_stages: Optional[int] = None, device=None,
                 act_dtype=None, schedule: str = "1f1b", grid=None):
        from ..models.config import CONFIGS
        from ..parallel.flat import FlatParamStore
        from ..optim import FlatAdamW
        from ..launcher.data import make_batches

        self.cfg = cfg
        mcfg = CONFIGS[cfg.model]
        self.device = torch.device(device or "cpu")
        if grid is not None:
            stage_idx, n_stages = grid.pp_rank, grid.pp_size
            pp_ranks = getattr(grid, "pp_global_ranks", None) or [
                (grid.dp_rank * grid.pp_size + p) * grid.tp_size
                + grid.tp_rank for p in range(grid.pp_size)]
            self.dp_size, self.dp_rank = grid.dp_size, grid.dp_rank
            self.dp_group = grid.dp_group
            # 4-axis MoE grids (groups.build_moe_grid) add true data
            # parallelism on top of the expert plane
            self.edp_size = getattr(grid, "edp_size", 1)
            self.edp_rank = getattr(grid, "edp_rank", 0)
            self.edp_group = getattr(grid, "edp_group", None)
            self.data_replicas = getattr(grid, "data_replicas",
                                         grid.dp_size)
            self.data_rank = getattr(grid, "data_rank", grid.dp_rank)
            self.dense_dp_group = getattr(grid, "dense_dp_group", None) \
                or grid.dp_group
        else:
            assert stage_idx is not None and n_stages is not None
            pp_ranks = None
            self.dp_size, self.dp_rank, self.dp_group = 1, 0, None
            self.edp_size, self.edp_rank, self.edp_group = 1, 0, None
            self.data_replicas, self.data_rank = 1, 0
            self.dense_dp_group = None
        self.grid = grid
        # all stages of MY pipeline (grad-norm seam); None == default
        # group in the pure-PP world where rank == stage
        self.pp_group = grid.pp_group if grid is not None else None
        from ..models.moe_llama import MoELlamaConfig
        self._is_moe = isinstance(mcfg, MoELlamaConfig)
        assert self.edp_size == 1 or self._is_moe, \
            "4-axis grids (edp) are for MoE pipelines"
        # identical init on every dp replica of a stage (same seed)
        torch.manual_seed(cfg.seed)
        if self._is_moe:
            # PP x EP (x TP): the grid's dp axis doubles as the EP plane —
            # the ranks holding the SAME stage in the other pipelines hold
            # the other expert shards of that stage (groups.py: "EP == DP
            # for MoE"); tp_size > 1 additionally tensor-shards attention
            # and the experts (TPExpert). Pure PP (no grid / dp 1) keeps
            # all experts local.
            assert not cfg.sequence_parallel, \
                "sequence parallelism with MoE stages is roadmap"
            from ..models.moe_llama import MoELlamaModel
            from .ep import diversify_experts, solo_group
            if grid is not None and grid.dp_size > 1:
                ep_plane, plane_rank = grid.dp_group, grid.dp_rank
            else:
                ep_plane, plane_rank = solo_group(), 0
            tp_grp = grid.tp_group if grid is not None and \
                grid.tp_size > 1 else None
            if tp_grp is not None:
                # build unsharded, diversify ALL experts, then slice both
                # the tp and ep shards (exact shard_from_full semantics)
                with torch.device(self.device):
                    full = MoELlamaModel(mcfg, ep_group=solo_group())
                diversify_experts(full, cfg.seed, ep_rank=0)
                with torch.device(self.device):
                    moem = MoELlamaModel(mcfg, ep_group=ep_plane,
                                         tp_group=tp_grp)
                moem.shard_from_full(full)
                del full
            else:
                with torch.device(self.device):
                    moem = MoELlamaModel(mcfg, ep_group=ep_plane)
                diversify_experts(moem, cfg.seed, ep_rank=plane_rank)
            self.stage = MoEStage.from_moe_model(moem, stage_idx, n_stages)
            assert not cfg.checkpoint_activations, \
                "activation checkpointing re-runs the MoE all-to-all at " \
                "backward time — unsupported for MoE stages (roadmap)"
        elif grid is not None and grid.tp_size > 1:
            # PP x TP: slice a tensor-parallel model into stages — the
            # stage machinery is block-generic (TPBlock outputs the same
            # full-size (x, residual) pair after its row-parallel
            # all-reduce, so the P2P seam is unchanged). With
            # cfg.sequence_parallel the blocks are SPBlocks and the
            # boundary payload is seq-SHARDED (S/tp per hop).
            from ..training import build_model
            if cfg.sequence_parallel:
                from .sp import SPLlamaModel as _M
            else:
                from .tp_llama import TPLlamaModel as _M
            full = build_model(mcfg, self.device,
                               cfg.checkpoint_activations)
            tpm = _M(mcfg, group=grid.tp_group).to(
                full.embed.weight.dtype).to(self.device)
            tpm.inv_freq = full.inv_freq
            tpm.shard_from_full(full)
            del full
            if cfg.sequence_parallel:
                self.stage = SPStage.from_model(tpm, stage_idx, n_stages)
                self.stage.group = grid.tp_group
            else:
                self.stage = LlamaStage.from_model(tpm, stage_idx,
                                                   n_stages)
        else:
            self.stage = LlamaStage.from_config(mcfg, stage_idx, n_stages,
                                                device=self.device)
        # the flat store keeps parameters in bf16; boundary activations
        # travel in the same dtype
        self.stage.checkpoint_activations = cfg.checkpoint_activations
        self.act_dtype = act_dtype or torch.bfloat16
        sched_cls = {"gpipe": GPipeSchedule,
                     "1f1b": OneFOneBSchedule}[schedule]
        self.sched = sched_cls(self.stage, stage_idx, n_stages,
                               device=self.device, pp_ranks=pp_ranks)
        self.store = FlatParamStore(self.stage, device=self.device)
        # the optimizer's LOCAL-norm clip would scale each stage by its
        # own norm; the trainer applies the cross-stage global clip
        self.opt = FlatAdamW(self.store, lr=cfg.lr, betas=cfg.betas,
                             weight_decay=cfg.weight_decay,
                             clip_grad_norm=0.0)
        self._tp_spans = None
        self.tp_group = None
        if grid is not None and grid.tp_size > 1:
            from .flat import classify_spans
            named = dict(self.stage.named_parameters())
            self._tp_spans = classify_spans(
                self.store,
                lambda n: getattr(named[n], "tp_sharded", False))
            self.tp_group = grid.tp_group
        self._expert_spans = self._moe_dense_spans = None
        self._attn_shard_spans = self._moe_rep_spans = None
        if self._is_moe:
            from .flat import classify_spans
            self._expert_spans, self._moe_dense_spans = classify_spans(
                self.store, lambda n: ".experts." in n)
            if self.tp_group is not None:
                # PP x EP x TP: three clip classes (experts sharded over
                # plane x tp; attention over tp; the rest replicated)
                named = dict(self.stage.named_parameters())
                self._attn_shard_spans, _ = classify_spans(
                    self.store,
                    lambda n: getattr(named[n], "tp_sharded", False)
                    and ".experts." not in n)
                _, self._moe_rep_spans = classify_spans(
                    self.store,
                    lambda n: getattr(named[n], "tp_sharded", False)
                    or ".experts." in n)
                self._tp_spans = None   # MoE branch owns the clip
        # every pp rank of a replica draws the same stream; replicas draw
        # DISTINCT streams (dp_rank-keyed), like the DP Trainer
        self.data = make_batches(cfg, self.device, rank=self.data_rank)
        seq = cfg.seq_len
        if grid is not None and cfg.sequence_parallel:
            assert seq % grid.tp_size == 0
            seq //= grid.tp_size        # seq-sharded boundary payload
        self.hidden_shape = (cfg.micro_batch, seq, mcfg.hidden_size)
        self.step_count = 0

    def _clip_grads(self) -> None:
        """Global grad-norm clip across the pipeline: each stage's normsq
        summed over the pp group so every stage applies the IDENTICAL
        factor — matching the unsharded model's clip semantics."""
        clip = self.cfg.clip_grad_norm
        if not clip or clip <= 0:
            return
        fg = self.store.flat_grad
        if self._attn_shard_spans is not None:
            # PP x EP x TP: experts are (plane x tp)-sharded — sum their
            # normsq over BOTH; attention shards over tp; replicated once
            nsq_e = fg.new_zeros((), dtype=torch.float32)
            for s_, e_ in self._expert_spans:
                nsq_e += fg[s_:e_].float().pow(2).sum()
            if self.dp_size > 1:
                dist.all_reduce(nsq_e, group=self.dp_group)
            nsq_a = fg.new_zeros((), dtype=torch.float32)
            for s_, e_ in self._attn_shard_spans:
                nsq_a += fg[s_:e_].float().pow(2).sum()
            nsq = nsq_e + nsq_a
            dist.all_reduce(nsq, group=self.tp_group)
            for s_, e_ in self._moe_rep_spans:
                nsq += fg[s_:e_].float().pow(2).sum()
        elif self._tp_spans is not None:
            # count each tp shard once: sharded normsq summed over the tp
            # group, replicated params counted locally
            sharded, replicated = self._tp_spans
            nsq = fg.new_zeros((), dtype=torch.float32)
            for s_, e_ in sharded:
                nsq += fg[s_:e_].float().pow(2).sum()
            dist.all_reduce(nsq, group=self.tp_group)
            for s_, e_ in replicated:
                nsq += fg[s_:e_].float().pow(2).sum()
        elif self._expert_spans is not None and self.dp_size > 1:
            # PP x EP: each plane member holds distinct experts — sum
            # their normsq over the plane; dense counted once (identical
            # across the plane after the all-reduce above)
            nsq = fg.new_zeros((), dtype=torch.float32)
            for s_, e_ in self._expert_spans:
                nsq += fg[s_:e_].float().pow(2).sum()
            dist.all_reduce(nsq, group=self.dp_group)
            for s_, e_ in self._moe_dense_spans:
                nsq += fg[s_:e_].float().pow(2).sum()
        else:
            nsq = fg.float().pow(2).sum()
        if dist.is_initialized() and self.sched.n > 1:
            dist.all_reduce(nsq, group=self.pp_group)
        gnorm = float(nsq.sqrt())
        if gnorm > clip:
            fg.mul_(clip / gnorm)

    def train_step(self):
        micros = [next(self.data) for _ in range(self.cfg.grad_accum)]
        if self.cfg.warmup_steps or self.cfg.lr_decay_steps:
            from ..optim import lr_at
            self.opt.lr = lr_at(self.opt.step_count, self.cfg.lr,
                                self.cfg.warmup_steps,
                                self.cfg.lr_decay_steps, self.cfg.min_lr)
        loss = self.sched.step(micros, self.hidden_shape, self.act_dtype)
        if self.cfg.sequence_parallel and isinstance(self.stage, SPStage):
            self.stage.allreduce_sp_grads()
        if self.data_replicas > 1:
            # stage-peer gradient seam: SUM with 1/replicas pre-scale
            # (gloo has no AVG; RCCL path matches ddp.py's convention)
            fg = self.store.flat_grad
            fg.mul_(1.0 / self.data_replicas)
            if self._expert_spans is None:
                dist.all_reduce(fg, group=self.dp_group)
            else:
                # PP x EP: the plane peers hold DIFFERENT experts at the
                # same flat offsets; expert grads are already complete
                # within the plane (the backward all-to-all summed every
                # pipeline's tokens) — they sync only across edp replicas
                for s_, e_ in self._moe_dense_spans:
                    dist.all_reduce(fg[s_:e_], group=self.dense_dp_group)
                if self.edp_size > 1:
                    for s_, e_ in self._expert_spans:
                        dist.all_reduce(fg[s_:e_], group=self.edp_group)
            if loss is not None:
                loss = loss / self.data_replicas
                dist.all_reduce(loss, group=self.dense_dp_group
                                if self._expert_spans is not None
                                else self.dp_group)
        self._clip_grads()
        self.opt.step()
        self.opt.zero_grad()
        self.step_count += 1
        return loss
