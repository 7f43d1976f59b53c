"""In-pod worker entrypoint.

The operator injects the rendezvous env (controller/envinject.py); this
launcher reads it, forms the RCCL world, restores the latest checkpoint
(fault restart and elastic resize both arrive here as fresh processes with
a new TRAININGJOB_RENDEZVOUS_EPOCH / WORLD_SIZE), trains with periodic
async checkpoints, and exits 0 on completion — the exit code is the
operator's completion signal (SURVEY.md §1 'contract with the container').

    python -m trainingjob_operator_amd.launcher.main \
        --model llama3-8b --steps 1000 --ckpt-dir /ckpt
"""
from __future__ import annotations

import argparse
import logging
import os
import signal
import sys
import time

log = logging.getLogger("launcher")


def _evaluate(trainer, cfg, n_batches: int):
    """Mean loss over a HELD-OUT stream (distinct seed from training) —
    dp-averaged so every rank logs the same number. None for trainers
    without a local loss (pipeline stages)."""
    import dataclasses

    import torch
    import torch.distributed as dist

    model = getattr(trainer, "model", None)
    if model is None:
        return None
    from .data import make_batches
    eval_cfg = dataclasses.replace(cfg, seed=cfg.seed + 7777)
    rank = getattr(getattr(trainer, "topo", None), "dp_rank",
                   getattr(getattr(trainer, "topo", None), "data_rank", 0))         if hasattr(trainer, "topo") else 0
    data = make_batches(eval_cfg, trainer.device
                        if hasattr(trainer, "device") else None, rank or 0)
    was_training = model.training
    model.eval()
    total = 0.0
    with torch.no_grad():
        for _ in range(n_batches):
            tokens, targets = next(data)
            total += float(model(tokens, targets))
    if was_training:
        model.train()
    loss = total / max(n_batches, 1)
    if dist.is_initialized() and dist.get_world_size() > 1:
        t = torch.tensor(loss / dist.get_world_size())
        dist.all_reduce(t)
        loss = float(t)
    return loss


def _run_ps_mode(args) -> int:
    """Asynchronous parameter-server training: ps ranks [0, N) + worker
    ranks after, one torch.distributed world. Role/index come from the
    operator's injected env (TRAININGJOB_REPLICA_NAME/_INDEX)."""
    import torch
    import torch.distributed as dist

    from ..parallel.ps import run_role
    from ..training import TrainConfig

    n_ps = args.ps_servers
    n_workers = args.ps_workers or int(os.environ.get(
        "TRAINER_INSTANCES_NUM", "1"))
    index = int(os.environ.get("TRAININGJOB_REPLICA_INDEX",
                               os.environ.get("RANK", "0")))
    role = "pserver" if args.role.lower().startswith("ps") else "trainer"
    rank = index if role == "pserver" else n_ps + index
    world = n_ps + n_workers
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "23456")
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend, rank=rank, world_size=world)
    log.info("ps mode: role=%s rank=%d/%d (%d ps + %d workers)",
             role, rank, world, n_ps, n_workers)
    cfg = TrainConfig(model=args.model, micro_batch=args.micro_batch,
                      grad_accum=args.grad_accum, seq_len=args.seq_len,
                      lr=args.lr, clip_grad_norm=0.0,
                      data_path=args.data_path, data_dtype=args.data_dtype)
    try:
        run_role(cfg, role, index, n_ps, n_workers, args.steps)
    finally:
        dist.destroy_process_group()
    log.info("ps mode done (role=%s)", role)
    return 0


def _saves_ckpt(args, ctx, trainer) -> bool:
    """Who writes checkpoints: DP -> rank 0; ZeRO-1 -> every rank (each
    owns a moment shard); PP -> every stage's dp-replica 0; EP -> every
    expert shard's edp-replica 0; TP -> every shard's dp-replica 0."""
    if args.zero1:
        return True                   # every rank owns a moment shard
    if args.pp > 1 and getattr(args, "moe_plane", 0):
        # 4-axis: every (stage, plane, tp) shard, edp-replica 0 only
        return trainer.edp_rank == 0
    if args.pp > 1:
        return trainer.dp_rank == 0   # every stage, dp-replica 0 only
    if args.ep:
        return trainer.topo.edp_rank == 0  # every shard, edp-replica 0
    if args.tp > 1:
        return trainer.topo.dp_rank == 0
    return ctx.is_rank0


def _maybe_reexec_torchrun(argv) -> bool:
    """Multi-GPU pod bootstrap: the operator injects NPROC_PER_NODE > 1
    (amd.com/gpu > 1 on the pod) plus NODE_RANK and a node-level
    WORLD_SIZE; the per-process RANK/LOCAL_RANK come from torchrun, so
    re-exec this launcher under torch.distributed.run once. Single-GPU
    pods (NPROC_PER_NODE absent or 1, LOCAL_RANK pre-set) skip this."""
    nproc = int(os.environ.get("NPROC_PER_NODE", "1") or 1)
    if nproc <= 1 or "LOCAL_RANK" in os.environ:
        return False
    nnodes = os.environ.get("TRAININGJOB_NNODES") or str(
        max(int(os.environ.get("WORLD_SIZE", nproc)) // nproc, 1))
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        f"--nnodes={nnodes}",
        f"--node-rank={os.environ.get('NODE_RANK', '0')}",
        f"--nproc-per-node={nproc}",
        f"--master-addr={os.environ.get('MASTER_ADDR', '127.0.0.1')}",
        f"--master-port={os.environ.get('MASTER_PORT', '23456')}",
        "-m", "trainingjob_operator_amd.launcher.main",
    ] + list(argv if argv is not None else sys.argv[1:])
    log.info("multi-GPU pod: re-exec under torchrun (%d procs)", nproc)
    os.execv(sys.executable, cmd)
    return True  # unreachable


def main(argv=None) -> int:
    if _maybe_reexec_torchrun(argv):
        return 0
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--steps", type=int, default=1000,
                    help="total optimizer steps (across restarts)")
    ap.add_argument("--micro-batch", type=int, default=1)
    ap.add_argument("--grad-accum", type=int, default=4)
    ap.add_argument("--seq-len", type=int, default=4096)
    ap.add_argument("--lr", type=float, default=3e-4)
    ap.add_argument("--warmup-steps", type=int, default=0,
                    help="linear LR warmup steps")
    ap.add_argument("--lr-decay-steps", type=int, default=0,
                    help="cosine-decay horizon after warmup (0 = constant)")
    ap.add_argument("--min-lr", type=float, default=0.0)
    ap.add_argument("--ckpt-dir", default=os.environ.get(
        "TRAININGJOB_CKPT_DIR", "/tmp/aitj-ckpt"))
    ap.add_argument("--ckpt-every", type=int, default=50)
    ap.add_argument("--log-every", type=int, default=10)
    ap.add_argument("--checkpoint-activations", action="store_true")
    ap.add_argument("--data-path", default=os.environ.get(
        "TRAININGJOB_DATA_PATH", ""),
        help="flat token-id binary to train on (uint16/uint32; "
             "memory-mapped); empty = synthetic data")
    ap.add_argument("--data-dtype", choices=("uint16", "uint32"),
                    default=os.environ.get("TRAININGJOB_DATA_DTYPE",
                                           "uint16"))
    ap.add_argument("--tp", type=int, default=int(os.environ.get(
        "TRAININGJOB_TP_SIZE", "1")),
        help="tensor-parallel degree (world = dp x tp; tp ranks adjacent)")
    ap.add_argument("--pp", type=int, default=int(os.environ.get(
        "TRAININGJOB_PP_SIZE", "1")),
        help="pipeline-parallel stages (world must equal pp; rank = stage)")
    ap.add_argument("--zero1", action="store_true",
                    default=os.environ.get("TRAININGJOB_ZERO1", "") == "1",
                    help="ZeRO-1: shard optimizer state across the dp "
                         "group (pure-DP jobs)")
    ap.add_argument("--vocab-parallel", action="store_true",
                    default=os.environ.get("TRAININGJOB_VOCAB_PARALLEL",
                                           "") == "1",
                    help="shard the LM head + CE over the tp group "
                         "(requires --tp > 1)")
    ap.add_argument("--cp", type=int, default=int(os.environ.get(
        "AITJ_CP_SIZE", "0")),
                    help="context parallelism: shard the sequence cp-ways "
                         "with ring attention (world = dp x cp)")
    ap.add_argument("--sp", action="store_true",
                    default=os.environ.get("TRAININGJOB_SP", "") == "1",
                    help="sequence parallelism on top of --tp (Megatron "
                         "SP: seq-sharded norms/residual)")
    ap.add_argument("--ep", type=int, default=int(os.environ.get(
        "TRAININGJOB_EP_SIZE", "0")),
        help="expert-parallel group size for MoE models (world = edp x ep; "
             "ep ranks adjacent; 0 = off)")
    ap.add_argument("--ps-servers", type=int, default=int(os.environ.get(
        "TRAININGJOB_PS_SERVERS", "0")),
        help="asynchronous parameter-server mode: N pserver ranks + the "
             "rest workers (role from TRAININGJOB_REPLICA_NAME; the "
             "reference's pserver/trainer job shape)")
    ap.add_argument("--ps-workers", type=int, default=0,
        help="worker count in ps mode (default: TRAINER_INSTANCES_NUM "
             "from the operator's env contract)")
    ap.add_argument("--role", default=os.environ.get(
        "TRAININGJOB_REPLICA_NAME", "trainer"),
        help="this pod's replica role (injected by the operator)")
    ap.add_argument("--moe-plane", type=int, default=int(os.environ.get(
        "TRAININGJOB_MOE_PLANE", "0")),
        help="expert-plane size for 4-axis MoE pipeline grids "
             "(world = edp x plane x pp x tp; requires --pp and a MoE "
             "model; 0 = plane inferred as world/(pp*tp))")
    ap.add_argument("--pp-schedule", choices=("1f1b", "gpipe"),
                    default=os.environ.get("TRAININGJOB_PP_SCHEDULE", "1f1b"),
                    help="pipeline schedule: 1f1b bounds live microbatches "
                         "per stage; gpipe holds all of them")
    ap.add_argument("--eval-every", type=int, default=int(os.environ.get(
        "TRAININGJOB_EVAL_EVERY", "0")),
        help="run a held-out eval every N steps (0 = off)")
    ap.add_argument("--eval-batches", type=int, default=8,
                    help="micro-batches per evaluation")
    ap.add_argument("--metrics-port", type=int, default=int(os.environ.get(
        "TRAININGJOB_METRICS_PORT", "0")),
        help="expose Prometheus worker metrics (tokens/s, step time, loss)")
    args = ap.parse_args(argv)

    logging.basicConfig(
        level=logging.INFO,
        format="%(asctime)s %(name)s [%(levelname)s] %(message)s")

    from ..parallel import dist_ctx
    from ..training import TrainConfig, Trainer
    from .checkpoint import Checkpointer

    if args.ps_servers:
        return _run_ps_mode(args)

    ctx = dist_ctx.from_env()
    epoch = os.environ.get("TRAININGJOB_RENDEZVOUS_EPOCH", "0")
    restart = os.environ.get("TRAININGJOB_REPLICA_RESTARTCOUNT", "0")
    log.info("rank %d/%d epoch=%s restart=%s master=%s:%s",
             ctx.rank, ctx.world_size, epoch, restart,
             ctx.master_addr, ctx.master_port)
    t_start = time.time()
    dist_ctx.init_process_group(ctx)

    if args.zero1:
        assert args.tp == 1 and args.pp == 1 and not args.ep, \
            "--zero1 launcher support is pure-DP (composed grids: roadmap)"
    if args.vocab_parallel:
        assert args.tp > 1 and args.pp == 1 and not args.ep, \
            "--vocab-parallel is a pure-TP option"
        assert not args.sp, "vocab-parallel with --sp is roadmap"
    if args.cp > 1:
        assert args.tp == 1 and args.pp == 1 and not args.ep \
            and not args.zero1, \
            "--cp composes with dp only (tp/pp/ep grids: roadmap)"
        assert args.seq_len % args.cp == 0, "--cp needs seq_len % cp == 0"
    if args.sp:
        assert args.tp > 1, "--sp requires --tp > 1"
        assert args.seq_len % args.tp == 0, "--sp needs seq_len % tp == 0"
        assert not args.ep, "--sp with --ep is not supported"
    cfg = TrainConfig(
        model=args.model, micro_batch=args.micro_batch,
        grad_accum=args.grad_accum, seq_len=args.seq_len, lr=args.lr,
        checkpoint_activations=args.checkpoint_activations,
        tp_size=args.tp, sequence_parallel=args.sp, zero1=args.zero1,
        vocab_parallel=args.vocab_parallel,
        data_path=args.data_path, data_dtype=args.data_dtype,
        warmup_steps=args.warmup_steps, lr_decay_steps=args.lr_decay_steps,
        min_lr=args.min_lr)
    if args.pp > 1:
        assert not args.ep, \
            "--pp with a MoE model shards experts automatically: the " \
            "grid's dp axis IS the EP plane (drop --ep)"
        assert ctx.world_size % (args.pp * max(args.tp, 1)) == 0, \
            f"world {ctx.world_size} not divisible by " \
            f"pp*tp={args.pp * max(args.tp, 1)}"
        from ..parallel.pp import PPTrainer
        if args.moe_plane:
            # 4-axis MoE grid: true data parallelism (edp) on top of the
            # expert plane x pp x tp
            from ..parallel.groups import build_moe_grid
            grid = build_moe_grid(plane_size=args.moe_plane,
                                  pp_size=args.pp,
                                  tp_size=max(args.tp, 1))
        else:
            from ..parallel.groups import build_grid
            grid = build_grid(tp_size=max(args.tp, 1), pp_size=args.pp)
        trainer = PPTrainer(cfg, schedule=args.pp_schedule, grid=grid)
        # per-(stage, tp-shard[, plane-shard]) checkpoint streams,
        # written by the dp_rank==0 (and edp_rank==0) replica
        sub = f"stage{grid.pp_rank}"
        if args.moe_plane:
            sub += f"_pl{grid.dp_rank}"
        if args.tp > 1:
            sub += f"_tp{grid.tp_rank}"
        ckpt = Checkpointer(os.path.join(args.ckpt_dir, sub))
    elif args.cp > 1:
        from ..parallel.cp import CPTrainer
        import torch
        dev = torch.device(f"cuda:{ctx.local_rank}"
                           if torch.cuda.is_available() else "cpu")
        trainer = CPTrainer(cfg, cp_size=args.cp, device=dev)
        # weights replicated everywhere: one rank-0 stream, like DP
        ckpt = Checkpointer(args.ckpt_dir)
    elif args.ep:
        from ..parallel.ep import EPTrainer
        import torch
        dev = torch.device(f"cuda:{ctx.local_rank}"
                           if torch.cuda.is_available() else "cpu")
        trainer = EPTrainer(cfg, ep_size=args.ep, device=dev,
                            tp_size=args.tp)
        # each (ep, tp) coordinate owns a distinct shard: its own stream,
        # written by the edp_rank==0 replica
        sub = (f"ep{trainer.topo.ep_rank}_tp{trainer.topo.tp_rank}"
               if args.tp > 1 else f"ep{trainer.topo.ep_rank}")
        ckpt = Checkpointer(os.path.join(args.ckpt_dir, sub))
    else:
        trainer = Trainer(cfg, ctx)
        if args.zero1 and ctx.world_size > 1:
            # every rank owns a distinct optimizer-state shard: its own
            # checkpoint stream (the full bf16 params ride each stream,
            # so params still resume anywhere — only moments are sharded)
            ckpt = Checkpointer(os.path.join(args.ckpt_dir,
                                             f"zero{ctx.rank}"))
        elif args.tp > 1:
            # each tp rank owns a distinct shard: per-tp-rank checkpoint
            # streams, written by the dp_rank==0 replica
            ckpt = Checkpointer(os.path.join(
                args.ckpt_dir, f"tp{trainer.topo.tp_rank}"))
        else:
            ckpt = Checkpointer(args.ckpt_dir)
    resumed = ckpt.load_latest(trainer)
    if resumed is not None:
        log.info("resumed from step %d (world=%d)", resumed, ctx.world_size)
    log.info("rejoin latency: %.2fs from exec to first step",
             time.time() - t_start)

    stop_requested = {"flag": False}
    ckpt_requested = {"flag": False}

    def on_term(signum, frame):
        stop_requested["flag"] = True
    signal.signal(signal.SIGTERM, on_term)

    def on_usr1(signum, frame):
        # on-demand checkpoint (e.g. before planned node maintenance):
        # kubectl exec ... -- kill -USR1 1
        ckpt_requested["flag"] = True
    signal.signal(signal.SIGUSR1, on_usr1)

    from ..utils.tracing import tracer
    trace = tracer("worker")
    metrics = None
    if args.metrics_port and ctx.is_rank0:
        from .worker_metrics import WorkerMetrics
        metrics = WorkerMetrics(args.metrics_port)
    # tp ranks and pp stages share one replica: whole-job tokens count DP
    # replicas only (EP ranks are all data workers: divisor 1)
    tokens_per_step = (cfg.tokens_per_step_per_rank()
                       * (ctx.world_size
                          // (max(args.tp, 1) * max(args.pp, 1))))
    # chaos knob: die with a retryable code at step N — exercises the
    # operator's ExitCode restart + checkpoint-resume path in-cluster
    fault_step = int(os.environ.get("AITJ_FAULT_STEP", "0"))
    t_last = time.time()
    while trainer.step_count < args.steps and not stop_requested["flag"]:
        loss = trainer.train_step()
        step = trainer.step_count
        if fault_step and step == fault_step:
            ckpt.wait()
            log.error("AITJ_FAULT_STEP=%d: injected fault, exiting 137",
                      fault_step)
            os._exit(137)
        if loss is None:  # non-last pipeline stages produce no loss
            if step % args.ckpt_every == 0:
                ckpt.save_async(trainer)
            continue
        if step % args.log_every == 0 and ctx.is_rank0:
            now_t = time.time()
            tps = tokens_per_step * args.log_every / max(now_t - t_last, 1e-9)
            t_last = now_t
            log.info("step %d loss %.4f tokens/s %.0f", step, loss.item(),
                     tps)
            trace.event("train_step", step=step, loss=round(loss.item(), 4),
                        tokens_per_sec=round(tps, 1),
                        world_size=ctx.world_size)
            if metrics:
                metrics.observe(step, loss.item(), tps)
        if args.eval_every and step % args.eval_every == 0:
            ev = _evaluate(trainer, cfg, args.eval_batches)
            if ev is not None and ctx.is_rank0:
                log.info("step %d eval_loss %.4f", step, ev)
                trace.event("eval", step=step, eval_loss=round(ev, 4))
                if metrics:
                    metrics.observe_eval(ev)
        want_ckpt = step % args.ckpt_every == 0 or ckpt_requested["flag"]
        if want_ckpt and _saves_ckpt(args, ctx, trainer):
            ckpt.save_async(trainer)
            trace.event("checkpoint", step=step,
                        on_demand=ckpt_requested["flag"])
        ckpt_requested["flag"] = False

    if _saves_ckpt(args, ctx, trainer):
        ckpt.save_async(trainer, blocking=True)
    ckpt.wait()
    dist_ctx.destroy_process_group()
    log.info("done at step %d", trainer.step_count)
    return 0


if __name__ == "__main__":
    sys.exit(main())
