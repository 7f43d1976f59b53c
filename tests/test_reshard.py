"""Checkpoint topology resharding (launcher/reshard.py): full <-> PP
stage slices <-> TP shards, parameters + AdamW moments."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from trainingjob_operator_amd.launcher.checkpoint import Checkpointer
from trainingjob_operator_amd.launcher.reshard import (
    load_stream, reshard, stream_to_named,
)
from trainingjob_operator_amd.training import TrainConfig, Trainer


def _trained_full_ckpt(tmp_path, steps=2):
    cfg = TrainConfig(model="llama-tiny", micro_batch=1, grad_accum=2,
                      seq_len=32, lr=1e-3)
    tr = Trainer(cfg)
    for _ in range(steps):
        tr.train_step()
    full_dir = os.path.join(str(tmp_path), "full")
    Checkpointer(full_dir).save_async(tr, blocking=True)
    return cfg, tr, full_dir


def _flats(state):
    return (state["flat_param"], state["opt"]["p32"],
            state["opt"]["m"], state["opt"]["v"])


def test_full_pp_full_roundtrip(tmp_path):
    cfg, tr, full_dir = _trained_full_ckpt(tmp_path)
    pp_dir = os.path.join(str(tmp_path), "pp")
    back_dir = os.path.join(str(tmp_path), "back")
    paths = reshard("llama-tiny", full_dir, pp_dir, "full", "pp=2")
    assert len(paths) == 2
    reshard("llama-tiny", pp_dir, back_dir, "pp=2", "full")
    orig = load_stream(Checkpointer(full_dir).latest())
    back = load_stream(Checkpointer(back_dir).latest())
    assert back["step"] == orig["step"]
    for a, b in zip(_flats(orig), _flats(back)):
        assert torch.equal(a, b)
    # and the resharded-back stream loads straight into a fresh trainer
    tr2 = Trainer(cfg)
    assert Checkpointer(back_dir).load_latest(tr2) == tr.step_count
    assert torch.equal(tr2.store.flat_param, tr.store.flat_param)


def test_full_tp_full_roundtrip(tmp_path):
    cfg, tr, full_dir = _trained_full_ckpt(tmp_path)
    tp_dir = os.path.join(str(tmp_path), "tp")
    back_dir = os.path.join(str(tmp_path), "back")
    paths = reshard("llama-tiny", full_dir, tp_dir, "full", "tp=2")
    assert len(paths) == 2
    # shard streams carry distinct q/o slices but identical norms
    s0 = stream_to_named(load_stream(Checkpointer(
        os.path.join(tp_dir, "tp0")).latest()))
    s1 = stream_to_named(load_stream(Checkpointer(
        os.path.join(tp_dir, "tp1")).latest()))
    assert torch.equal(s0["final_norm_weight"]["param"],
                       s1["final_norm_weight"]["param"])
    assert not torch.equal(s0["blocks.0.attn.q_proj.weight"]["param"],
                           s1["blocks.0.attn.q_proj.weight"]["param"])
    reshard("llama-tiny", tp_dir, back_dir, "tp=2", "full")
    orig = load_stream(Checkpointer(full_dir).latest())
    back = load_stream(Checkpointer(back_dir).latest())
    for a, b in zip(_flats(orig), _flats(back)):
        assert torch.equal(a, b)


def test_pp_to_tp_cross(tmp_path):
    """pp=2 -> tp=2 via the tool chains both transforms."""
    _, _, full_dir = _trained_full_ckpt(tmp_path)
    pp_dir = os.path.join(str(tmp_path), "pp")
    tp_dir = os.path.join(str(tmp_path), "tp")
    tp_ref = os.path.join(str(tmp_path), "tpref")
    reshard("llama-tiny", full_dir, pp_dir, "full", "pp=2")
    reshard("llama-tiny", pp_dir, tp_dir, "pp=2", "tp=2")
    reshard("llama-tiny", full_dir, tp_ref, "full", "tp=2")
    for r in range(2):
        a = load_stream(Checkpointer(os.path.join(tp_dir, f"tp{r}")).latest())
        b = load_stream(Checkpointer(os.path.join(tp_ref, f"tp{r}")).latest())
        for x, y in zip(_flats(a), _flats(b)):
            assert torch.equal(x, y)


def test_cli(tmp_path):
    import subprocess
    import sys
    _, _, full_dir = _trained_full_ckpt(tmp_path)
    out = os.path.join(str(tmp_path), "out")
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(root, "scripts", "reshard_ckpt.py"),
         "--model", "llama-tiny", "--in-dir", full_dir, "--src", "full",
         "--out-dir", out, "--dst", "pp=3"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    assert len(r.stdout.strip().splitlines()) == 3


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _tp_resume_worker(rank, world, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from trainingjob_operator_amd.parallel.dist_ctx import DistContext
        ctx = DistContext(rank=rank, world_size=world, backend="gloo")
        cfg = TrainConfig(model="llama-tiny", micro_batch=1, grad_accum=2,
                          seq_len=32, lr=1e-3)
        # train the FULL model (dp2 over the world), snapshot on rank 0
        full = Trainer(cfg, ctx)
        for _ in range(2):
            full.train_step()
        full_dir = os.path.join(tmpdir, "full")
        tp_dir = os.path.join(tmpdir, "tp")
        if rank == 0:
            Checkpointer(full_dir).save_async(full, blocking=True)
            reshard("llama-tiny", full_dir, tp_dir, "full", "tp=2")
        dist.barrier()

        # resume as tp=2 from the resharded streams
        tcfg = TrainConfig(model="llama-tiny", micro_batch=1, grad_accum=2,
                           seq_len=32, lr=1e-3, tp_size=2)
        tp = Trainer(tcfg, ctx)
        step = Checkpointer(os.path.join(
            tp_dir, f"tp{tp.topo.tp_rank}")).load_latest(tp)
        assert step == full.step_count

        # same batch through both: the TP resume must reproduce the full
        # model's loss (exact reformulation up to bf16 collective rounding)
        g = torch.Generator().manual_seed(99)
        tokens = torch.randint(0, 512, (1, 32), generator=g)
        with torch.no_grad():
            lf = float(full.model(tokens, tokens))
            lt = float(tp.model(tokens, tokens))
        assert lt == pytest.approx(lf, abs=1e-2), (lt, lf)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_resharded_ckpt_resumes_as_tp(tmp_path):
    """End-to-end topology change: train full (DP), reshard to tp=2,
    resume with the TP trainer, losses match."""
    mp.spawn(_tp_resume_worker, args=(2, _free_port(), str(tmp_path)),
             nprocs=2, join=True)


def test_full_ep_full_roundtrip(tmp_path):
    """MoE: full <-> ep=2 expert-shard streams (experts renumbered,
    dense replicated)."""
    from trainingjob_operator_amd.parallel.ep import EPTrainer
    cfg = TrainConfig(model="moe-tiny", micro_batch=1, grad_accum=2,
                      seq_len=16, lr=1e-3)
    tr = EPTrainer(cfg)        # no dist -> single-process full MoE
    for _ in range(2):
        tr.train_step()
    full_dir = os.path.join(str(tmp_path), "full")
    ep_dir = os.path.join(str(tmp_path), "ep")
    back_dir = os.path.join(str(tmp_path), "back")
    Checkpointer(full_dir).save_async(tr, blocking=True)
    paths = reshard("moe-tiny", full_dir, ep_dir, "full", "ep=2")
    assert len(paths) == 2
    s0 = stream_to_named(load_stream(Checkpointer(
        os.path.join(ep_dir, "ep0")).latest()))
    s1 = stream_to_named(load_stream(Checkpointer(
        os.path.join(ep_dir, "ep1")).latest()))
    # shard 0 local expert 1 == full global expert 1; shard 1 local 0 == 2
    full_named = stream_to_named(load_stream(Checkpointer(full_dir).latest()))
    assert torch.equal(
        s0["blocks.0.moe.experts.1.gate_proj.weight"]["param"],
        full_named["blocks.0.moe.experts.1.gate_proj.weight"]["param"])
    assert torch.equal(
        s1["blocks.0.moe.experts.0.gate_proj.weight"]["param"],
        full_named["blocks.0.moe.experts.2.gate_proj.weight"]["param"])
    # router replicated on both shards
    assert torch.equal(s0["blocks.0.moe.router.weight"]["param"],
                       s1["blocks.0.moe.router.weight"]["param"])
    reshard("moe-tiny", ep_dir, back_dir, "ep=2", "full")
    orig = load_stream(Checkpointer(full_dir).latest())
    back = load_stream(Checkpointer(back_dir).latest())
    for a, b in zip(_flats(orig), _flats(back)):
        assert torch.equal(a, b)


def _zero_save_worker(rank, world, port, outdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from trainingjob_operator_amd.parallel.dist_ctx import DistContext
        ctx = DistContext(rank=rank, world_size=world, backend="gloo")
        cfg = TrainConfig(model="llama-tiny", micro_batch=1, grad_accum=2,
                          seq_len=32, lr=1e-3, zero1=True)
        tr = Trainer(cfg, ctx)
        for _ in range(2):
            tr.train_step()
        Checkpointer(os.path.join(outdir, f"zero{rank}")).save_async(
            tr, blocking=True)
        # also snapshot an UNsharded run on the same data for comparison
        if rank == 0:
            pass
    finally:
        dist.destroy_process_group()


def test_zero_streams_reshard_to_full_and_back(tmp_path):
    """zero=2 -> full -> zero=3: moments reassemble exactly and re-split
    to a new dp size (any-world ZeRO resume via the tool)."""
    out = str(tmp_path)
    mp.spawn(_zero_save_worker, args=(2, _free_port(), out), nprocs=2,
             join=True)
    full_dir = os.path.join(out, "full")
    z3_dir = os.path.join(out, "z3")
    paths = reshard("llama-tiny", out, full_dir, "zero=2", "full")
    assert len(paths) == 1
    full = load_stream(Checkpointer(full_dir).latest())
    s0 = load_stream(Checkpointer(os.path.join(out, "zero0")).latest())
    s1 = load_stream(Checkpointer(os.path.join(out, "zero1")).latest())
    n0 = s0["opt"]["p32"].numel()
    assert torch.equal(full["opt"]["p32"][:n0], s0["opt"]["p32"])
    assert torch.equal(full["opt"]["p32"][n0:n0 + s1["opt"]["p32"].numel()],
                       s1["opt"]["p32"])
    assert torch.equal(full["flat_param"], s0["flat_param"])
    # the merged stream loads into an UNsharded trainer directly
    cfg = TrainConfig(model="llama-tiny", micro_batch=1, grad_accum=2,
                      seq_len=32, lr=1e-3)
    tr = Trainer(cfg)
    assert Checkpointer(full_dir).load_latest(tr) == 2
    # and re-splits to a different dp size
    paths = reshard("llama-tiny", full_dir, z3_dir, "full", "zero=3")
    assert len(paths) == 3
    z0 = load_stream(Checkpointer(os.path.join(z3_dir, "zero0")).latest())
    m = z0["opt"]["p32"].numel()
    assert torch.equal(z0["opt"]["p32"], full["opt"]["p32"][:m])


def test_reshard_rejects_wrong_model(tmp_path):
    """Converting with the wrong --model fails loudly (name/shape
    mismatches) instead of writing a corrupt stream."""
    _, _, full_dir = _trained_full_ckpt(tmp_path)
    out = os.path.join(str(tmp_path), "out")
    with pytest.raises(Exception):
        reshard("llama-1b", full_dir, out, "full", "tp=2")
    assert not os.path.exists(os.path.join(out, "tp1"))
