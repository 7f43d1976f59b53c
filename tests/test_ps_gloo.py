"""Asynchronous parameter-server training (parallel/ps.py): 1-2 ps ranks
serving request/reply updates to async workers over gloo."""
import json
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from trainingjob_operator_amd.parallel.ps import chunk_bounds
from trainingjob_operator_amd.training import TrainConfig


def test_chunk_bounds():
    b = chunk_bounds(1000, 3)
    assert b[0][0] == 0 and b[-1][1] == 1000
    assert all(s <= e for s, e in b)
    got = sorted(b)
    for (s1, e1), (s2, e2) in zip(got, got[1:]):
        assert e1 == s2 or s2 >= e1  # contiguous, non-overlapping


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _ps_worker(rank, world, port, n_ps, steps, outdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from trainingjob_operator_amd.parallel.ps import run_role
        cfg = TrainConfig(model="llama-tiny", micro_batch=2, grad_accum=1,
                          seq_len=32, lr=2e-3)
        n_workers = world - n_ps
        if rank < n_ps:
            served = run_role(cfg, "pserver", rank, n_ps, n_workers, steps)
            with open(os.path.join(outdir, f"ps{rank}.json"), "w") as f:
                json.dump({"served": served}, f)
        else:
            # record the worker's own loss trajectory
            from trainingjob_operator_amd.parallel.ps import PSWorker
            w = PSWorker(cfg, rank - n_ps, n_ps)
            losses = [float(w.train_step()) for _ in range(steps)]
            w.finish()
            with open(os.path.join(outdir,
                                   f"w{rank - n_ps}.json"), "w") as f:
                json.dump(losses, f)
    finally:
        dist.destroy_process_group()


def _ps_parity_worker(rank, world, port, outdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from trainingjob_operator_amd.parallel.ps import PSServer, PSWorker
        cfg = TrainConfig(model="llama-tiny", micro_batch=2, grad_accum=2,
                          seq_len=32, lr=2e-3, clip_grad_norm=0.0)
        if rank == 0:
            PSServer(cfg, 0, 1, 1).serve()
            return
        w = PSWorker(cfg, 0, 1)
        losses = [float(w.train_step()) for _ in range(4)]
        w.finish()

        # local reference: same model/data, FlatAdamW without clipping
        from trainingjob_operator_amd.optim import FlatAdamW
        from trainingjob_operator_amd.parallel.flat import FlatParamStore
        from trainingjob_operator_amd.training import (
            build_model, synthetic_batches,
        )
        torch.manual_seed(cfg.seed)
        model = build_model(cfg.model_config, torch.device("cpu"))
        store = FlatParamStore(model)
        opt = FlatAdamW(store, lr=cfg.lr, betas=cfg.betas,
                        weight_decay=cfg.weight_decay, clip_grad_norm=0.0)
        data = synthetic_batches(cfg, torch.device("cpu"), rank=0)
        ref = []
        for _ in range(4):
            for _ in range(cfg.grad_accum):
                loss = model(*next(data))
                (loss / cfg.grad_accum).backward()
            ref.append(float(loss))
            opt.step()
            opt.zero_grad()
        assert losses == ref, (losses, ref)          # bit-identical path
        assert torch.equal(w.store.flat_param, store.flat_param)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_ps_single_worker_matches_local_adamw(tmp_path):
    """1 ps + 1 worker must be BIT-identical to a local FlatAdamW loop
    (same grads, the ps applies the identical decoupled-AdamW formula)."""
    mp.spawn(_ps_parity_worker, args=(2, _free_port(), str(tmp_path)),
             nprocs=2, join=True)


@pytest.mark.timeout(600)
def test_ps_async_two_workers_liveness(tmp_path):
    """1 ps + 2 async workers: all requests served, params move, losses
    stay finite (uniform-random tokens are already at the entropy floor,
    so a decrease test would be meaningless here)."""
    out = str(tmp_path)
    mp.spawn(_ps_worker, args=(3, _free_port(), 1, 8, out), nprocs=3,
             join=True)
    ps = json.load(open(os.path.join(out, "ps0.json")))
    assert ps["served"] == 2 * 8           # every step, every worker
    for w in range(2):
        losses = json.load(open(os.path.join(out, f"w{w}.json")))
        assert len(losses) == 8
        assert all(l == l and l < 20 for l in losses)


@pytest.mark.timeout(600)
def test_ps_sharded_across_two_servers(tmp_path):
    """2 ps ranks each own half the flat space; all chunk requests land."""
    out = str(tmp_path)
    mp.spawn(_ps_worker, args=(4, _free_port(), 2, 6, out), nprocs=4,
             join=True)
    total = sum(json.load(open(os.path.join(out, f"ps{p}.json")))["served"]
                for p in range(2))
    assert total == 2 * 2 * 6              # both chunks, both workers
    for w in range(2):
        losses = json.load(open(os.path.join(out, f"w{w}.json")))
        assert all(l == l and l < 20 for l in losses)


def _ps_launcher_worker(rank, world, port, outdir):
    # simulate the operator's injected env for a pserver/trainer pair job
    n_ps = 1
    role = "pserver" if rank < n_ps else "trainer"
    index = rank if rank < n_ps else rank - n_ps
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "TRAININGJOB_REPLICA_NAME": role,
        "TRAININGJOB_REPLICA_INDEX": str(index),
        "TRAINER_INSTANCES_NUM": str(world - n_ps),
    })
    os.environ.pop("RANK", None)
    os.environ.pop("WORLD_SIZE", None)
    from trainingjob_operator_amd.launcher.main import main
    rc = main(["--model", "llama-tiny", "--steps", "3", "--seq-len", "32",
               "--grad-accum", "1", "--micro-batch", "1",
               "--ps-servers", str(n_ps)])
    assert rc == 0


@pytest.mark.timeout(600)
def test_ps_launcher_role_pair(tmp_path):
    """The launcher runs the reference's pserver/trainer job shape from
    the injected env alone (--ps-servers + TRAININGJOB_REPLICA_NAME)."""
    mp.spawn(_ps_launcher_worker, args=(3, _free_port(), str(tmp_path)),
             nprocs=3, join=True)
