"""Context parallelism (ring attention, parallel/cp.py) over gloo: the
sequence-sharded model must match the unsharded LlamaModel — global
loss AND every replicated weight's gradient — and the ring attention
primitive must match plain causal attention fwd+bwd."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _ring_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from trainingjob_operator_amd.parallel.cp import ring_attention
        torch.manual_seed(5)
        B, H, S, D = 2, 3, 16, 8
        Sb = S // world
        q = torch.randn(B, H, S, D)
        k = torch.randn(B, H, S, D)
        v = torch.randn(B, H, S, D)
        dy = torch.randn(B, H, S, D)

        sl = slice(rank * Sb, (rank + 1) * Sb)
        qs = q[:, :, sl].clone().requires_grad_()
        ks = k[:, :, sl].clone().requires_grad_()
        vs = v[:, :, sl].clone().requires_grad_()
        o = ring_attention(qs, ks, vs, None)
        (o * dy[:, :, sl]).sum().backward()

        qr = q.clone().requires_grad_()
        kr = k.clone().requires_grad_()
        vr = v.clone().requires_grad_()
        scale = D ** -0.5
        s = (qr @ kr.transpose(-1, -2)) * scale
        mask = torch.ones(S, S, dtype=torch.bool).triu(1)
        p = torch.softmax(s.masked_fill(mask, float("-inf")), dim=-1)
        (((p @ vr) * dy).sum()).backward()

        assert torch.allclose(o, (p @ vr)[:, :, sl], atol=1e-5), "fwd"
        assert torch.allclose(qs.grad, qr.grad[:, :, sl], atol=1e-5), "dq"
        assert torch.allclose(ks.grad, kr.grad[:, :, sl], atol=1e-5), "dk"
        assert torch.allclose(vs.grad, vr.grad[:, :, sl], atol=1e-5), "dv"
        results[rank] = True
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4])
def test_ring_attention_matches_full(world):
    results = mp.Manager().dict()
    mp.spawn(_ring_worker, args=(world, _free_port(), results),
             nprocs=world, join=True)
    assert len(results) == world


def _cp_model_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from trainingjob_operator_amd.models.config import CONFIGS
        from trainingjob_operator_amd.models.llama import LlamaModel
        from trainingjob_operator_amd.parallel.cp import CPLlamaModel
        cfg = CONFIGS["llama-tiny"]
        torch.manual_seed(17)
        full = LlamaModel(cfg).float()
        cp = CPLlamaModel(cfg).float()
        cp.shard_from_full(full)

        g = torch.Generator().manual_seed(23)
        tokens = torch.randint(0, cfg.vocab_size, (2, 32), generator=g)
        targets = torch.randint(0, cfg.vocab_size, (2, 32), generator=g)

        loss = cp(tokens, targets)
        loss.backward()
        cp.allreduce_cp_grads()

        ref_loss = full(tokens, targets)
        ref_loss.backward()

        assert torch.allclose(loss, ref_loss, atol=1e-4), \
            (float(loss), float(ref_loss))
        ref = dict(full.named_parameters())
        for name, p in cp.named_parameters():
            assert p.grad is not None, name
            assert torch.allclose(p.grad, ref[name].grad, atol=1e-4,
                                  rtol=1e-3), name
        results[rank] = float(loss)
    finally:
        dist.destroy_process_group()


def test_cp_llama_matches_unsharded():
    world = 2
    results = mp.Manager().dict()
    mp.spawn(_cp_model_worker, args=(world, _free_port(), results),
             nprocs=world, join=True)
    assert len(results) == world
    # every rank reports the identical global loss
    assert abs(results[0] - results[1]) < 1e-6


def _cp_trainer_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from trainingjob_operator_amd.parallel.cp import CPTrainer
        from trainingjob_operator_amd.training import (
            TrainConfig, Trainer,
        )
        cfg = TrainConfig(model="llama-tiny", micro_batch=2, grad_accum=2,
                          seq_len=32, lr=1e-3)
        tr = CPTrainer(cfg, cp_size=world, device="cpu")
        losses = [float(tr.train_step()) for _ in range(3)]
        assert all(l == l for l in losses), losses
        # params stay identical across the cp group after updates
        fp = tr.store.flat_param
        ref = fp.clone()
        dist.broadcast(ref, src=0)
        assert torch.equal(fp, ref), "cp replicas diverged"
        results[rank] = losses
    finally:
        dist.destroy_process_group()


def test_cp_trainer_steps_and_stays_in_sync():
    world = 2
    results = mp.Manager().dict()
    mp.spawn(_cp_trainer_worker, args=(world, _free_port(), results),
             nprocs=world, join=True)
    assert len(results) == world
    assert results[0] == results[1]      # same global loss both ranks


def _dpcp_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from trainingjob_operator_amd.parallel.cp import CPTrainer
        from trainingjob_operator_amd.training import TrainConfig
        cfg = TrainConfig(model="llama-tiny", micro_batch=2, grad_accum=1,
                          seq_len=32, lr=1e-3)
        tr = CPTrainer(cfg, cp_size=2, device="cpu")   # world 4 = dp2 x cp2
        assert tr.topo.dp_size == 2 and tr.topo.cp_size == 2
        losses = [float(tr.train_step()) for _ in range(2)]
        # params bit-identical on EVERY rank (dp mean + cp sum both in
        # the one world all-reduce)
        fp = tr.store.flat_param
        ref = fp.clone()
        dist.broadcast(ref, src=0)
        assert torch.equal(fp, ref), "dp x cp replicas diverged"
        results[rank] = (tr.topo.dp_rank, losses)
    finally:
        dist.destroy_process_group()


def test_cp_composes_with_dp():
    world = 4
    results = mp.Manager().dict()
    mp.spawn(_dpcp_worker, args=(world, _free_port(), results),
             nprocs=world, join=True)
    assert len(results) == world
    # cp peers (same dp row) see the same data -> identical global loss;
    # different dp rows stream different batches
    assert results[0] == results[1]
    assert results[2] == results[3]
    assert results[0][1] != results[2][1]


def test_cp_checkpoint_is_a_dp_checkpoint(tmp_path):
    """CP replicates weights with the same module names/order as the
    dense model, so its flat checkpoint IS a DP checkpoint: save from
    CPTrainer, resume in the plain Trainer (and back) with no reshard."""
    import torch as t
    from trainingjob_operator_amd.launcher.checkpoint import Checkpointer
    from trainingjob_operator_amd.parallel.cp import CPTrainer
    from trainingjob_operator_amd.training import TrainConfig, Trainer
    cfg = TrainConfig(model="llama-tiny", micro_batch=1, grad_accum=1,
                      seq_len=32, lr=1e-3)
    cp = CPTrainer(cfg, device="cpu")
    cp.train_step()
    Checkpointer(str(tmp_path)).save_async(cp, blocking=True)
    dense = Trainer(cfg, device=t.device("cpu"))
    assert Checkpointer(str(tmp_path)).load_latest(dense) == 1
    assert t.equal(dense.store.flat_param, cp.store.flat_param)
    dense.train_step()          # resumed state actually trains
