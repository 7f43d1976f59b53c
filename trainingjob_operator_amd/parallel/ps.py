"""Asynchronous parameter-server training (the reference's heritage: its
example jobs are pserver/trainer role pairs — example/paddle-mnist.yaml —
though the operator itself never contained the training side).

Roles over ONE torch.distributed world:
  * ps ranks [0, n_ps): each owns a contiguous chunk of the flat
    parameter space plus that chunk's AdamW state, and serves
    request/reply: recv a gradient chunk from ANY worker, apply the
    update, send the fresh parameters back.
  * worker ranks [n_ps, world): compute grads on their own batches and
    exchange per-chunk with every ps — no worker-worker synchronization,
    so stragglers never stall the fleet (bounded-staleness async SGD).

Protocol per (worker, ps) exchange:
    worker -> ps : header [1] int64 (1 = grad follows, 0 = worker done)
    worker -> ps : grad chunk (fp32)
    ps -> worker : updated param chunk (fp32)

The ps exits when every worker has sent a done-header. gloo's
recv-anysource backs the CPU tests; the same calls run over RCCL p2p on
MI355X. Collective DP (ddp.py) remains the throughput path — PS mode
exists for the heterogeneous/elastic-worker topologies the reference's
lineage targeted (workers can join/leave between steps without a world
restart).
"""
from __future__ import annotations

from typing import List, Tuple

import torch
import torch.distributed as dist

from ..models.config import CONFIGS
from .flat import FlatParamStore, _aligned


def chunk_bounds(total: int, n_ps: int) -> List[Tuple[int, int]]:
    chunk = _aligned(-(-total // n_ps))
    return [(min(i * chunk, total), min((i + 1) * chunk, total))
            for i in range(n_ps)]


class PSServer:
    """One parameter-server rank: owns chunk `ps_index` of the flat
    space + its AdamW state."""

    def __init__(self, cfg, ps_index: int, n_ps: int, n_workers: int,
                 device=None):
        from ..training import build_model
        self.cfg = cfg
        self.device = torch.device(device or "cpu")
        torch.manual_seed(cfg.seed)       # same init as every worker
        model = build_model(CONFIGS[cfg.model], self.device)
        store = FlatParamStore(model, device=self.device)
        self.n_workers = n_workers
        s, e = chunk_bounds(store.total, n_ps)[ps_index]
        self.start, self.end = s, e
        self.params = store.flat_param[s:e].float()
        self.m = torch.zeros_like(self.params)
        self.v = torch.zeros_like(self.params)
        self.step = 0
        self.updates_served = 0
        del model, store                  # only the chunk lives here

    def _apply(self, grad: torch.Tensor) -> None:
        # exactly ops/reference.py adamw_step (decoupled weight decay),
        # so a 1-ps/1-worker run is BIT-identical to the local FlatAdamW
        c = self.cfg
        self.step += 1
        beta1, beta2 = c.betas
        self.m.mul_(beta1).add_(grad, alpha=1 - beta1)
        self.v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
        bc1 = 1 - beta1 ** self.step
        bc2 = 1 - beta2 ** self.step
        mhat = self.m / bc1
        vhat = self.v / bc2
        self.params.add_(-c.lr * (mhat / (vhat.sqrt() + 1e-8)
                                  + c.weight_decay * self.params))

    def serve(self) -> int:
        """Blocking request loop; returns updates served."""
        n = self.end - self.start
        header = torch.zeros(1, dtype=torch.int64)
        grad = torch.zeros(n, dtype=torch.float32)
        done = 0
        while done < self.n_workers:
            src = dist.recv(header, src=None)
            if int(header) == 0:
                done += 1
                continue
            dist.recv(grad, src=src)
            self._apply(grad)
            dist.send(self.params, dst=src)
            self.updates_served += 1
        return self.updates_served


class PSWorker:
    """One worker rank: local fwd/bwd, per-chunk exchange with every ps."""

    def __init__(self, cfg, worker_index: int, n_ps: int, device=None):
        from ..launcher.data import make_batches
        from ..training import build_model
        self.cfg = cfg
        self.device = torch.device(device or "cpu")
        torch.manual_seed(cfg.seed)
        self.model = build_model(CONFIGS[cfg.model], self.device)
        self.store = FlatParamStore(self.model, device=self.device)
        self.bounds = chunk_bounds(self.store.total, n_ps)
        self.n_ps = n_ps
        self.data = make_batches(cfg, self.device, rank=worker_index)
        self.step_count = 0

    def train_step(self):
        cfg = self.cfg
        loss = None
        for _ in range(cfg.grad_accum):
            tokens, targets = next(self.data)
            loss = self.model(tokens, targets)
            (loss / cfg.grad_accum).backward()
        header = torch.ones(1, dtype=torch.int64)
        for p, (s, e) in enumerate(self.bounds):
            if e <= s:
                continue
            dist.send(header, dst=p)
            dist.send(self.store.flat_grad[s:e].float(), dst=p)
            fresh = torch.zeros(e - s, dtype=torch.float32)
            dist.recv(fresh, src=p)
            self.store.flat_param[s:e].copy_(fresh.to(torch.bfloat16))
        self.store.zero_grad()
        self.step_count += 1
        return loss.detach()

    def finish(self) -> None:
        header = torch.zeros(1, dtype=torch.int64)
        for p in range(self.n_ps):
            dist.send(header, dst=p)


def run_role(cfg, role: str, index: int, n_ps: int, n_workers: int,
             steps: int, device=None):
    """Entry used by the launcher for pserver/trainer role pairs."""
    if role == "pserver":
        server = PSServer(cfg, index, n_ps, n_workers, device)
        return server.serve()
    worker = PSWorker(cfg, index, n_ps, device)
    loss = None
    while worker.step_count < steps:
        loss = worker.train_step()
    worker.finish()
    return loss
