"""Distributed bootstrap from the operator's injected env contract.

Reads both the torch-style env (MASTER_ADDR/MASTER_PORT/WORLD_SIZE/RANK/
LOCAL_RANK — the MI355X-native extension this operator injects) and falls
back to the reference contract (TRAININGJOB_REPLICA_INDEX + {RT}_HOSTS,
reference pkg/controller/pod.go:548-652) so workloads run under either.
"""
from __future__ import annotations

import datetime
import os
from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class DistContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    master_addr: str = "127.0.0.1"
    master_port: int = 23456
    backend: str = "gloo"

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    @property
    def is_rank0(self) -> bool:
        return self.rank == 0


def from_env() -> DistContext:
    env = os.environ
    rank = int(env.get("RANK", env.get("TRAININGJOB_REPLICA_INDEX", 0)))
    world = env.get("WORLD_SIZE")
    if world is None:
        # reference contract: count instances of this replica role
        rt = env.get("TRAININGJOB_REPLICA_NAME", "").upper()
        world = env.get(f"{rt}_INSTANCES_NUM", "1") if rt else "1"
    world_size = int(world)
    local_rank = int(env.get("LOCAL_RANK", rank % max(torch.cuda.device_count(), 1)
                             if torch.cuda.is_available() else rank))
    master_addr = env.get("MASTER_ADDR")
    master_port = int(env.get("MASTER_PORT", 23456))
    if master_addr is None:
        rt = env.get("TRAININGJOB_REPLICA_NAME", "")
        insts = env.get(f"{rt.upper()}_INSTANCES", "")
        master_addr = insts.split(",")[0] if insts else "127.0.0.1"
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    return DistContext(rank=rank, world_size=world_size, local_rank=local_rank,
                       master_addr=master_addr, master_port=master_port,
                       backend=backend)


def init_process_group(ctx: DistContext,
                       timeout_s: float = 600.0) -> DistContext:
    """init torch.distributed (backend "nccl" IS RCCL on ROCm).

    A 1-rank world launched under torchrun (RANK present in the env) still
    initializes the process group: single-rank RCCL communicator init +
    barriers are the smallest on-silicon proof of the collective path, and
    it keeps the torchrun-launched code path identical at every N."""
    if ctx.world_size <= 1 and os.environ.get("RANK") is None:
        return ctx
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", ctx.master_addr)
        os.environ.setdefault("MASTER_PORT", str(ctx.master_port))
        dist.init_process_group(
            backend=ctx.backend, rank=ctx.rank, world_size=ctx.world_size,
            timeout=datetime.timedelta(seconds=timeout_s))
    if torch.cuda.is_available():
        torch.cuda.set_device(ctx.local_rank)
    return ctx


def destroy_process_group() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
