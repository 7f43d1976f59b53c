"""Token-file data loading: train on real corpora, not just synthetic.

The format is the conventional flat binary of token ids (uint16 for
vocab < 65536, else uint32) that tokenizer preprocessing pipelines emit
(`*.bin`). The file is MEMORY-MAPPED — a TB-scale corpus streams from
page cache without touching the 288 GB of HBM; each batch copies only
micro_batch x (seq_len+1) ids to the device.

Sampling is deterministic per (seed, rank, batch-index): every dp data
replica draws disjoint random windows, and a restarted rank re-draws the
identical sequence — fault restarts and elastic resizes (the launcher's
rendezvous epoch) stay reproducible without a sampler checkpoint.

The reference operator has no data path at all (the
training container is opaque to it); this is worker-side capability.
"""
from __future__ import annotations

import os
from typing import Iterator, Tuple

import numpy as np
import torch

DTYPES = {"uint16": np.uint16, "uint32": np.uint32}


class TokenFileDataset:
    """Random fixed-length windows over a flat token-id file."""

    def __init__(self, path: str, seq_len: int, micro_batch: int,
                 dtype: str = "uint16", seed: int = 1234, rank: int = 0):
        if dtype not in DTYPES:
            raise ValueError(f"dtype {dtype!r} not in {sorted(DTYPES)}")
        self.path = path
        self.seq_len = seq_len
        self.micro_batch = micro_batch
        self.rank = rank
        self.seed = seed
        self.tokens = np.memmap(path, dtype=DTYPES[dtype], mode="r")
        need = seq_len + 1
        if len(self.tokens) < need:
            raise ValueError(
                f"{path}: {len(self.tokens)} tokens < seq_len+1 ({need})")
        self.n_windows = len(self.tokens) - need + 1

    def __len__(self) -> int:
        return len(self.tokens)

    def batches(self, device=None) -> Iterator[
            Tuple[torch.Tensor, torch.Tensor]]:
        """Infinite (tokens, targets) stream: [B, S] int64 on `device`."""
        g = torch.Generator().manual_seed(self.seed * 1000 + self.rank)
        need = self.seq_len + 1
        while True:
            starts = torch.randint(0, self.n_windows, (self.micro_batch,),
                                   generator=g)
            rows = np.stack([np.asarray(self.tokens[s:s + need])
                             for s in starts.tolist()])
            batch = torch.from_numpy(rows.astype(np.int64))
            if device is not None:
                batch = batch.to(device, non_blocking=True)
            yield batch[:, :-1].contiguous(), batch[:, 1:].contiguous()


def write_token_file(path: str, tokens, dtype: str = "uint16") -> str:
    """Helper for tests/tools: dump token ids to the flat-binary format."""
    arr = np.asarray(tokens, dtype=DTYPES[dtype])
    arr.tofile(path)
    return path


def make_batches(cfg, device, rank: int = 0):
    """The launcher's data dispatch: a token file when cfg.data_path is
    set, else the deterministic synthetic stream (no network here)."""
    if getattr(cfg, "data_path", ""):
        if not os.path.exists(cfg.data_path):
            raise FileNotFoundError(cfg.data_path)
        ds = TokenFileDataset(cfg.data_path, cfg.seq_len, cfg.micro_batch,
                              dtype=getattr(cfg, "data_dtype", "uint16"),
                              seed=cfg.seed, rank=rank)
        return ds.batches(device)
    from ..training import synthetic_batches
    return synthetic_batches(cfg, device, rank)
