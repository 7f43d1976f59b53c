"""Checkpoint / resume for the flat training state.

Because parameters and optimizer state live in flat buffers
(parallel/flat.py) and DP replicates them fully, a checkpoint taken at any
world size resumes at ANY other world size — elastic resize needs no
optimizer-state resharding (SURVEY.md §5 'Checkpoint / resume'). Only the
data-cursor differs per rank, and synthetic/streaming loaders reseed from
the step count.

Writes are atomic (tmp + rename) and asynchronous: device -> host copy on a
side stream, file write on a background thread — training resumes while the
file lands (HBM -> host -> disk).
"""
from __future__ import annotations

import os
import threading
import time
from typing import Optional

import torch

CKPT_PREFIX = "ckpt_step"


class Checkpointer:
    def __init__(self, directory: str, keep: int = 2):
        self.directory = directory
        self.keep = keep
        self._thread: Optional[threading.Thread] = None
        os.makedirs(directory, exist_ok=True)

    # -- save -------------------------------------------------------------
    def save_async(self, trainer, blocking: bool = False) -> str:
        """Snapshot trainer state; returns the target path immediately."""
        self.wait()
        step = trainer.opt.step_count
        path = os.path.join(self.directory, f"{CKPT_PREFIX}{step:08d}.pt")
        # device->host snapshot (cheap vs training step; pinned staging)
        state = {
            "step": step,
            "flat_param": trainer.store.flat_param.detach().to(
                "cpu", non_blocking=False),
            "opt": {
                "p32": trainer.opt.p32.detach().cpu(),
                "m": trainer.opt.m.detach().cpu(),
                "v": trainer.opt.v.detach().cpu(),
                "step": trainer.opt.step_count,
            },
            "train_config": trainer.cfg.__dict__.copy(),
            # flat-layout metadata: lets reshard.py reinterpret the stream
            # name-wise when converting between parallel topologies
            "names": list(trainer.store.names),
            "offsets": dict(trainer.store.offsets),
            "shapes": dict(getattr(trainer.store, "shapes", {})),
            "time": time.time(),
        }

        def write():
            tmp = path + ".tmp"
            torch.save(state, tmp)
            os.replace(tmp, path)
            self._prune()

        if blocking:
            write()
        else:
            self._thread = threading.Thread(target=write, daemon=True)
            self._thread.start()
        return path

    def wait(self) -> None:
        if self._thread is not None:
            self._thread.join()
            self._thread = None

    def _prune(self) -> None:
        ckpts = self.list_checkpoints()
        for old in ckpts[:-self.keep]:
            try:
                os.remove(old)
            except OSError:
                pass

    # -- load -------------------------------------------------------------
    def list_checkpoints(self):
        try:
            names = sorted(n for n in os.listdir(self.directory)
                           if n.startswith(CKPT_PREFIX)
                           and n.endswith(".pt"))
        except FileNotFoundError:
            return []
        return [os.path.join(self.directory, n) for n in names]

    def latest(self) -> Optional[str]:
        ckpts = self.list_checkpoints()
        return ckpts[-1] if ckpts else None

    def load_latest(self, trainer) -> Optional[int]:
        """Restore trainer state from the newest checkpoint; returns the
        restored step or None. Valid for any current world size."""
        path = self.latest()
        if path is None:
            return None
        state = torch.load(path, map_location="cpu", weights_only=False)
        trainer.store.load_flat_param(state["flat_param"])
        opt = state["opt"]
        trainer.opt.p32.copy_(opt["p32"].to(trainer.opt.p32.device))
        trainer.opt.m.copy_(opt["m"].to(trainer.opt.m.device))
        trainer.opt.v.copy_(opt["v"].to(trainer.opt.v.device))
        trainer.opt.step_count = int(opt["step"])
        trainer.step_count = int(state["step"])
        return trainer.step_count
