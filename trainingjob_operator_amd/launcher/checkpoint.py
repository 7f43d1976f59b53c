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


def _to_host(t: "torch.Tensor") -> "torch.Tensor":
    """Pinned-staging D2H (~2x a pageable .cpu() copy); the caller
    synchronizes once after issuing all copies."""
    if not t.is_cuda:
        return t.detach().clone()
    out = torch.empty_like(t, device="cpu", pin_memory=True)
    out.copy_(t.detach(), non_blocking=True)
    return out


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
        # The device->host snapshot is SYNCHRONOUS on the training thread
        # (pinned staging + non_blocking issue, then one synchronize): an
        # overlapped copy would race the next optimizer step mutating
        # flat_param/p32/m/v in place, and a device-side staging clone of
        # the fp32 optimizer state is too large to double-buffer at 8B+
        # scale. Only the disk write runs in the background.
        state = {
            "step": step,
            "flat_param": _to_host(trainer.store.flat_param),
            "opt": {
                "p32": _to_host(trainer.opt.p32),
                "m": _to_host(trainer.opt.m),
                "v": _to_host(trainer.opt.v),
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
        if trainer.store.flat_param.is_cuda:
            import torch as _t
            _t.cuda.synchronize()

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


def load_model_only(directory: str, model) -> Optional[int]:
    """Serving-side weight load: copy the model's named parameters out of
    the latest flat checkpoint (ignores optimizer state; works from a
    checkpoint taken at any world size because DP replicates the flat
    stream). Returns the checkpoint step, or None if none found."""
    ck = Checkpointer(directory)
    path = ck.latest()
    if path is None:
        return None
    state = torch.load(path, map_location="cpu", weights_only=False)
    flat = state["flat_param"]
    offsets, shapes = state["offsets"], state["shapes"]
    named = dict(model.named_parameters())
    with torch.no_grad():
        for name, (off, numel) in offsets.items():
            p = named.get(name)
            if p is None:
                continue
            p.copy_(flat[off:off + numel]
                    .view(shapes.get(name, p.shape)).to(p.dtype))
    return state["step"]
