"""Fused flat-buffer AdamW for MI355X.

One kernel launch updates every parameter: fp32 master weights + moments
live in three flat fp32 buffers; the bf16 compute copy is written back in
the same pass. Optional global-norm clipping is fused (the norm is computed
on-device and consumed by the update kernel — no host sync anywhere in the
step). CPU path uses the equivalent reference implementation.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from .ops import native, reference
from .parallel.flat import FlatParamStore

_N_PARTIALS = 2048


def lr_at(step: int, base_lr: float, warmup_steps: int = 0,
          decay_steps: int = 0, min_lr: float = 0.0) -> float:
    """Warmup + cosine-decay schedule (the standard LLM pretraining
    shape): linear 0 -> base over warmup_steps, cosine base -> min_lr
    over decay_steps, then flat min_lr. decay_steps == 0 disables decay
    (constant base after warmup)."""
    import math
    if warmup_steps > 0 and step < warmup_steps:
        return base_lr * (step + 1) / warmup_steps
    if decay_steps <= 0:
        return base_lr
    t = min(max(step - warmup_steps, 0), decay_steps) / decay_steps
    return min_lr + 0.5 * (base_lr - min_lr) * (1.0 + math.cos(math.pi * t))


class FlatAdamW:
    def __init__(self, store: FlatParamStore, lr: float = 3e-4,
                 betas=(0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.1, clip_grad_norm: float = 0.0,
                 shard: Optional[Tuple[int, int]] = None,
                 shard_norm_group=None, bf16_moments: bool = False):
        """shard=(start, end): ZeRO-1 — master weights and moments cover
        only that slice of the flat buffer; step() updates only the slice.
        The grad-norm clip reads the FULL gradient, UNLESS
        shard_norm_group is set (the reduce-scatter layout, where only
        the local shard holds globally-reduced values): then the norm is
        the shard normsq all-reduced over that group. The caller
        re-assembles flat_param across the dp group after the step."""
        self.store = store
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.clip = clip_grad_norm
        self.step_count = 0
        self.shard = shard
        self.shard_norm_group = shard_norm_group
        s0, s1 = shard if shard is not None else (0, store.total)
        self._s0, self._s1 = s0, s1
        dev = store.flat_param.device
        self.p32 = store.flat_param[s0:s1].to(torch.float32)
        # bf16 moments (GPU fused path): halves optimizer HBM traffic and
        # checkpoint moment size; update math stays fp32 in-kernel. The
        # CPU reference emulates the same bf16 round-trip for parity.
        self.bf16_moments = bf16_moments
        mdtype = torch.bfloat16 if bf16_moments else torch.float32
        self.m = torch.zeros(self.p32.numel(), dtype=mdtype, device=dev)
        self.v = torch.zeros_like(self.m)
        if dev.type == "cuda":
            self._partials = torch.empty(_N_PARTIALS, dtype=torch.float32,
                                         device=dev)
            self._normsq = torch.empty(1, dtype=torch.float32, device=dev)
            # bias-correction terms in device memory so a hipGraph-captured
            # step stays correct across replays (the kernel reads them)
            self._bc = torch.ones(2, dtype=torch.float32, device=dev)
        else:
            self._partials = None
            self._normsq = None
            self._bc = None

    def update_bias_correction(self, step_count: Optional[int] = None) -> None:
        """Refresh the device-side [bc1, bc2]. Called inside eager step();
        a graph-replay driver calls it (plus bump_step) BEFORE replay."""
        if self._bc is None:
            return
        t = self.step_count if step_count is None else step_count
        self._bc.copy_(torch.tensor(
            [1.0 - self.beta1 ** t, 1.0 - self.beta2 ** t],
            dtype=torch.float32))

    def bump_step(self) -> int:
        self.step_count += 1
        return self.step_count

    @torch.no_grad()
    def step(self, grad_pre_scale: float = 1.0,
             in_graph_capture: bool = False) -> None:
        """One fused update. With in_graph_capture=True the step counter and
        bc buffer are managed by the caller (bump_step +
        update_bias_correction before each replay)."""
        if not in_graph_capture:
            self.bump_step()
        st = self.store
        s0, s1 = self._s0, self._s1
        n = s1 - s0
        if n <= 0:
            return
        grad = st.flat_grad[s0:s1]
        param = st.flat_param[s0:s1]
        bc1 = 1.0 - self.beta1 ** max(self.step_count, 1)
        bc2 = 1.0 - self.beta2 ** max(self.step_count, 1)
        if st.flat_param.is_cuda:
            if not in_graph_capture:
                self.update_bias_correction()
            lib = native.load(require=True)
            sp = native.stream_ptr()
            normsq_ptr = None
            if self.clip > 0.0:
                if self.shard_norm_group is not None:
                    # RS layout: only [s0, s1) holds reduced grads — the
                    # global normsq = sum of shard normsqs over the group
                    import torch.distributed as dist
                    rc = lib.l2normsq(sp, grad.data_ptr(), n,
                                      self._partials.data_ptr(),
                                      _N_PARTIALS, self._normsq.data_ptr())
                    native.check_rc(rc, "l2normsq", f"n={n}")
                    dist.all_reduce(self._normsq,
                                    group=self.shard_norm_group)
                else:
                    # the clip norm is GLOBAL over the full gradient,
                    # even when the update covers only this rank's shard
                    rc = lib.l2normsq(sp, st.flat_grad.data_ptr(), st.total,
                                      self._partials.data_ptr(), _N_PARTIALS,
                                      self._normsq.data_ptr())
                    native.check_rc(rc, "l2normsq", f"n={st.total}")
                normsq_ptr = self._normsq.data_ptr()
            fn = (lib.adamw_step_bf16mom if self.bf16_moments
                  else lib.adamw_step)
            rc = fn(sp, self.p32.data_ptr(), self.m.data_ptr(),
                    self.v.data_ptr(), grad.data_ptr(),
                    param.data_ptr(), normsq_ptr, n, self.lr,
                    self.beta1, self.beta2, self.eps,
                    self.weight_decay, bc1, bc2, self.clip,
                    grad_pre_scale, self._bc.data_ptr())
            native.check_rc(rc, "adamw_step", f"n={n}")
        else:
            normsq = None
            if self.clip > 0.0:
                if self.shard_norm_group is not None:
                    import torch.distributed as dist
                    normsq = grad.float().pow(2).sum()
                    dist.all_reduce(normsq, group=self.shard_norm_group)
                else:
                    normsq = st.flat_grad.float().pow(2).sum()
            if self.bf16_moments:
                m32 = self.m.float()
                v32 = self.v.float()
                reference.adamw_step(self.p32, m32, v32, grad,
                                     param, self.lr, self.beta1,
                                     self.beta2, self.eps,
                                     self.weight_decay, self.step_count,
                                     self.clip, normsq, grad_pre_scale)
                self.m.copy_(m32.to(torch.bfloat16))
                self.v.copy_(v32.to(torch.bfloat16))
            else:
                reference.adamw_step(self.p32, self.m, self.v, grad,
                                     param, self.lr, self.beta1,
                                     self.beta2, self.eps,
                                     self.weight_decay,
                                     self.step_count, self.clip, normsq,
                                     grad_pre_scale)

    def zero_grad(self) -> None:
        self.store.zero_grad()

    def grad_norm(self) -> Optional[torch.Tensor]:
        """Post-clip-input grad norm (device tensor; only after a clip step)."""
        if self._normsq is None:
            return None
        return self._normsq.sqrt()

    def state_dict(self) -> dict:
        return {"p32": self.p32, "m": self.m, "v": self.v,
                "step": self.step_count}

    def load_state_dict(self, sd: dict) -> None:
        if sd["p32"].numel() != self.p32.numel():
            raise ValueError(
                f"optimizer state length {sd['p32'].numel()} does not "
                f"match this rank's {self.p32.numel()} — a ZeRO-1 "
                f"checkpoint must resume at the same dp size (or be "
                f"resharded)")
        self.p32.copy_(sd["p32"])
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.step_count = int(sd["step"])
        self.store.flat_param[self._s0:self._s1].copy_(
            self.p32.to(torch.bfloat16))
