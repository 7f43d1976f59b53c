"""Fused flat-buffer AdamW for MI355X.

One kernel launch updates every parameter: fp32 master weights + moments
live in three flat fp32 buffers; the bf16 compute copy is written back in
the same pass. Optional global-norm clipping is fused (the norm is computed
on-device and consumed by the update kernel — no host sync anywhere in the
step). CPU path uses the equivalent reference implementation.
"""
from __future__ import annotations

from typing import Optional

import torch

from .ops import native, reference
from .parallel.flat import FlatParamStore

_N_PARTIALS = 2048


class FlatAdamW:
    def __init__(self, store: FlatParamStore, lr: float = 3e-4,
                 betas=(0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.1, clip_grad_norm: float = 0.0):
        self.store = store
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.clip = clip_grad_norm
        self.step_count = 0
        dev = store.flat_param.device
        self.p32 = store.flat_param.to(torch.float32)
        self.m = torch.zeros_like(self.p32)
        self.v = torch.zeros_like(self.p32)
        if dev.type == "cuda":
            self._partials = torch.empty(_N_PARTIALS, dtype=torch.float32,
                                         device=dev)
            self._normsq = torch.empty(1, dtype=torch.float32, device=dev)
        else:
            self._partials = None
            self._normsq = None

    @torch.no_grad()
    def step(self, grad_pre_scale: float = 1.0) -> None:
        self.step_count += 1
        st = self.store
        n = st.total
        bc1 = 1.0 - self.beta1 ** self.step_count
        bc2 = 1.0 - self.beta2 ** self.step_count
        if st.flat_param.is_cuda:
            lib = native.load(require=True)
            sp = native.stream_ptr()
            normsq_ptr = None
            if self.clip > 0.0:
                lib.l2normsq(sp, st.flat_grad.data_ptr(), n,
                             self._partials.data_ptr(), _N_PARTIALS,
                             self._normsq.data_ptr())
                normsq_ptr = self._normsq.data_ptr()
            lib.adamw_step(sp, self.p32.data_ptr(), self.m.data_ptr(),
                           self.v.data_ptr(), st.flat_grad.data_ptr(),
                           st.flat_param.data_ptr(), normsq_ptr, n, self.lr,
                           self.beta1, self.beta2, self.eps,
                           self.weight_decay, bc1, bc2, self.clip,
                           grad_pre_scale)
        else:
            normsq = None
            if self.clip > 0.0:
                normsq = st.flat_grad.float().pow(2).sum()
            reference.adamw_step(self.p32, self.m, self.v, st.flat_grad,
                                 st.flat_param, self.lr, self.beta1,
                                 self.beta2, self.eps, self.weight_decay,
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
        self.p32.copy_(sd["p32"])
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.step_count = int(sd["step"])
        self.store.flat_param.copy_(self.p32.to(torch.bfloat16))
