"""Flat parameter/gradient store.

All trainable parameters live as views into ONE contiguous bf16 buffer, and
all gradients accumulate into ONE contiguous bf16 buffer:

  * the optimizer is a single fused HIP kernel over the flat buffers
    (ops/hip/ops.hip adamw_kernel) — no per-tensor launch storm;
  * DDP all-reduce operates on contiguous slices of the flat grad buffer —
    bucket sizes are chosen for xGMI ring bandwidth, not per-tensor shapes;
  * with 288 GB HBM3E per MI355X there is no reason to scatter state.

Parameters are laid out in REVERSE registration order so gradient buckets
become ready roughly front-to-back during backward (last layers first).
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

import torch
import torch.nn as nn

ALIGN = 64  # elements; keeps every view 16-byte aligned for bf16x8 kernels


def _aligned(n: int) -> int:
    return (n + ALIGN - 1) // ALIGN * ALIGN


class FlatParamStore:
    def __init__(self, model: nn.Module, device: Optional[torch.device] = None):
        named = [(n, p) for n, p in model.named_parameters() if p.requires_grad]
        named.reverse()  # backward-completion-friendly layout
        self.names: List[str] = [n for n, _ in named]
        self.params: List[nn.Parameter] = [p for _, p in named]
        if device is None:
            device = self.params[0].device

        offset = 0
        self.offsets: Dict[str, Tuple[int, int]] = {}
        self.shapes: Dict[str, Tuple[int, ...]] = {}
        for n, p in named:
            self.offsets[n] = (offset, p.numel())
            self.shapes[n] = tuple(p.shape)
            offset += _aligned(p.numel())
        self.total = _aligned(offset)

        self.flat_param = torch.zeros(self.total, dtype=torch.bfloat16,
                                      device=device)
        self.flat_grad = torch.zeros(self.total, dtype=torch.bfloat16,
                                     device=device)

        # move weights into the flat buffer and rebind params as views
        for n, p in named:
            off, numel = self.offsets[n]
            self.flat_param[off:off + numel].copy_(
                p.data.reshape(-1).to(torch.bfloat16))
            p.data = self.flat_param[off:off + numel].view(p.shape)

        self._grad_views: Dict[int, torch.Tensor] = {
            id(p): self.flat_grad[self.offsets[n][0]:
                                  self.offsets[n][0] + self.offsets[n][1]]
            .view(p.shape)
            for n, p in named
        }
        self._hook_handles = []
        self._ready_cb: Optional[Callable[[nn.Parameter], None]] = None
        for p in self.params:
            self._hook_handles.append(
                p.register_post_accumulate_grad_hook(self._on_grad_ready))
        self._prime_grads()

    def _prime_grads(self) -> None:
        """Preset every p.grad to its flat view so AccumulateGrad adds
        in-place into the flat buffer (no separate fold pass — measured
        3.7% of step time as elementwise adds)."""
        for p in self.params:
            p.grad = self._grad_views[id(p)]

    # -- gradient flow ----------------------------------------------------
    def _on_grad_ready(self, p: nn.Parameter) -> None:
        """Runs once per param per backward, after autograd accumulated into
        p.grad. Normally p.grad IS the flat view (primed) and there is
        nothing to do; if autograd replaced the tensor (out-of-place
        accumulation), the replacement already contains old+new, so copy it
        back and re-prime."""
        view = self._grad_views[id(p)]
        g = p.grad
        if g is not None and g.data_ptr() != view.data_ptr():
            view.copy_(g.reshape(view.shape).to(view.dtype))
            p.grad = view
        if self._ready_cb is not None:
            self._ready_cb(p)

    def on_param_grad_ready(self, cb: Optional[Callable]) -> None:
        self._ready_cb = cb

    def zero_grad(self) -> None:
        self.flat_grad.zero_()
        self._prime_grads()

    # -- conveniences -----------------------------------------------------
    def grad_view(self, name: str) -> torch.Tensor:
        off, numel = self.offsets[name]
        return self.flat_grad[off:off + numel]

    def param_view(self, name: str) -> torch.Tensor:
        off, numel = self.offsets[name]
        return self.flat_param[off:off + numel]

    def state_dict(self) -> dict:
        return {"flat_param": self.flat_param, "names": self.names,
                "offsets": self.offsets}

    def load_flat_param(self, flat: torch.Tensor) -> None:
        self.flat_param.copy_(flat.to(self.flat_param.device,
                                      self.flat_param.dtype))


def classify_spans(store: FlatParamStore, predicate):
    """Split the flat buffer into (matching, rest) span lists by
    predicate(param_name). Adjacent same-class params are merged INCLUDING
    the alignment padding between them — padding is zero-initialized and
    only param slices are ever written, so collectives/norms over it are
    harmless. Spans are [start, end) offsets into flat_param/flat_grad."""
    items = sorted(
        ((off, off + _aligned(numel), bool(predicate(name)))
         for name, (off, numel) in store.offsets.items()),
        key=lambda t: t[0])
    match, rest = [], []
    for start, end, is_match in items:
        spans = match if is_match else rest
        if spans and spans[-1][1] == start:
            spans[-1] = (spans[-1][0], end)
        else:
            spans.append((start, end))
    return match, rest
