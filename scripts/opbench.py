#!/usr/bin/env python3
"""Per-op microbenchmarks at Llama-3-8B shapes; reports achieved GB/s against
the ~6.3 TB/s MI355X streaming ceiling (HBM-bound ops should be close)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from trainingjob_operator_amd.ops import (
    apply_rope, fused_cross_entropy, fused_rmsnorm, make_inv_freq, swiglu,
)

DEV = "cuda"


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def report(name, sec, bytes_moved):
    print(f"{name:28s} {sec * 1e6:9.1f} us  {bytes_moved / sec / 1e9:8.1f} GB/s")


def main():
    T, H, F, V = 4096, 4096, 14336, 128256
    torch.manual_seed(0)
    x = torch.randn(T, H, device=DEV).bfloat16()
    r = torch.randn(T, H, device=DEV).bfloat16()
    w = torch.randn(H, device=DEV).bfloat16()

    # rmsnorm fwd: read x,r write y,res (+w) = 4*T*H*2
    y, res = fused_rmsnorm(x, w, r, 1e-5)
    report("rmsnorm_fwd(+res)", timeit(lambda: fused_rmsnorm(x, w, r, 1e-5)),
           4 * T * H * 2)

    # rmsnorm bwd via autograd: read dy,res,(dres),w write dx + dw partials
    def bwd():
        xg = x.detach().requires_grad_()
        rg = r.detach().requires_grad_()
        yy, rr = fused_rmsnorm(xg, w, rg, 1e-5)
        torch.autograd.backward([yy, rr], [x, r])
    report("rmsnorm_fwd+bwd(+dres)", timeit(bwd), 9 * T * H * 2)

    g = torch.randn(T, F, device=DEV).bfloat16()
    u = torch.randn(T, F, device=DEV).bfloat16()
    report("swiglu_fwd", timeit(lambda: swiglu(g, u)), 3 * T * F * 2)

    q = torch.randn(T, 32, 128, device=DEV).bfloat16()
    inv_freq = make_inv_freq(128, 500000.0, device=DEV)
    report("rope_q", timeit(lambda: apply_rope(q, inv_freq, T)),
           2 * q.numel() * 2)

    logits = torch.randn(T, V, device=DEV).bfloat16()
    tgt = torch.randint(0, V, (T,), device=DEV)
    report("ce_fwd", timeit(lambda: fused_cross_entropy(logits, tgt)),
           T * V * 2)

    def ce_bwd():
        lg = logits.detach().requires_grad_()
        fused_cross_entropy(lg, tgt).sum().backward()
    report("ce_fwd+bwd", timeit(ce_bwd, iters=10), 3 * T * V * 2)

    # fused AdamW at 8B scale
    from trainingjob_operator_amd.ops.native import load, stream_ptr
    lib = load()
    n = 8_030_000_000 // 4  # quarter-size to keep alloc fast
    n -= n % 1024
    p32 = torch.zeros(n, device=DEV)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    grad = torch.zeros(n, device=DEV, dtype=torch.bfloat16)
    pb = torch.zeros(n, device=DEV, dtype=torch.bfloat16)

    def adamw():
        lib.adamw_step(stream_ptr(), p32.data_ptr(), m.data_ptr(),
                       v.data_ptr(), grad.data_ptr(), pb.data_ptr(), None, n,
                       1e-4, 0.9, 0.95, 1e-8, 0.1, 0.1, 0.05, 0.0, 1.0, None)
    report("adamw(2B params)", timeit(adamw, iters=10),
           n * (3 * 4 * 2 + 2 + 2 + 2))

    # bf16-moment variant: m/v bf16 (20 B/param/step vs 28)
    m16 = torch.zeros(n, device=DEV, dtype=torch.bfloat16)
    v16 = torch.zeros(n, device=DEV, dtype=torch.bfloat16)

    def adamw16():
        lib.adamw_step_bf16mom(stream_ptr(), p32.data_ptr(), m16.data_ptr(),
                               v16.data_ptr(), grad.data_ptr(), pb.data_ptr(),
                               None, n, 1e-4, 0.9, 0.95, 1e-8, 0.1, 0.1,
                               0.05, 0.0, 1.0, None)
    report("adamw_bf16mom(2B)", timeit(adamw16, iters=10),
           n * (2 * 4 * 2 + 2 * 2 * 2 + 2 + 2 + 2))


if __name__ == "__main__":
    sys.exit(main())
