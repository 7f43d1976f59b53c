#!/usr/bin/env python3
"""Attention fwd+bwd microbench: native kernels vs aten/aotriton."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 20
    which = sys.argv[2] if len(sys.argv) > 2 else "native"
    B, H, HKV, S, D = 2, 32, 8, 4096, 128   # the flagship mb2 shape
    torch.manual_seed(0)
    q = (torch.randn(B, H, S, D, device="cuda") * .5).bfloat16().requires_grad_()
    k = (torch.randn(B, HKV, S, D, device="cuda") * .5).bfloat16().requires_grad_()
    v = (torch.randn(B, HKV, S, D, device="cuda") * .5).bfloat16().requires_grad_()
    gout = (torch.randn(B, H, S, D, device="cuda") * .5).bfloat16()

    if which == "aten-bwd":
        os.environ["AITJ_ATTN_BWD"] = "aten"
        which = "native"

    def run():
        if which == "native":
            from trainingjob_operator_amd.ops.attention import flash_attention
            out = flash_attention(q, k, v)
        else:
            out = F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                                 enable_gqa=True)
        out.backward(gout)
        q.grad = k.grad = v.grad = None

    for _ in range(3):
        run()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        run()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / n * 1e6
    print(f"{which} fwd+bwd: {us:.1f} us/iter")


if __name__ == "__main__":
    main()
