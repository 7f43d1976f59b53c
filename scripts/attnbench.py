#!/usr/bin/env python3
"""Attention fwd microbench (for timing and rocprofv3 PMC runs)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from trainingjob_operator_amd.ops.attention import (
    flash_attention_fwd_only, flash_attention_fwd_nw8, flash_attention_fwd_v5,
    flash_attention_fwd_v7)


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 30
    which = sys.argv[2] if len(sys.argv) > 2 else "native"
    B, H, S, D = 1, 32, 4096, 128
    torch.manual_seed(0)
    q = (torch.randn(B, H, S, D, device="cuda") * .5).bfloat16()
    k = (torch.randn(B, H, S, D, device="cuda") * .5).bfloat16()
    v = (torch.randn(B, H, S, D, device="cuda") * .5).bfloat16()

    def run():
        if which == "native":
            flash_attention_fwd_only(q, k, v)
        elif which == "v5":
            flash_attention_fwd_v5(q, k, v)
        elif which == "nw8":
            flash_attention_fwd_nw8(q, k, v)
        elif which == "v7":
            flash_attention_fwd_v7(q, k, v)
        else:
            F.scaled_dot_product_attention(q, k, v, is_causal=True)

    for _ in range(5):
        run()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        run()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / n * 1e6
    # causal flops: 2 gemms, half the square
    flops = 2 * 2 * B * H * S * S * D * 0.5
    print(f"{which}: {us:.1f} us/call  {flops / (us * 1e-6) / 1e12:.0f} TF")


if __name__ == "__main__":
    main()
