#!/usr/bin/env python3
"""Within-box A/B of GEMV variants (gemv_bf16_ab) on the llama3-8b decode
shapes, plus hipBLASLt (F.linear) for reference. Effective TB/s = W bytes
/ time."""
import ctypes
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from trainingjob_operator_amd.ops import native

VARIANTS = {0: "unroll4 base", 1: "unroll8", 2: "unroll8+nt",
            3: "wavewg+u8+nt"}


def main():
    lib = native.load(require=True)
    lib.gemv_bf16_ab.restype = ctypes.c_int
    lib.gemv_bf16_ab.argtypes = [ctypes.c_void_p, ctypes.c_int] + \
        [ctypes.c_void_p] * 3 + [ctypes.c_int] * 2
    st = torch.cuda.current_stream().cuda_stream
    shapes = [("qkv", 6144, 4096), ("o", 4096, 4096),
              ("gate_up", 28672, 4096), ("down", 4096, 14336),
              ("lm_head", 128256, 4096)]
    torch.manual_seed(0)
    for name, M, K in shapes:
        w = (torch.randn(M, K, device="cuda") * .1).bfloat16()
        x = (torch.randn(1, K, device="cuda") * .1).bfloat16()
        y = torch.empty(1, M, device="cuda", dtype=torch.bfloat16)
        gb = M * K * 2 / 1e9
        res = []
        for which, label in VARIANTS.items():
            def run():
                assert lib.gemv_bf16_ab(st, which, w.data_ptr(),
                                        x.data_ptr(), y.data_ptr(),
                                        M, K) == 0
            for _ in range(5):
                run()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(50):
                run()
            torch.cuda.synchronize()
            us = (time.perf_counter() - t0) / 50 * 1e6
            res.append(f"{label} {us:.1f}us {gb/us*1e3:.1f}TB/s")
            ref = (x.float() @ w.float().t())
            assert (y.float() - ref).abs().max() < \
                ref.abs().max() * 2e-2 + 2e-2, (name, which)
        # hipBLASLt
        def runl():
            torch.nn.functional.linear(x, w)
        for _ in range(5):
            runl()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(50):
            runl()
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / 50 * 1e6
        res.append(f"blaslt {us:.1f}us {gb/us*1e3:.1f}TB/s")
        print(f"{name} [{M}x{K}]: " + " | ".join(res))


if __name__ == "__main__":
    main()
