#!/usr/bin/env python3
"""KV-cached generation (serving) bench: prefill + decode tokens/s."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    model = sys.argv[1] if len(sys.argv) > 1 else "llama3-8b"
    batch = int(sys.argv[2]) if len(sys.argv) > 2 else 1
    prompt_len = int(sys.argv[3]) if len(sys.argv) > 3 else 512
    new_tokens = int(sys.argv[4]) if len(sys.argv) > 4 else 128
    from trainingjob_operator_amd.models.config import CONFIGS
    from trainingjob_operator_amd.models.generate import (
        build_inference_model, generate,
    )
    cfg = CONFIGS[model]
    m = build_inference_model(cfg, torch.device("cuda:0"))
    g = torch.Generator().manual_seed(1)
    prompt = torch.randint(0, cfg.vocab_size, (batch, prompt_len),
                           generator=g).to("cuda:0")
    # warmup
    generate(m, prompt, max_new_tokens=8)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = generate(m, prompt, max_new_tokens=new_tokens)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert out.shape == (batch, prompt_len + new_tokens)
    tps = batch * new_tokens / dt
    print(f"{model} b{batch} prompt{prompt_len} +{new_tokens}: "
          f"{dt*1e3:.0f} ms total, {tps:.1f} decode tok/s, "
          f"{dt*1e3/new_tokens:.2f} ms/token")


if __name__ == "__main__":
    main()
