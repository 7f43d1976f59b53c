#!/usr/bin/env python3
"""Validate + time the hipGraph decode path (AITJ_DECODE_GRAPH=1).

Run via gpurun under an explicit `timeout`: hipGraph capture hung on a
large training step on ROCm 7.2, so the graph path is exercised here, not
in the driver-run GPU suite.

  python scripts/graphcheck.py            # correctness on llama-smoke
  python scripts/graphcheck.py bench      # llama3-8b b1/b8 decode tok/s
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def tokens_per_sec(model, batch, prompt_len, new_tokens):
    from trainingjob_operator_amd.models.config import CONFIGS
    from trainingjob_operator_amd.models.generate import generate
    from trainingjob_operator_amd.training import build_model
    cfg = CONFIGS[model]
    m = build_model(cfg, torch.device("cuda:0"))
    g = torch.Generator().manual_seed(1)
    prompt = torch.randint(0, cfg.vocab_size, (batch, prompt_len),
                           generator=g).to("cuda:0")
    generate(m, prompt, max_new_tokens=8)          # warmup (+graph capture)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = generate(m, prompt, max_new_tokens=new_tokens)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert out.shape == (batch, prompt_len + new_tokens)
    return batch * new_tokens / dt, dt * 1e3 / new_tokens


def check():
    from trainingjob_operator_amd.models.config import CONFIGS
    from trainingjob_operator_amd.models.generate import (
        build_inference_model, generate,
    )
    for name in ("llama-smoke", "moe-mid"):
        cfg = CONFIGS[name]
        m = build_inference_model(cfg, torch.device("cuda:0"))
        g = torch.Generator().manual_seed(3)
        prompt = torch.randint(0, cfg.vocab_size, (2, 16),
                               generator=g).to("cuda:0")
        os.environ["AITJ_DECODE_GRAPH"] = "0"
        ref = generate(m, prompt, max_new_tokens=24)
        os.environ["AITJ_DECODE_GRAPH"] = "1"
        got = generate(m, prompt, max_new_tokens=24)
        match = (ref == got).float().mean().item()
        print(f"{name}: token agreement graph vs eager: {match:.3f} "
              f"({ref.shape[1]} positions)")
        assert match > 0.95, (name, ref, got)
        del m
        torch.cuda.empty_cache()
    print("graphcheck OK")


def bench():
    os.environ["AITJ_DECODE_GRAPH"] = "1"
    for b in (1, 8):
        tps, ms = tokens_per_sec("llama3-8b", b, 512, 128)
        print(f"llama3-8b b{b} graph decode: {tps:.1f} tok/s, "
              f"{ms:.2f} ms/token")


def session():
    """DecodeSession: capture once, serve many requests (correctness on
    llama-smoke + amortized timing on llama3-8b)."""
    from trainingjob_operator_amd.models.config import CONFIGS
    from trainingjob_operator_amd.models.decode_graph import DecodeSession
    from trainingjob_operator_amd.models.generate import (
        build_inference_model, generate,
    )
    cfg = CONFIGS["llama-smoke"]
    m = build_inference_model(cfg, torch.device("cuda:0"))
    ses = DecodeSession(m, batch=2, max_len=256)
    g = torch.Generator().manual_seed(5)
    for i in range(3):                     # 3 requests, one capture
        prompt = torch.randint(0, cfg.vocab_size, (2, 12 + i),
                               generator=g).to("cuda:0")
        got = ses.generate(prompt, max_new_tokens=16)
        os.environ["AITJ_DECODE_GRAPH"] = "0"
        ref = generate(m, prompt, max_new_tokens=16)
        match = (ref == got).float().mean().item()
        print(f"request {i}: agreement {match:.3f}")
        assert match > 0.95
    del m
    torch.cuda.empty_cache()

    cfg = CONFIGS["llama3-8b"]
    m = build_inference_model(cfg, torch.device("cuda:0"))
    ses = DecodeSession(m, batch=1, max_len=1024)
    prompt = torch.randint(0, cfg.vocab_size, (1, 512),
                           generator=g).to("cuda:0")
    ses.generate(prompt, max_new_tokens=8)           # capture + warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = ses.generate(prompt, max_new_tokens=128)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"llama3-8b b1 session decode (no capture in-request): "
          f"{128 / dt:.1f} tok/s, {dt * 1e3 / 128:.2f} ms/token")


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "bench":
        bench()
    elif len(sys.argv) > 1 and sys.argv[1] == "session":
        session()
    else:
        check()
