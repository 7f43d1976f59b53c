#!/usr/bin/env python3
"""Isolate which op breaks hipGraph replay at Llama-3-8B scale.
Usage: python scripts/graph_repro.py sdpa|ce|rmsnorm|mlp|attnblock|embed [S]"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from trainingjob_operator_amd.ops import (
    apply_rope, fused_cross_entropy, fused_rmsnorm, make_inv_freq, swiglu,
)

DEV = "cuda"


def capture_and_replay(fn, n_replays=6):
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(2):
            fn()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        fn()
    for i in range(n_replays):
        g.replay()
        torch.cuda.synchronize()
        print(f"replay {i} ok", flush=True)


def main():
    which = sys.argv[1]
    S = int(sys.argv[2]) if len(sys.argv) > 2 else 4096
    torch.manual_seed(0)

    if which == "sdpa":
        q = torch.randn(1, 32, S, 128, device=DEV, dtype=torch.bfloat16,
                        requires_grad=True)
        k = torch.randn(1, 32, S, 128, device=DEV, dtype=torch.bfloat16,
                        requires_grad=True)
        v = torch.randn(1, 32, S, 128, device=DEV, dtype=torch.bfloat16,
                        requires_grad=True)

        def fn():
            o = F.scaled_dot_product_attention(q, k, v, is_causal=True)
            o.sum().backward()
            q.grad = k.grad = v.grad = None
        capture_and_replay(fn)

    elif which == "ce":
        logits = torch.randn(S, 128256, device=DEV,
                             dtype=torch.bfloat16, requires_grad=True)
        targets = torch.randint(0, 128256, (S,), device=DEV)

        def fn():
            loss = fused_cross_entropy(logits, targets).sum()
            loss.backward()
            logits.grad = None
        capture_and_replay(fn)

    elif which == "rmsnorm":
        x = torch.randn(S, 4096, device=DEV, dtype=torch.bfloat16,
                        requires_grad=True)
        r = torch.randn(S, 4096, device=DEV, dtype=torch.bfloat16,
                        requires_grad=True)
        w = torch.randn(4096, device=DEV, dtype=torch.bfloat16,
                        requires_grad=True)

        def fn():
            y, res = fused_rmsnorm(x, w, r, 1e-5)
            (y.sum() + res.sum()).backward()
            x.grad = r.grad = w.grad = None
        capture_and_replay(fn)

    elif which == "mlp":
        x = torch.randn(S, 4096, device=DEV, dtype=torch.bfloat16,
                        requires_grad=True)
        wg = torch.randn(14336, 4096, device=DEV, dtype=torch.bfloat16,
                         requires_grad=True)
        wu = torch.randn(14336, 4096, device=DEV, dtype=torch.bfloat16,
                         requires_grad=True)
        wd = torch.randn(4096, 14336, device=DEV, dtype=torch.bfloat16,
                         requires_grad=True)

        def fn():
            out = F.linear(swiglu(F.linear(x, wg), F.linear(x, wu)), wd)
            out.sum().backward()
            for t in (x, wg, wu, wd):
                t.grad = None
        capture_and_replay(fn)

    elif which == "rope":
        x = torch.randn(S, 32, 128, device=DEV, dtype=torch.bfloat16,
                        requires_grad=True)
        inv_freq = make_inv_freq(128, 500000.0, device=DEV)

        def fn():
            apply_rope(x, inv_freq, S).sum().backward()
            x.grad = None
        capture_and_replay(fn)

    elif which == "sdpa_gqa":
        # smoke-model attention shape: GQA repeat_interleave + rope
        q = torch.randn(1, S, 16, 128, device=DEV, dtype=torch.bfloat16,
                        requires_grad=True)
        kv = torch.randn(1, S, 8, 128, device=DEV, dtype=torch.bfloat16,
                         requires_grad=True)
        inv_freq = make_inv_freq(128, 500000.0, device=DEV)

        def fn():
            qr = apply_rope(q, inv_freq, S).transpose(1, 2)
            kr = apply_rope(kv, inv_freq, S).transpose(1, 2)
            vr = kv.transpose(1, 2)
            kr = kr.repeat_interleave(2, dim=1)
            vr = vr.repeat_interleave(2, dim=1)
            o = F.scaled_dot_product_attention(qr, kr, vr, is_causal=True)
            o.sum().backward()
            q.grad = kv.grad = None
        capture_and_replay(fn)

    elif which == "trainer":
        from trainingjob_operator_amd.training import TrainConfig, Trainer
        model = sys.argv[3] if len(sys.argv) > 3 else "llama-smoke"
        ga = int(sys.argv[4]) if len(sys.argv) > 4 else 1
        cfg = TrainConfig(model=model, micro_batch=1, grad_accum=ga,
                          seq_len=S, use_graphs=True)
        trainer = Trainer(cfg)
        for i in range(6):
            loss = trainer.train_step()
            torch.cuda.synchronize()
            print(f"step {i} ok loss={loss.item():.3f}", flush=True)
        print(f"trainer {model} S={S} ga={ga}: ALL OK")
        return

    elif which == "embed":
        emb = torch.nn.Embedding(128256, 4096).to(DEV).bfloat16()
        tokens = torch.randint(0, 128256, (1, S), device=DEV)

        def fn():
            emb(tokens).sum().backward()
            emb.weight.grad = None
        capture_and_replay(fn)

    print(f"{which}: ALL REPLAYS OK")


if __name__ == "__main__":
    main()
