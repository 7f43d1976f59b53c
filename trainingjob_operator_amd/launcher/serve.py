"""HTTP inference server over the KV-cached generation path.

The reference operator manages training workloads only and ships no
serving component; this gives the framework's model
families an OpenAI-ish token-in/token-out endpoint on an MI355X node:

    python -m trainingjob_operator_amd.launcher.serve \
        --model llama3-8b --port 8000 [--ckpt-dir DIR] [--graph]

The API is deliberately token-level (``prompt_tokens`` → ``tokens``):
tokenizers are workload assets the cluster may not have offline, and the
serving hot path — decode GEMV, fused flash-decoding attention, the
optional hipGraph-captured step (AITJ_DECODE_GRAPH=1 via --graph) — is
independent of them. Requests are served one at a time from a worker
thread (decode saturates HBM bandwidth at batch 1; request-level
batching happens by sending multiple sequences in one request).

Endpoints:
    GET  /healthz   liveness + model name
    GET  /info      model config, device, decode-path flags
    POST /generate  {"prompt_tokens": [[int]], "max_new_tokens": int,
                     "temperature": float, "top_k": int,
                     "eos_token": int|null, "seed": int|null}
                 -> {"tokens": [[int]], "decode_tok_s": float}
"""
# NB: no `from __future__ import annotations` here — FastAPI resolves
# endpoint type hints at runtime, and the locally-scoped request model
# would become an unresolvable string under postponed annotations.
import argparse
import threading
import time
from typing import List, Optional

import torch


def create_app(model, model_name: str = "model"):
    """Build the FastAPI app around an already-constructed model (tests
    pass a CPU llama-tiny; the CLI builds the named config on cuda:0)."""
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    from ..models.generate import generate

    class GenerateRequest(BaseModel):
        prompt_tokens: List[List[int]]
        max_new_tokens: int = 64
        temperature: float = 0.0
        top_k: int = 0
        eos_token: Optional[int] = None
        seed: Optional[int] = None
        # greedy batch-1 only: propose continuations from earlier n-gram
        # matches in the context, verify in one forward (output identical
        # to plain greedy; wins on repetitive text)
        prompt_lookup: int = 0

    app = FastAPI(title="trainingjob-operator-amd serving", version="1.0")
    lock = threading.Lock()          # one generation at a time per GPU
    device = next(model.parameters()).device
    sessions = {}                    # batch -> DecodeSession (graph mode)
    try:                             # prometheus (same registry the
        from prometheus_client import (  # operator's metrics use)
            Counter, Histogram, generate_latest,
        )
        mx = {
            "requests": Counter("aitj_serve_requests_total",
                                "Generation requests", ["status"]),
            "tokens": Counter("aitj_serve_tokens_total",
                              "Tokens decoded"),
            "latency": Histogram("aitj_serve_request_seconds",
                                 "Request wall time",
                                 buckets=(.05, .2, .5, 1., 2., 5., 15.)),
        }
    except Exception:                # prometheus_client optional
        mx, generate_latest = None, None

    def _session(batch: int, need_len: int):
        """Reuse a captured-graph session per batch size (the capture is
        position-independent; see models/decode_graph.DecodeSession).
        Returns None when the graph path is off or doesn't fit."""
        import os
        if os.environ.get("AITJ_DECODE_GRAPH") != "1" \
                or device.type != "cuda":
            return None
        ses = sessions.get(batch)
        if ses is not None and need_len <= ses.max_len:
            return ses
        if ses is not None:          # longer request: rebuild bigger
            del sessions[batch]
        if len(sessions) < 4:
            from ..models.decode_graph import DecodeSession
            max_len = max(need_len + 8, 2048)
            ses = DecodeSession(model, batch, max_len)
            sessions[batch] = ses
            return ses
        return None                  # session pool full: eager path

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "model": model_name}

    @app.get("/info")
    def info():
        import os
        cfg = model.cfg
        return {
            "model": model_name, "device": str(device),
            "vocab_size": cfg.vocab_size, "num_layers": cfg.num_layers,
            "hidden_size": cfg.hidden_size,
            "graph_decode": os.environ.get("AITJ_DECODE_GRAPH") == "1",
        }

    @app.get("/metrics")
    def metrics():
        from fastapi.responses import PlainTextResponse
        if generate_latest is None:
            raise HTTPException(404, "prometheus_client not installed")
        return PlainTextResponse(generate_latest().decode())

    @app.post("/generate")
    def gen(req: GenerateRequest):
        if mx:
            mx["requests"].labels(status="received").inc()
        if not req.prompt_tokens or not all(req.prompt_tokens):
            raise HTTPException(400, "prompt_tokens must be non-empty")
        lens = {len(p) for p in req.prompt_tokens}
        if len(lens) != 1:
            raise HTTPException(
                400, "all prompts in a batch must share a length "
                     "(pad on the client)")
        if req.max_new_tokens < 1 or req.max_new_tokens > 8192:
            raise HTTPException(400, "max_new_tokens out of range")
        vocab = model.cfg.vocab_size
        if any(t < 0 or t >= vocab for p in req.prompt_tokens for t in p):
            raise HTTPException(400, f"token id out of range [0,{vocab})")
        prompt = torch.tensor(req.prompt_tokens, dtype=torch.long,
                              device=device)
        with lock:
            t0 = time.perf_counter()
            use_lookup = (req.prompt_lookup > 0 and req.temperature <= 0
                          and prompt.shape[0] == 1)
            ses = None if use_lookup else _session(
                prompt.shape[0], prompt.shape[1] + req.max_new_tokens)
            if use_lookup:
                from ..models.generate import generate_lookup
                out = generate_lookup(model, prompt, req.max_new_tokens,
                                      lookup_k=req.prompt_lookup,
                                      eos_token=req.eos_token)
            elif ses is not None:
                out = ses.generate(prompt, req.max_new_tokens,
                                   temperature=req.temperature,
                                   top_k=req.top_k,
                                   eos_token=req.eos_token, seed=req.seed)
            else:
                out = generate(model, prompt,
                               max_new_tokens=req.max_new_tokens,
                               temperature=req.temperature,
                               top_k=req.top_k,
                               eos_token=req.eos_token, seed=req.seed)
            if device.type == "cuda":
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
        n_new = out.shape[1] - prompt.shape[1]
        if mx:
            mx["requests"].labels(status="ok").inc()
            mx["tokens"].inc(out.shape[0] * n_new)
            mx["latency"].observe(dt)
        return {"tokens": out.tolist(),
                "decode_tok_s": round(out.shape[0] * n_new / dt, 1)}

    return app


def main(argv=None):
    import os

    import uvicorn

    from ..models.config import CONFIGS
    from ..models.generate import build_inference_model

    ap = argparse.ArgumentParser()
    ap.add_argument("--model", required=True, choices=sorted(CONFIGS))
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--device", default="cuda:0")
    ap.add_argument("--ckpt-dir", default=None,
                    help="load latest checkpoint weights (else random "
                         "init — benchmarking mode)")
    ap.add_argument("--graph", action="store_true",
                    help="hipGraph-captured decode (AITJ_DECODE_GRAPH=1)")
    args = ap.parse_args(argv)
    if args.graph:
        os.environ["AITJ_DECODE_GRAPH"] = "1"
    device = torch.device(args.device)
    model = build_inference_model(CONFIGS[args.model], device)
    if args.ckpt_dir:
        from .checkpoint import load_model_only
        step = load_model_only(args.ckpt_dir, model)
        print(f"loaded checkpoint step {step} from {args.ckpt_dir}")
    app = create_app(model, args.model)
    uvicorn.run(app, host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
