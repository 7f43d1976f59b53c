#!/usr/bin/env python3
"""Generate token ids from a trained checkpoint.

    python scripts/generate.py --model llama3-8b --ckpt-dir /ckpt \
        --prompt-tokens 128000,15339,1917 --max-new-tokens 64 \
        --temperature 0.8 --top-k 40

Loads the newest full (DP) checkpoint stream, runs KV-cached decode on
cuda:0 when available (CPU otherwise), and prints the generated ids as a
comma-separated line (tokenizers live outside this image — no network).
"""
import argparse
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--ckpt-dir", default="",
                    help="checkpoint stream to load (empty: random init)")
    ap.add_argument("--prompt-tokens", required=True,
                    help="comma-separated prompt token ids")
    ap.add_argument("--max-new-tokens", type=int, default=64)
    ap.add_argument("--temperature", type=float, default=0.0)
    ap.add_argument("--top-k", type=int, default=0)
    ap.add_argument("--eos-token", type=int, default=None)
    ap.add_argument("--seed", type=int, default=None)
    args = ap.parse_args(argv)

    import torch

    from trainingjob_operator_amd.models.config import CONFIGS
    from trainingjob_operator_amd.models.generate import generate
    from trainingjob_operator_amd.training import build_model

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    cfg = CONFIGS[args.model]
    model = build_model(cfg, device)
    if args.ckpt_dir:
        from trainingjob_operator_amd.launcher.checkpoint import Checkpointer
        from trainingjob_operator_amd.parallel.flat import FlatParamStore
        store = FlatParamStore(model, device=device)
        path = Checkpointer(args.ckpt_dir).latest()
        if path is None:
            print(f"no checkpoint under {args.ckpt_dir}", file=sys.stderr)
            return 1
        state = torch.load(path, map_location="cpu", weights_only=False)
        store.load_flat_param(state["flat_param"])
        print(f"# loaded step {state['step']} from {path}", file=sys.stderr)

    ids = [int(t) for t in args.prompt_tokens.split(",") if t.strip()]
    prompt = torch.tensor([ids], dtype=torch.int64, device=device)
    out = generate(model, prompt, args.max_new_tokens,
                   temperature=args.temperature, top_k=args.top_k,
                   eos_token=args.eos_token, seed=args.seed)
    print(",".join(str(int(t)) for t in out[0][len(ids):]))
    return 0


if __name__ == "__main__":
    sys.exit(main())
