"""KV-cached generation (models/generate.py): cache parity with the full
forward, greedy/sampled decoding, EOS handling."""
import pytest
import torch

from trainingjob_operator_amd.models.config import LLAMA_TINY
from trainingjob_operator_amd.models.generate import (
    KVCache, _forward_cached, generate,
)
from trainingjob_operator_amd.models.llama import LlamaModel


@pytest.fixture(scope="module")
def model():
    torch.manual_seed(9)
    return LlamaModel(LLAMA_TINY).eval()


def test_cached_logits_match_full_forward(model):
    g = torch.Generator().manual_seed(10)
    tokens = torch.randint(0, LLAMA_TINY.vocab_size, (2, 12), generator=g)
    with torch.no_grad():
        full = model(tokens)                       # [B, S, V]
    cache = KVCache(LLAMA_TINY, 2, 12, tokens.device, torch.float32)
    pre = _forward_cached(model, tokens[:, :8], cache)
    assert torch.allclose(pre, full[:, :8], atol=1e-4)
    # one-token decode steps continue the SAME distribution
    for i in range(8, 12):
        step = _forward_cached(model, tokens[:, i:i + 1], cache)
        assert torch.allclose(step[:, 0], full[:, i], atol=1e-4), i


def test_greedy_matches_uncached_loop(model):
    g = torch.Generator().manual_seed(11)
    prompt = torch.randint(0, LLAMA_TINY.vocab_size, (1, 6), generator=g)
    out = generate(model, prompt, max_new_tokens=6)
    assert out.shape == (1, 12)
    # naive reference: full forward per step, argmax
    ref = prompt.clone()
    with torch.no_grad():
        for _ in range(6):
            nxt = model(ref)[:, -1].argmax(dim=-1)
            ref = torch.cat([ref, nxt[:, None]], dim=1)
    assert torch.equal(out, ref)


def test_sampling_reproducible_and_topk(model):
    g = torch.Generator().manual_seed(12)
    prompt = torch.randint(0, LLAMA_TINY.vocab_size, (2, 4), generator=g)
    a = generate(model, prompt, 5, temperature=0.8, top_k=8, seed=7)
    b = generate(model, prompt, 5, temperature=0.8, top_k=8, seed=7)
    c = generate(model, prompt, 5, temperature=0.8, top_k=8, seed=8)
    assert torch.equal(a, b)
    assert not torch.equal(a, c)


def test_eos_stops_early(model):
    g = torch.Generator().manual_seed(13)
    prompt = torch.randint(0, LLAMA_TINY.vocab_size, (1, 4), generator=g)
    greedy = generate(model, prompt, 8)
    eos = int(greedy[0, 4])       # the first token it would emit
    out = generate(model, prompt, 8, eos_token=eos)
    assert out.shape[1] == 5      # stopped right after emitting EOS


def test_generate_cli(tmp_path):
    """scripts/generate.py resumes a trained checkpoint and decodes."""
    import os
    import subprocess
    import sys
    from trainingjob_operator_amd.launcher.checkpoint import Checkpointer
    from trainingjob_operator_amd.training import TrainConfig, Trainer
    cfg = TrainConfig(model="llama-tiny", micro_batch=1, grad_accum=1,
                      seq_len=16)
    tr = Trainer(cfg)
    tr.train_step()
    ck = os.path.join(str(tmp_path), "ck")
    Checkpointer(ck).save_async(tr, blocking=True)
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(root, "scripts", "generate.py"),
         "--model", "llama-tiny", "--ckpt-dir", ck,
         "--prompt-tokens", "1,2,3", "--max-new-tokens", "4"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    ids = [int(t) for t in r.stdout.strip().split(",")]
    assert len(ids) == 4 and all(0 <= t < 512 for t in ids)
    assert "loaded step 1" in r.stderr


def test_moe_generate_matches_uncached():
    from trainingjob_operator_amd.models.moe_llama import (
        MOE_TINY, MoELlamaModel,
    )
    torch.manual_seed(21)
    moe = MoELlamaModel(MOE_TINY).eval()
    g = torch.Generator().manual_seed(22)
    prompt = torch.randint(0, MOE_TINY.vocab_size, (1, 6), generator=g)
    out = generate(moe, prompt, max_new_tokens=5)
    assert out.shape == (1, 11)
    ref = prompt.clone()
    with torch.no_grad():
        for _ in range(5):
            nxt = moe(ref)[:, -1].argmax(dim=-1)
            ref = torch.cat([ref, nxt[:, None]], dim=1)
    assert torch.equal(out, ref)


def test_prompt_lookup_identical_to_greedy(model):
    """Prompt-lookup speculative decode must emit EXACTLY the greedy
    sequence — speculation changes the forward count, never the
    output — on repetitive (draft-accepting) and random prompts."""
    from trainingjob_operator_amd.models.generate import generate_lookup
    rep = torch.tensor([[5, 6, 7, 5, 6, 7, 5, 6]])
    g = torch.Generator().manual_seed(9)
    rnd = torch.randint(0, 512, (1, 8), generator=g)
    for prompt in (rep, rnd):
        ref = generate(model, prompt, max_new_tokens=16)
        for k in (2, 4, 8):
            got = generate_lookup(model, prompt, max_new_tokens=16,
                                  lookup_k=k)
            assert torch.equal(ref, got), (k, ref, got)


def test_prompt_lookup_eos(model):
    from trainingjob_operator_amd.models.generate import generate_lookup
    rep = torch.tensor([[5, 6, 7, 5, 6, 7, 5, 6]])
    ref = generate(model, rep, max_new_tokens=16)
    eos = int(ref[0, 12])
    ref_e = generate(model, rep, max_new_tokens=16, eos_token=eos)
    got = generate_lookup(model, rep, max_new_tokens=16, lookup_k=4,
                          eos_token=eos)
    assert torch.equal(ref_e[0, :got.shape[1]], got[0])
    assert int(got[0, -1]) == eos
