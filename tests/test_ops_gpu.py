"""GPU numerics: each hand-written HIP kernel vs the plain PyTorch fp32
reference of the same op (repo rule; run with -m gpu on an MI355X box)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from trainingjob_operator_amd.ops import (  # noqa: E402
    apply_rope, fused_cross_entropy, fused_rmsnorm, make_inv_freq, swiglu,
)
from trainingjob_operator_amd.ops import native, reference  # noqa: E402

DEV = "cuda:0"
torch.manual_seed(0)


@pytest.fixture(scope="module", autouse=True)
def _require_native():
    native.load(require=True)  # fail loudly, never eager-fallback on GPU


def _mk(shape, dtype=torch.bfloat16, scale=1.0):
    return (torch.randn(*shape, dtype=torch.float32, device=DEV) * scale).to(dtype)


# relative-ish tolerance for bf16 compute vs fp32 reference
BF16_ATOL = 2e-2


def test_rmsnorm_fwd_matches_reference():
    for H in (2048, 4096, 8192):
        T = 257  # deliberately odd row count
        x = _mk((T, H))
        r = _mk((T, H))
        w = _mk((H,), scale=0.5)
        y, res = fused_rmsnorm(x, w, r, 1e-5)
        y_ref, res_ref, rrms_ref = reference.rmsnorm_fwd(
            x.float().cpu(), r.float().cpu(), w.float().cpu(), 1e-5)
        assert torch.allclose(res.float().cpu(),
                              (x.float() + r.float()).cpu(), atol=BF16_ATOL)
        assert torch.allclose(y.float().cpu(), y_ref, atol=BF16_ATOL,
                              rtol=1e-2), f"H={H}"


def test_rmsnorm_bwd_matches_reference():
    T, H = 512, 4096
    x = _mk((T, H)).requires_grad_()
    r = _mk((T, H)).requires_grad_()
    w = _mk((H,), scale=0.5).requires_grad_()
    y, res = fused_rmsnorm(x, w, r, 1e-5)
    dy = _mk((T, H))
    (y.float() * dy.float()).sum().backward()

    # fp32 CPU autograd reference
    xa = x.detach().float().cpu().requires_grad_()
    ra = r.detach().float().cpu().requires_grad_()
    wa = w.detach().float().cpu().requires_grad_()
    resa = xa + ra
    ya = resa * torch.rsqrt(resa.pow(2).mean(-1, keepdim=True) + 1e-5) * wa
    (ya * dy.float().cpu()).sum().backward()

    assert torch.allclose(x.grad.float().cpu(), xa.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(r.grad.float().cpu(), ra.grad, atol=5e-2, rtol=5e-2)
    # dw sums over 512 rows — scale tolerance with magnitude
    assert torch.allclose(w.grad.float().cpu(), wa.grad,
                          atol=wa.grad.abs().max() * 0.03 + 0.05, rtol=5e-2)


def test_rope_matches_reference():
    T, NH, D, S = 1024, 8, 128, 256
    inv_freq = make_inv_freq(D, 500000.0, device=DEV)
    x = _mk((T, NH, D))
    y = apply_rope(x, inv_freq, S)
    y_ref = reference.rope_rotate(x.float().cpu(), inv_freq.cpu(), S, 1.0)
    assert torch.allclose(y.float().cpu(), y_ref, atol=BF16_ATOL, rtol=1e-2)
    # inverse property on device
    back = apply_rope(y, inv_freq, S)  # not inverse; use grad for -sign
    dy = y.clone().detach()
    xg = x.clone().requires_grad_()
    out = apply_rope(xg, inv_freq, S)
    (out.float() * dy.float()).sum().backward()
    dx_ref = reference.rope_rotate(dy.float().cpu(), inv_freq.cpu(), S, -1.0)
    assert torch.allclose(xg.grad.float().cpu(), dx_ref, atol=BF16_ATOL,
                          rtol=1e-2)


def test_swiglu_matches_reference():
    N = 1 << 20
    g = _mk((N,), scale=2.0).requires_grad_()
    u = _mk((N,)).requires_grad_()
    out = swiglu(g, u)
    out_ref = reference.swiglu_fwd(g.detach().float().cpu(),
                                   u.detach().float().cpu())
    assert torch.allclose(out.float().cpu(), out_ref, atol=BF16_ATOL,
                          rtol=1e-2)
    dy = _mk((N,))
    (out.float() * dy.float()).sum().backward()
    dg_ref, du_ref = reference.swiglu_bwd(dy.float().cpu(),
                                          g.detach().float().cpu(),
                                          u.detach().float().cpu())
    assert torch.allclose(g.grad.float().cpu(), dg_ref, atol=5e-2, rtol=5e-2)
    assert torch.allclose(u.grad.float().cpu(), du_ref, atol=5e-2, rtol=5e-2)


def test_cross_entropy_matches_reference():
    T, V = 512, 128256
    logits = _mk((T, V), scale=3.0).requires_grad_()
    targets = torch.randint(0, V, (T,), device=DEV)
    targets[7] = -100
    loss_vec = fused_cross_entropy(logits, targets)
    n_valid = (targets != -100).sum()
    loss = loss_vec.sum() / n_valid
    loss.backward()

    la = logits.detach().float().cpu().requires_grad_()
    loss_ref = torch.nn.functional.cross_entropy(la, targets.cpu(),
                                                 ignore_index=-100)
    loss_ref.backward()
    assert abs(loss.item() - loss_ref.item()) < 2e-2
    assert loss_vec[7].item() == 0.0
    assert torch.allclose(logits.grad.float().cpu(), la.grad, atol=1e-4,
                          rtol=5e-2)


def test_adamw_matches_reference():
    n = 1 << 16
    g0 = torch.randn(n)
    p0 = torch.randn(n)
    # GPU fused
    from trainingjob_operator_amd.ops.native import load, stream_ptr
    lib = load()
    p32 = p0.clone().to(DEV)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    grad = g0.to(torch.bfloat16).to(DEV)
    pb = p0.to(torch.bfloat16).to(DEV)
    partials = torch.empty(2048, dtype=torch.float32, device=DEV)
    normsq = torch.empty(1, dtype=torch.float32, device=DEV)
    lr, b1, b2, eps, wd, clip, pre = 1e-3, 0.9, 0.95, 1e-8, 0.1, 1.0, 0.5
    lib.l2normsq(stream_ptr(), grad.data_ptr(), n, partials.data_ptr(), 2048,
                 normsq.data_ptr())
    lib.adamw_step(stream_ptr(), p32.data_ptr(), m.data_ptr(), v.data_ptr(),
                   grad.data_ptr(), pb.data_ptr(), normsq.data_ptr(), n,
                   lr, b1, b2, eps, wd, 1 - b1, 1 - b2, clip, pre, None)
    torch.cuda.synchronize()
    # CPU reference
    p32r = p0.clone()
    mr = torch.zeros(n)
    vr = torch.zeros(n)
    pbr = p0.to(torch.bfloat16)
    gr = g0.to(torch.bfloat16)
    nsq = gr.float().pow(2).sum()
    assert abs(normsq.item() - nsq.item()) / nsq.item() < 1e-4
    reference.adamw_step(p32r, mr, vr, gr, pbr, lr, b1, b2, eps, wd, 1,
                         clip, nsq, pre)
    assert torch.allclose(p32.cpu(), p32r, atol=1e-5, rtol=1e-4)
    assert torch.allclose(m.cpu(), mr, atol=1e-5)
    assert torch.allclose(v.cpu(), vr, atol=1e-6)


def test_smoke_training_step_gpu():
    from trainingjob_operator_amd.training import TrainConfig, Trainer
    cfg = TrainConfig(model="llama-smoke", micro_batch=1, grad_accum=2,
                      seq_len=512, lr=1e-3)
    trainer = Trainer(cfg)
    l0 = trainer.train_step().item()
    l1 = trainer.train_step().item()
    assert l0 == l0 and l1 == l1  # finite
    torch.cuda.synchronize()


def test_swiglu_packed_matches_reference():
    from trainingjob_operator_amd.ops import swiglu_packed
    T, F = 512, 14336
    gu = _mk((T, 2 * F), scale=2.0).requires_grad_()
    out = swiglu_packed(gu)
    g, u = gu.detach().float().cpu().split([F, F], dim=-1)
    ref = reference.swiglu_fwd(g.contiguous(), u.contiguous())
    assert torch.allclose(out.float().cpu(), ref, atol=BF16_ATOL, rtol=1e-2)
    dy = _mk((T, F))
    (out.float() * dy.float()).sum().backward()
    dg_ref, du_ref = reference.swiglu_bwd(dy.float().cpu(), g.contiguous(),
                                          u.contiguous())
    assert torch.allclose(gu.grad[:, :F].float().cpu(), dg_ref, atol=5e-2,
                          rtol=5e-2)
    assert torch.allclose(gu.grad[:, F:].float().cpu(), du_ref, atol=5e-2,
                          rtol=5e-2)


def test_rmsnorm_generic_small_h():
    """Round-1 SIGFPE regression: H outside the templated 2048*k set must
    run the generic kernel (not crash, not return garbage). Covers the
    moe-tiny/llama-tiny H=64 shape that killed GPUTEST_r01."""
    for H in (64, 256, 1024, 2560):
        T = 33
        x = _mk((T, H)).requires_grad_()
        r = _mk((T, H)).requires_grad_()
        w = _mk((H,), scale=0.5).requires_grad_()
        y, res = fused_rmsnorm(x, w, r, 1e-5)
        dy = _mk((T, H))
        (y.float() * dy.float()).sum().backward()

        xa = x.detach().float().cpu().requires_grad_()
        ra = r.detach().float().cpu().requires_grad_()
        wa = w.detach().float().cpu().requires_grad_()
        resa = xa + ra
        ya = resa * torch.rsqrt(resa.pow(2).mean(-1, keepdim=True) + 1e-5) * wa
        (ya * dy.float().cpu()).sum().backward()
        assert torch.allclose(y.float().cpu(), ya.detach(), atol=BF16_ATOL,
                              rtol=1e-2), f"H={H}"
        assert torch.allclose(x.grad.float().cpu(), xa.grad, atol=5e-2,
                              rtol=5e-2), f"H={H}"
        assert torch.allclose(w.grad.float().cpu(), wa.grad,
                              atol=wa.grad.abs().max() * 0.03 + 0.05,
                              rtol=5e-2), f"H={H}"


def test_launchers_reject_unsupported_shapes():
    """Launchers return nonzero for shapes they cannot run and Python
    raises — never a silent no-op with uninitialized outputs."""
    x = _mk((4, 60))  # H % 8 != 0
    w = _mk((60,))
    with pytest.raises(RuntimeError, match="unsupported shape"):
        fused_rmsnorm(x, w, None, 1e-5)
    lib = native.load()
    # direct launcher probes (no tensors touched on failure)
    assert lib.rmsnorm_fwd(None, None, None, None, None, None, None,
                           4, 60, 1e-5) != 0
    assert lib.ce_fwd(None, None, None, None, None, 4, 100, -100) != 0
    assert lib.rope(None, None, None, None, 4, 8, 16, 24, 1.0) != 0
    assert lib.attn_fwd(None, None, None, None, None, None,
                        0, 0, 0, 0, 0, 0, 0, 0, 0, 1, 4, 4, 100, 1.0) != 0


def test_rows_gather_scatter_and_moe_combine():
    """MoE dispatch kernels vs torch indexing (fwd + autograd bwd)."""
    from trainingjob_operator_amd.ops import gather_rows, moe_combine
    torch.manual_seed(11)
    N, H, T, K = 64, 256, 32, 2
    x = _mk((N, H)).requires_grad_()
    idx = torch.randint(0, N, (48,), device=DEV)   # duplicates
    out = gather_rows(x, idx, bijective=False)
    dy = _mk((48, H))
    (out.float() * dy.float()).sum().backward()
    xa = x.detach().float().cpu().requires_grad_()
    (xa[idx.cpu()] * dy.float().cpu()).sum().backward()
    assert torch.allclose(out.float().cpu(), xa.detach()[idx.cpu()],
                          atol=1e-2)
    assert torch.allclose(x.grad.float().cpu(), xa.grad, atol=2e-2,
                          rtol=2e-2)

    # bijective path
    x2 = _mk((N, H)).requires_grad_()
    perm = torch.randperm(N, device=DEV)
    out2 = gather_rows(x2, perm, bijective=True)
    dy2 = _mk((N, H))
    (out2.float() * dy2.float()).sum().backward()
    assert torch.allclose(x2.grad.float().cpu()[perm.cpu()],
                          dy2.float().cpu(), atol=1e-2)

    # fused combine
    src = _mk((T * K, H)).requires_grad_()
    inv = torch.randperm(T * K, device=DEV)
    gates = torch.rand(T, K, device=DEV, dtype=torch.float32) \
        .requires_grad_()
    y = moe_combine(src, inv, gates)
    gy = _mk((T, H))
    (y.float() * gy.float()).sum().backward()

    sa = src.detach().float().cpu().requires_grad_()
    ga = gates.detach().cpu().requires_grad_()
    pair = sa[inv.cpu()] * ga.reshape(-1, 1)
    ya = pair.reshape(T, K, H).sum(1)
    (ya * gy.float().cpu()).sum().backward()
    assert torch.allclose(y.float().cpu(), ya.detach(), atol=2e-2, rtol=1e-2)
    assert torch.allclose(src.grad.float().cpu(), sa.grad, atol=2e-2,
                          rtol=2e-2)
    assert torch.allclose(gates.grad.cpu(), ga.grad, atol=2e-1, rtol=2e-2)


def test_dispatch_rows_backward_no_atomics():
    """dispatch_rows: gather fwd; backward = gather-sum via the inverse
    pair permutation (vs index_add reference)."""
    from trainingjob_operator_amd.ops import dispatch_rows
    torch.manual_seed(13)
    T, K, H = 40, 2, 256
    x = _mk((T, H)).requires_grad_()
    flat_t = torch.arange(T, device=DEV).repeat_interleave(K)
    order = torch.randperm(T * K, device=DEV)
    inv = torch.empty_like(order)
    inv[order] = torch.arange(T * K, device=DEV)
    out = dispatch_rows(x, flat_t[order], inv, K)
    dy = _mk((T * K, H))
    (out.float() * dy.float()).sum().backward()
    xa = x.detach().float().cpu().requires_grad_()
    (xa[flat_t[order].cpu()] * dy.float().cpu()).sum().backward()
    assert torch.allclose(out.float().cpu(), xa.detach()[flat_t[order].cpu()],
                          atol=1e-2)
    assert torch.allclose(x.grad.float().cpu(), xa.grad, atol=2e-2,
                          rtol=2e-2)


def test_gemv_bf16_matches_fp32():
    """Decode GEMV (wave-per-row, hip/ops.hip gemv_bf16) vs an fp32
    matmul on the decode projection shapes (llama3-8b qkv/o/gate_up/down
    plus a small odd-M case)."""
    from trainingjob_operator_amd.ops import decode_linear
    torch.manual_seed(17)
    for M, K in [(6144, 4096), (4096, 4096), (1026, 512), (512, 14336)]:
        for N in (1, 3, 8):
            w = _mk((M, K))
            x = _mk((N, K))
            y = decode_linear(x, w)
            ref = x.float() @ w.float().t()
            err = (y.float() - ref).abs().max().item()
            tol = ref.abs().max().item() * 2e-2 + 2e-2
            assert err < tol, f"M={M} K={K} N={N}: {err} vs {tol}"


def test_gemv_bf16_fallback_shapes():
    """Unsupported shapes (K % 512 != 0, > 8 rows) fall back to the
    library GEMM and stay correct."""
    from trainingjob_operator_amd.ops import decode_linear
    torch.manual_seed(18)
    for M, K, N in [(256, 320, 2), (256, 512, 17)]:
        w = _mk((M, K))
        x = _mk((N, K))
        y = decode_linear(x, w)
        ref = x.float() @ w.float().t()
        assert (y.float() - ref).abs().max().item() < \
            ref.abs().max().item() * 2e-2 + 2e-2


def test_decode_attention_matches_fp32():
    """Fused flash-decoding (attn_decode: split-L partials + LSE combine)
    vs an fp32 masked-softmax reference, GQA and MHA, positions that
    land mid-chunk and chunk-aligned."""
    import math
    from trainingjob_operator_amd.ops import decode_attention
    torch.manual_seed(21)
    for B, H, n_kv, Lmax, pos in [(1, 32, 8, 640, 517), (2, 8, 8, 256, 255),
                                  (2, 16, 4, 384, 127), (1, 8, 1, 128, 0),
                                  (1, 8, 4, 200, 150)]:   # Lmax % 64 != 0
        D = 128
        q = _mk((B, H, 1, D))
        kc = _mk((B, n_kv, Lmax, D))
        vc = _mk((B, n_kv, Lmax, D))
        pos_t = torch.full((1,), pos, dtype=torch.int64, device=DEV)
        o = decode_attention(q, kc, vc, pos_t, 1.0 / math.sqrt(D))
        assert o is not None
        G = H // n_kv
        qg = q.float().reshape(B, n_kv, G, D)
        s = torch.einsum("bkgd,bksd->bkgs", qg, kc.float()) / math.sqrt(D)
        s[..., pos + 1:] = float("-inf")
        p = torch.softmax(s, dim=-1)
        ref = torch.einsum("bkgs,bksd->bkgd", p, vc.float())
        ref = ref.reshape(B, H, 1, D)
        err = (o.float() - ref).abs().max().item()
        assert err < 2e-2, f"B{B} H{H} kv{n_kv} L{Lmax} pos{pos}: {err}"


def test_rope_cache_matches_reference():
    """Fused decode rope+cache-write (rope_cache) vs the torch Neox
    rotation: q roped to a buffer, k roped into cache[pos], v copied."""
    from trainingjob_operator_amd.ops import make_inv_freq, native
    lib = native.load(require=True)
    torch.manual_seed(23)
    B, nh, nkv, Lmax, D, pos = 2, 8, 2, 96, 64, 37
    inv_freq = make_inv_freq(D, 10000.0, device=DEV)
    qkv = _mk((B, (nh + 2 * nkv) * D))
    kc = torch.zeros(B, nkv, Lmax, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    q = torch.empty(B, nh, D, dtype=torch.bfloat16, device=DEV)
    pos_t = torch.full((1,), pos, dtype=torch.int64, device=DEV)
    rc = lib.rope_cache(native.stream_ptr(), qkv.data_ptr(), q.data_ptr(),
                        kc.data_ptr(), vc.data_ptr(), inv_freq.data_ptr(),
                        pos_t.data_ptr(), B, nh, nkv, Lmax, D)
    native.check_rc(rc, "rope_cache", "test")
    torch.cuda.synchronize()

    def rope_ref(x):                       # [B, n, D] fp32 at position pos
        half = D // 2
        ang = pos * inv_freq.float()
        c, s = ang.cos(), ang.sin()
        x1, x2 = x[..., :half], x[..., half:]
        return torch.cat([x1 * c - x2 * s, x1 * s + x2 * c], dim=-1)

    f = qkv.float().reshape(B, nh + 2 * nkv, D)
    qr, kr, vr = f[:, :nh], f[:, nh:nh + nkv], f[:, nh + nkv:]
    assert (q.float() - rope_ref(qr)).abs().max().item() < 2e-2
    assert (kc[:, :, pos].float() - rope_ref(kr)).abs().max().item() < 2e-2
    assert (vc[:, :, pos].float() - vr).abs().max().item() < 1e-6
    assert kc[:, :, pos + 1].abs().max().item() == 0   # only row pos written


def test_gemv_swiglu_matches_fp32():
    """Fused decode GEMV+SwiGLU vs fp32 silu(x@Wg)*(x@Wu)."""
    from trainingjob_operator_amd.ops import decode_swiglu
    torch.manual_seed(29)
    for F, K, N in [(14336, 4096, 1), (512, 512, 3), (1024, 1024, 8)]:
        w = _mk((2 * F, K)) * 0.05
        x = _mk((N, K)) * 0.05
        y = decode_swiglu(x, w)
        assert y is not None
        g = x.float() @ w.float()[:F].t()
        u = x.float() @ w.float()[F:].t()
        ref = torch.nn.functional.silu(g) * u
        err = (y.float() - ref).abs().max().item()
        assert err < ref.abs().max().item() * 2e-2 + 2e-2, (F, K, N, err)


def test_moe_routed_decode_matches_grouped():
    """The sync-free routed decode path (pointer-table GEMVs) must match
    the grouped dispatch path on the same MoEMLP."""
    from trainingjob_operator_amd.parallel.ep import MoEMLP
    torch.manual_seed(31)
    m = MoEMLP(512, 512, n_experts=4, top_k=2).to(torch.bfloat16).to(DEV)
    x = _mk((3, 512)) * 0.1
    with torch.no_grad():
        fast = m(x)                        # T*k = 6 -> routed kernels
    grouped = m(x.requires_grad_())        # grad on -> grouped dispatch
    err = (fast.float() - grouped.float()).abs().max().item()
    assert err < 3e-2, err
