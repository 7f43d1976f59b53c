"""CPU numerics: the ops' hand-written forward/backward (reference.py, which is
also the math the HIP kernels implement) vs plain torch autograd in fp32."""
import torch

from trainingjob_operator_amd.ops import (
    apply_rope, fused_cross_entropy, fused_rmsnorm, make_inv_freq, swiglu,
)
from trainingjob_operator_amd.ops import reference

torch.manual_seed(0)


def test_rmsnorm_matches_autograd():
    T, H = 32, 256
    x = torch.randn(T, H, dtype=torch.float32)
    r = torch.randn(T, H, dtype=torch.float32)
    w = torch.randn(H, dtype=torch.float32)
    eps = 1e-5

    xa = x.clone().requires_grad_()
    ra = r.clone().requires_grad_()
    wa = w.clone().requires_grad_()
    res = xa + ra
    y_ref = res * torch.rsqrt(res.pow(2).mean(-1, keepdim=True) + eps) * wa
    dy = torch.randn_like(y_ref)
    dres_up = torch.randn_like(res)  # grad also arrives via the residual stream
    (y_ref * dy).sum().add_((res * dres_up).sum()).backward()

    xb = x.clone().requires_grad_()
    rb = r.clone().requires_grad_()
    wb = w.clone().requires_grad_()
    y, res_out = fused_rmsnorm(xb, wb, rb, eps)
    (y * dy).sum().add_((res_out * dres_up).sum()).backward()

    assert torch.allclose(y, y_ref, atol=1e-5)
    assert torch.allclose(xb.grad, xa.grad, atol=1e-4)
    assert torch.allclose(rb.grad, ra.grad, atol=1e-4)
    assert torch.allclose(wb.grad, wa.grad, atol=1e-3)


def test_rmsnorm_no_residual():
    T, H = 8, 64
    x = torch.randn(T, H, requires_grad=True)
    w = torch.randn(H, requires_grad=True)
    y, res_out = fused_rmsnorm(x, w, None, 1e-5)
    assert torch.equal(res_out, x.detach())
    y.sum().backward()
    assert x.grad is not None and w.grad is not None


def test_rope_inverse_and_autograd():
    T, NH, D = 16, 4, 64
    S = 8
    inv_freq = make_inv_freq(D, theta=10000.0)
    x = torch.randn(T, NH, D)
    fwd = reference.rope_rotate(x, inv_freq, S, 1.0)
    back = reference.rope_rotate(fwd, inv_freq, S, -1.0)
    assert torch.allclose(back, x, atol=1e-5)

    xa = x.clone().requires_grad_()
    y = apply_rope(xa, inv_freq, S)
    dy = torch.randn_like(y)
    (y * dy).sum().backward()
    # rotation is orthogonal: dx = R^T dy
    dx_ref = reference.rope_rotate(dy, inv_freq, S, -1.0)
    assert torch.allclose(xa.grad, dx_ref, atol=1e-5)


def test_rope_norm_preserved():
    T, NH, D = 8, 2, 32
    x = torch.randn(T, NH, D)
    y = reference.rope_rotate(x, make_inv_freq(D), T, 1.0)
    # pairwise rotation preserves L2 norm
    assert torch.allclose(y.norm(dim=-1), x.norm(dim=-1), atol=1e-4)


def test_swiglu_matches_autograd():
    N = 1024
    g = torch.randn(N, requires_grad=True)
    u = torch.randn(N, requires_grad=True)
    out_ref = torch.nn.functional.silu(g) * u
    dy = torch.randn_like(out_ref)
    (out_ref * dy).sum().backward()
    g2 = g.detach().clone().requires_grad_()
    u2 = u.detach().clone().requires_grad_()
    out = swiglu(g2, u2)
    (out * dy).sum().backward()
    assert torch.allclose(out, out_ref, atol=1e-5)
    assert torch.allclose(g2.grad, g.grad, atol=1e-5)
    assert torch.allclose(u2.grad, u.grad, atol=1e-5)


def test_cross_entropy_matches_torch():
    T, V = 64, 512
    logits = torch.randn(T, V)
    targets = torch.randint(0, V, (T,))
    targets[5] = -100
    targets[17] = -100

    la = logits.clone().requires_grad_()
    loss_ref = torch.nn.functional.cross_entropy(la, targets,
                                                 ignore_index=-100)
    loss_ref.backward()

    lb = logits.clone().requires_grad_()
    per_tok = fused_cross_entropy(lb, targets, ignore_index=-100)
    n_valid = (targets != -100).sum()
    loss = per_tok.sum() / n_valid
    loss.backward()

    assert torch.allclose(loss, loss_ref, atol=1e-5)
    assert torch.allclose(lb.grad, la.grad, atol=1e-5)
    assert per_tok[5] == 0 and per_tok[17] == 0


def test_adamw_matches_torch():
    n = 256
    p0 = torch.randn(n)
    p32 = p0.clone()
    m = torch.zeros(n)
    v = torch.zeros(n)
    pb = p0.to(torch.bfloat16)
    lr, b1, b2, eps, wd = 1e-3, 0.9, 0.95, 1e-8, 0.1

    pt = p0.clone().requires_grad_()
    opt = torch.optim.AdamW([pt], lr=lr, betas=(b1, b2), eps=eps,
                            weight_decay=wd)
    for step in range(1, 4):
        g = torch.randn(n, generator=torch.Generator().manual_seed(step))
        pt.grad = g.clone()
        opt.step()
        reference.adamw_step(p32, m, v, g.to(torch.bfloat16), pb, lr, b1, b2,
                             eps, wd, step)
    # bf16 grads introduce small drift vs fp32 grads — loose tolerance
    assert torch.allclose(p32, pt.detach(), atol=5e-3)


def test_swiglu_packed_matches_unpacked():
    from trainingjob_operator_amd.ops import swiglu_packed
    T, F = 8, 64
    gu = torch.randn(T, 2 * F, requires_grad=True)
    out = swiglu_packed(gu)
    g, u = gu.detach().split([F, F], dim=-1)
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(out, ref, atol=1e-5)
    dy = torch.randn_like(out)
    (out * dy).sum().backward()
    ga = g.clone().requires_grad_()
    ua = u.clone().requires_grad_()
    ((torch.nn.functional.silu(ga) * ua) * dy).sum().backward()
    assert torch.allclose(gu.grad[:, :F], ga.grad, atol=1e-5)
    assert torch.allclose(gu.grad[:, F:], ua.grad, atol=1e-5)
