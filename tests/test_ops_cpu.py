"""CPU numerics tests: the reference op implementations vs torch autograd.

These pin down the math that the gfx950 HIP kernels are later tested
against (tests/test_kernels_gpu.py compares HIP vs these in fp32), mirroring
the reference's oracle pattern: parallel/custom impl vs stock serial impl
(alpa/testing.py:28 assert_allclose)."""
import math

import pytest
import torch
import torch.nn.functional as F

from alpa_amd.ops import reference as ref


def test_layer_norm_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(8, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    b = torch.randn(64, requires_grad=True)
    y, mean, rstd = ref.layer_norm_fwd(x, w, b, 1e-5)
    y_ref = F.layer_norm(x, (64,), w, b, 1e-5)
    torch.testing.assert_close(y, y_ref, rtol=1e-5, atol=1e-5)

    dy = torch.randn_like(y)
    y_ref.backward(dy)
    dx, dw, db = ref.layer_norm_bwd(dy, x.detach(), w.detach(), mean, rstd)
    torch.testing.assert_close(dx, x.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dw, w.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(db, b.grad, rtol=1e-4, atol=1e-4)


def test_bias_gelu_matches_autograd():
    torch.manual_seed(1)
    x = torch.randn(16, 32, dtype=torch.float64, requires_grad=True)
    b = torch.randn(32, dtype=torch.float64, requires_grad=True)
    y = ref.bias_gelu_fwd(x, b)
    y_ag = ref.gelu(x + b)
    y_ag.sum().backward()
    dy = torch.ones_like(y)
    dx, db = ref.bias_gelu_bwd(dy, x.detach(), b.detach())
    torch.testing.assert_close(dx, x.grad, rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(db, b.grad, rtol=1e-6, atol=1e-6)


@pytest.mark.parametrize("causal", [True, False])
def test_attention_matches_sdpa(causal):
    torch.manual_seed(2)
    B, H, S, D = 2, 3, 17, 8
    q = torch.randn(B, H, S, D, requires_grad=True)
    k = torch.randn(B, H, S, D, requires_grad=True)
    v = torch.randn(B, H, S, D, requires_grad=True)
    o, lse = ref.attention_fwd(q, k, v, causal=causal)
    o_ref = F.scaled_dot_product_attention(q, k, v, is_causal=causal)
    torch.testing.assert_close(o, o_ref, rtol=1e-4, atol=1e-4)

    do = torch.randn_like(o)
    o_ref.backward(do)
    dq, dk, dv = ref.attention_bwd(do, q.detach(), k.detach(), v.detach(),
                                   o.detach(), lse, causal=causal)
    torch.testing.assert_close(dq, q.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dk, k.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dv, v.grad, rtol=1e-4, atol=1e-4)


def test_cross_entropy_matches_torch():
    torch.manual_seed(3)
    N, V = 32, 100
    logits = torch.randn(N, V, requires_grad=True)
    targets = torch.randint(0, V, (N,))
    loss, lse = ref.softmax_cross_entropy_fwd(logits, targets)
    loss_ref = F.cross_entropy(logits, targets, reduction="none")
    torch.testing.assert_close(loss, loss_ref, rtol=1e-5, atol=1e-5)

    loss_ref.mean().backward()
    dloss = torch.full((N,), 1.0 / N)
    dlogits = ref.softmax_cross_entropy_bwd(dloss, logits.detach(), targets,
                                            lse)
    torch.testing.assert_close(dlogits, logits.grad, rtol=1e-5, atol=1e-5)


def test_adamw_matches_torch():
    torch.manual_seed(4)
    shapes = [(10, 4), (7,), (3, 3, 3)]
    params = [torch.randn(s) for s in shapes]
    grads = [torch.randn(s) for s in shapes]
    params_t = [p.clone().requires_grad_(True) for p in params]
    opt = torch.optim.AdamW(params_t, lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                            weight_decay=0.1)
    for p, g in zip(params_t, grads):
        p.grad = g.clone()
    opt.step()
    opt.step()  # second step exercises bias correction

    ms = [torch.zeros_like(p) for p in params]
    vs = [torch.zeros_like(p) for p in params]
    for step in (1, 2):
        ref.adamw_step(params, grads, ms, vs, step, lr=1e-2, beta1=0.9,
                       beta2=0.95, eps=1e-8, weight_decay=0.1)
    for p, pt in zip(params, params_t):
        torch.testing.assert_close(p, pt.detach(), rtol=1e-5, atol=1e-6)


def test_adamw_grad_scale():
    """grad_scale folds the microbatch/dp division into the update."""
    torch.manual_seed(5)
    p1 = [torch.randn(5)]
    p2 = [p1[0].clone()]
    g = torch.randn(5)
    st = lambda: ([torch.zeros(5)], [torch.zeros(5)])
    m1, v1 = st()
    m2, v2 = st()
    ref.adamw_step(p1, [g * 0.25], m1, v1, 1, 1e-3, 0.9, 0.95, 1e-8, 0.0)
    ref.adamw_step(p2, [g], m2, v2, 1, 1e-3, 0.9, 0.95, 1e-8, 0.0,
                   grad_scale=0.25)
    torch.testing.assert_close(p1[0], p2[0])


def test_rms_norm_reference():
    """RMS-norm fp32 reference (the oracle for a future rms_norm HIP
    kernel — LLaMA-family models): fwd matches the closed form, bwd
    matches autograd."""
    import torch
    from alpa_amd.ops import reference as ref
    torch.manual_seed(5)
    x = torch.randn(8, 32, requires_grad=True)
    w = torch.randn(32, requires_grad=True)
    y, rstd = ref.rms_norm_fwd(x, w, 1e-6)
    ref_y = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * w
    torch.testing.assert_close(y, ref_y, rtol=1e-6, atol=1e-6)
    dy = torch.randn_like(y)
    ref_y.backward(dy)
    dx, dw = ref.rms_norm_bwd(dy, x.detach(), w.detach(), rstd)
    torch.testing.assert_close(dx, x.grad, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(dw, w.grad, rtol=1e-5, atol=1e-5)
